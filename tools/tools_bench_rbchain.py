"""A/B: whole-resblock chain kernel vs the 3-pair loop, real decode
shapes.  SONATA_RB_CHAIN gates the python dispatch per call."""
import os
import statistics
import sys
import time

import torch

sys.path.insert(0, '.')
from sonata_amd.models.vits import ResBlock1  # noqa: E402

dev = "cuda:0"
B, F = 8, 600
shapes = [(32, 3), (32, 7), (32, 11), (64, 3)]
for C, k in shapes:
    T = (256 if C == 32 else 128) * F
    torch.manual_seed(C + k)
    rb = ResBlock1(C, k, [1, 3, 5]).to(dev, torch.bfloat16)
    x = (torch.randn(B, T, C) / 4).to(torch.bfloat16).to(dev)
    for mode in ("0", "1"):
        os.environ["SONATA_RB_CHAIN"] = mode
        for _ in range(3):
            rb.forward_cl(x)
        torch.cuda.synchronize()
        ts = []
        for _ in range(11):
            t0 = time.perf_counter()
            for _ in range(3):
                rb.forward_cl(x)
            torch.cuda.synchronize()
            ts.append((time.perf_counter() - t0) / 3)
        med = statistics.median(ts)
        flops = 3 * 2 * 2 * B * T * C * C * k
        gb = 2 * B * T * C * (3 * 3 if mode == "0" else 3)
        print(f"C={C:3d} k={k:2d} chain={mode}: med {med*1e3:7.3f} ms "
              f"(min {min(ts)*1e3:.3f} max {max(ts)*1e3:.3f}) "
              f"{flops/med/1e12:6.1f} TF  {gb/med/1e9:6.0f} GB/s")
