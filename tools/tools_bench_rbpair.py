"""Microbench: resblock pair kernel, persistent vs generic, on the real
bench decode shapes.  SONATA_PERSIST_RB gates the persistent path (read
once per process, so A/B runs this script twice)."""
import os
import sys
import time

import torch

sys.path.insert(0, '.')
from sonata_amd.ops.functional import resblock_pair_cl  # noqa: E402

B = 64
F = 600  # ~frames per utterance in the flagship bench
dev = "cuda:0"
# (C, k, dil, T)
shapes = [(32, 3, 1, 256 * F), (32, 3, 3, 256 * F), (32, 7, 3, 256 * F),
          (32, 11, 5, 256 * F), (64, 3, 1, 128 * F), (64, 3, 5, 128 * F),
          (64, 7, 3, 128 * F), (128, 3, 1, 64 * F), (128, 7, 3, 64 * F),
          (128, 11, 5, 64 * F), (256, 3, 1, 8 * F), (256, 7, 3, 8 * F),
          (256, 11, 5, 8 * F)]
print(f"# persist={os.environ.get('SONATA_PERSIST_RB')} geom={os.environ.get('SONATA_RB_GEOM')}")
for C, k, dil, T in shapes:
    x = (torch.randn(B // 8, T, C) / 4).to(torch.bfloat16).to(dev)
    # B//8 keeps memory sane; per-shape time scales linearly in B
    w1 = (torch.randn(C, C, k) / (C * k) ** 0.5).to(torch.bfloat16).to(dev)
    w2 = (torch.randn(C, C, k) / (C * k) ** 0.5).to(torch.bfloat16).to(dev)
    b1 = (torch.randn(C) / 10).to(dev)
    b2 = (torch.randn(C) / 10).to(dev)
    for _ in range(3):
        y = resblock_pair_cl(x, w1, b1, w2, b2, dilation=dil)
    torch.cuda.synchronize()
    N = 10
    t0 = time.perf_counter()
    for _ in range(N):
        y = resblock_pair_cl(x, w1, b1, w2, b2, dilation=dil)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / N
    flops = 2 * 2 * x.shape[0] * T * C * C * k  # two convs
    gbytes = 2 * x.shape[0] * T * C * 3  # read x + write + residual read
    print(f"C={C:3d} k={k:2d} d={dil} T={T:7d}: {dt*1e3:7.3f} ms  "
          f"{flops/dt/1e12:6.1f} TF  {gbytes/dt/1e9:7.0f} GB/s")
