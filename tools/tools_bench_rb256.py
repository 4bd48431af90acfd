"""C=256 pair shapes with interleaved repeats (median + min + max) to
screen the round-1 'rare 7x outlier' from real geometry effects."""
import os, statistics, sys, time
import torch
sys.path.insert(0, '.')
from sonata_amd.ops.functional import resblock_pair_cl

dev = "cuda:0"
B, F = 8, 600
shapes = [(256, 3, 1), (256, 7, 3), (256, 11, 5), (128, 3, 1), (128, 11, 5)]
print(f"# geom={os.environ.get('SONATA_RB_GEOM')}")
for C, k, dil in shapes:
    T = (8 if C == 256 else 64) * F
    x = (torch.randn(B, T, C) / 4).to(torch.bfloat16).to(dev)
    w1 = (torch.randn(C, C, k) / (C * k) ** 0.5).to(torch.bfloat16).to(dev)
    w2 = (torch.randn(C, C, k) / (C * k) ** 0.5).to(torch.bfloat16).to(dev)
    b1 = (torch.randn(C) / 10).to(dev)
    b2 = (torch.randn(C) / 10).to(dev)
    for _ in range(3):
        resblock_pair_cl(x, w1, b1, w2, b2, dilation=dil)
    torch.cuda.synchronize()
    ts = []
    for _ in range(15):
        t0 = time.perf_counter()
        for _ in range(4):
            resblock_pair_cl(x, w1, b1, w2, b2, dilation=dil)
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) / 4)
    med = statistics.median(ts)
    flops = 2 * 2 * B * T * C * C * k
    print(f"C={C:3d} k={k:2d} d={dil}: med {med*1e3:7.3f} ms "
          f"(min {min(ts)*1e3:.3f} max {max(ts)*1e3:.3f})  "
          f"{flops/med/1e12:6.1f} TF")
