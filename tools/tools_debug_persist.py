"""Debug probe for resblock_pair_persist_kernel: structured inputs,
mismatch-coordinate classification vs the fp32 oracle."""
import os
import sys

import torch

sys.path.insert(0, '.')
from sonata_amd.ops.functional import resblock_pair_cl  # noqa: E402

dev = "cuda:0"


def probe(C, k, dil, B, T, mode):
    torch.manual_seed(0)
    if mode == "ramp":
        t_ramp = (torch.arange(T) % 97).float() / 97.0
        c_ramp = torch.arange(C).float() / C
        x = (t_ramp[None, :, None] + 0.01 * c_ramp[None, None, :]
             ).expand(B, T, C).contiguous() - 0.5
        w1 = torch.zeros(C, C, k)
        w1[torch.arange(C), torch.arange(C), k // 2] = 1.0  # identity tap
        w2 = w1.clone()
        b1 = torch.zeros(C)
        b2 = torch.zeros(C)
    else:
        x = torch.randn(B, T, C) / 4
        w1 = torch.randn(C, C, k) / (C * k) ** 0.5
        w2 = torch.randn(C, C, k) / (C * k) ** 0.5
        b1 = torch.randn(C) / 10
        b2 = torch.randn(C) / 10
    got = resblock_pair_cl(x.to(dev, torch.bfloat16),
                           w1.to(dev, torch.bfloat16), b1.to(dev),
                           w2.to(dev, torch.bfloat16), b2.to(dev),
                           dilation=dil)
    ref = resblock_pair_cl(x, w1, b1, w2, b2, dilation=dil)
    err = (got.float().cpu() - ref).abs()
    tol = 0.02 * max(ref.abs().max().item(), 1.0)
    bad = (err > tol)
    BM = 256 - (k - 1)
    print(f"C={C} k={k} d={dil} B={B} T={T} [{mode}]: "
          f"{int(bad.sum())} bad of {bad.numel()}  max_err={err.max():.4f}")
    if bad.any():
        idx = bad.nonzero()[:2000]
        ts = idx[:, 1]
        tiles = torch.unique(ts // BM)
        offs = torch.unique(ts % BM)
        cs = torch.unique(idx[:, 2])
        print(f"  bad tiles (t//{BM}): {tiles[:12].tolist()}"
              f"{'...' if len(tiles) > 12 else ''} ({len(tiles)} tiles)")
        print(f"  bad offsets in tile: {offs[:16].tolist()}"
              f"{'...' if len(offs) > 16 else ''} ({len(offs)} offsets)")
        print(f"  bad channels: {cs[:16].tolist()} ({len(cs)})")
        t0 = int(idx[0, 1])
        print(f"  sample t={t0}: got={got[0, t0, :6].float().tolist()}")
        print(f"             ref={ref[0, t0, :6].tolist()}")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "ramp"):
        probe(32, 3, 1, 1, 65024, "ramp")
    if which in ("all", "rand"):
        probe(32, 3, 1, 1, 65024, "rand")
        probe(64, 3, 1, 1, 65024, "rand")
        probe(32, 11, 5, 1, 63222, "rand")
