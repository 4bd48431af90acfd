#!/usr/bin/env python3
"""Flagship benchmark: Piper/VITS (en_US-lessac-medium class) synthesis
throughput on MI355X.

Metric (BASELINE.json): RTF + audio-seconds/sec for Piper
en_US-lessac-medium at 1/2/4/8 MI355X — data-parallel utterance batching,
one rank per GPU over RCCL, weak scaling (fixed per-GPU batch).

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1 is launched via torch.distributed.run, one rank per GPU.

A "step" = one full synthesis of a fixed batch of synthetic phoneme
sequences (text encoder + duration predictor + flow + HiFi-GAN decode)
with random-init weights (no network for real checkpoints — data:
synthetic).  Output value = WHOLE-JOB audio seconds synthesized per
wall second, summed over ranks; ms_per_step is the max over ranks.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from sonata_amd.models.config import ModelConfig, VitsArchitecture, QUALITY_PRESETS
from sonata_amd.models.vits import VitsModel
from sonata_amd.text.ids import default_phoneme_id_map, num_symbols


def make_batch(batch: int, seq_len: int, n_vocab: int, device, seed: int):
    """Synthetic phoneme-id batch shaped like real Piper input: BOS/EOS
    wrapped, PAD-interleaved ids (every odd position is PAD=0)."""
    g = torch.Generator().manual_seed(seed)
    ids = torch.zeros(batch, seq_len, dtype=torch.long)
    ids[:, 1:-1:2] = torch.randint(3, n_vocab, (batch, (seq_len - 2 + 1) // 2),
                                   generator=g)
    ids[:, 0] = 1  # BOS
    ids[:, -1] = 2  # EOS
    lengths = torch.full((batch,), seq_len, dtype=torch.long)
    return ids.to(device), lengths.to(device)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=64,
                    help="utterances per rank per step (64/rank makes the "
                         "8-GPU run exactly the 512-concurrent-utterance "
                         "serving config of BASELINE.json)")
    ap.add_argument("--seq-len", type=int, default=256,
                    help="phoneme ids per utterance (PAD-interleaved)")
    ap.add_argument("--quality", default="medium")
    ap.add_argument("--device", default=None)
    ap.add_argument("--engine", choices=["cpp", "python"], default="cpp",
                    help="serving runtime: the C++ VitsEngine (default) or "
                         "the Python model path (same kernels either way)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
        use_gpu = device.type == "cuda"
    else:
        # modulo device count so world>n_gpus rehearsal runs share GPUs
        # (no-op on a real N-GPU node where local_rank < count)
        dev_i = local_rank % max(torch.cuda.device_count(), 1)
        device = torch.device(f"cuda:{dev_i}" if use_gpu else "cpu")
    if use_gpu:
        torch.cuda.set_device(device)

    dist = None
    backend = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        # SONATA_BENCH_BACKEND=gloo lets world>1 rehearse on a 1-GPU box
        # (both ranks compute on cuda:0, collectives go over gloo/host)
        backend = os.environ.get(
            "SONATA_BENCH_BACKEND", "nccl" if use_gpu else "gloo")
        dist.init_process_group(backend=backend)

    # ---- build the flagship voice (random init, medium preset) ---------- #
    preset = QUALITY_PRESETS[args.quality]
    arch = VitsArchitecture(**preset["arch"])
    config = ModelConfig(
        key="en_US-lessac-" + args.quality,
        language_code="en-us",
        sample_rate=preset["sample_rate"],
        quality=args.quality,
        architecture=arch,
    )
    torch.manual_seed(0)  # same weights on every rank
    id_map = default_phoneme_id_map()
    nv = num_symbols(id_map)
    ids, lengths = make_batch(args.batch, args.seq_len, nv, device,
                              seed=1234 + rank)
    sample_rate = config.sample_rate
    dtype = torch.bfloat16 if use_gpu else torch.float32

    engine = None
    if args.engine == "cpp":
        try:
            import tempfile

            from sonata_amd.models import create_random_voice
            from sonata_amd.ops import hip_ext

            ext = hip_ext(required=False)
            if ext is not None and hasattr(ext, "VitsEngine"):
                tmp = tempfile.mkdtemp(prefix=f"bench_voice_r{rank}_")
                pack = create_random_voice(
                    tmp, "bench", quality=args.quality, seed=0)
                engine = ext.VitsEngine(
                    pack, str(device), "bf16" if use_gpu else "f32")
        except Exception as e:  # noqa: BLE001 - fall back to python path
            print(f"# engine=cpp unavailable ({e}); using python path",
                  flush=True)
            engine = None

    if engine is None:
        net = VitsModel(config.num_symbols, arch, n_speakers=1).eval()
        net = net.to(device=device, dtype=dtype)

    seeds = [1234 * (rank + 1) + i for i in range(args.batch)]

    def one_step():
        if engine is not None:
            audio, audio_lengths = engine.infer(
                ids, lengths, None, 0.667, 1.0, 0.8, seeds)
        else:
            with torch.no_grad():
                audio, audio_lengths = net.infer(ids, lengths)
        return float(audio_lengths.sum().item()) / sample_rate

    # ---- warmup --------------------------------------------------------- #
    audio_sec_per_step = 0.0
    for _ in range(max(args.warmup, 1)):
        audio_sec_per_step = one_step()
    if use_gpu:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()

    # ---- timed region --------------------------------------------------- #
    t0 = time.perf_counter()
    total_audio_sec = 0.0
    for _ in range(args.steps):
        total_audio_sec += one_step()
    if use_gpu:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max elapsed over ranks; sum audio over ranks
    if dist:
        comm_dev = device if (use_gpu and backend == "nccl") else "cpu"
        t = torch.tensor([elapsed, total_audio_sec], dtype=torch.float64,
                         device=comm_dev)
        gathered = [torch.zeros_like(t) for _ in range(world)]
        dist.all_gather(gathered, t)
        elapsed = max(float(g[0].item()) for g in gathered)
        total_audio_sec = sum(float(g[1].item()) for g in gathered)

    value = total_audio_sec / elapsed
    ms_per_step = elapsed * 1000.0 / args.steps
    rtf = (elapsed / args.steps) / (total_audio_sec / world / args.steps) \
        if total_audio_sec > 0 else 0.0

    if rank == 0:
        print(json.dumps({
            "metric": "audio_seconds_per_second",
            "value": round(value, 3),
            "unit": "audio_sec/s",
            "n_gpus": world if use_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": "en_US-lessac-" + args.quality + " (VITS, random-init)",
                "global_batch": args.batch * world,
                "seq_len": args.seq_len,
                "parallelism": f"dp{world}",
                "engine": "cpp" if engine is not None else "python",
                "sample_rate": sample_rate,
                "per_rank_rtf": round(rtf, 5),
            },
        }))

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
