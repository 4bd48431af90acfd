"""Data-parallel utterance serving primitives.

Design notes (MI355X/xGMI-first):
  * Weight broadcast happens ONCE per voice load as a handful of large
    flat buckets (not per-tensor): xGMI is point-to-point (7 links x
    ~153 GB/s), so few big transfers beat many small ones.
  * Waveform gather to rank 0 uses ONE all_gather of a packed f32 buffer
    per round (utterances concatenated, lengths exchanged first) instead
    of per-utterance sends — RCCL latency dominates small messages
    (SURVEY.md §7 hard part 6: chunk aggregation before all-gather).
  * Scheduling is static round-robin by utterance index: deterministic
    (same text -> same audio on any rank thanks to per-utterance seeds,
    models/voice.py) and needs no control-plane traffic.
"""

from __future__ import annotations

import os
from typing import List, Optional, Sequence

import numpy as np
import torch
import torch.distributed as dist

BUCKET_BYTES = 64 << 20  # broadcast bucket size


def init_distributed(backend: Optional[str] = None) -> tuple:
    """Initialize torch.distributed from torchrun env vars; no-op for
    world_size 1.  Returns (rank, world_size, local_rank)."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)
    return rank, world, local_rank


def broadcast_module(module: torch.nn.Module, src: int = 0) -> None:
    """Broadcast all parameters+buffers of `module` from rank `src` in
    large flat buckets (one RCCL broadcast per ~64 MB)."""
    if not (dist.is_available() and dist.is_initialized()):
        return
    tensors = [t.data for t in module.parameters()]
    tensors += list(module.buffers())
    # Group buckets by dtype and broadcast each in its NATIVE dtype:
    # casting through float32 would silently corrupt int64 buffers with
    # values > 2^24 and fp64 tensors.  Iterate dtypes in a
    # deterministic order so all ranks issue identical collectives.
    by_dtype: dict = {}
    for t in tensors:
        by_dtype.setdefault(t.dtype, []).append(t)
    for dtype in sorted(by_dtype, key=str):
        bucket: List[torch.Tensor] = []
        nbytes = 0
        for t in by_dtype[dtype]:
            bucket.append(t)
            nbytes += t.numel() * t.element_size()
            if nbytes >= BUCKET_BYTES:
                _broadcast_bucket(bucket, src)
                bucket, nbytes = [], 0
        if bucket:
            _broadcast_bucket(bucket, src)


def _broadcast_bucket(bucket: List[torch.Tensor], src: int) -> None:
    flat = torch.cat([t.reshape(-1) for t in bucket])
    dist.broadcast(flat, src=src)
    off = 0
    for t in bucket:
        n = t.numel()
        t.copy_(flat[off:off + n].reshape(t.shape))
        off += n


def shard_round_robin(n_items: int, rank: int, world: int) -> List[int]:
    """Indices this rank owns (static schedule, no communication)."""
    return list(range(rank, n_items, world))


def comm_device(compute_device: torch.device) -> torch.device:
    """Device collective tensors must live on for the active backend:
    nccl/RCCL wants them on the GPU (xGMI path), gloo wants host memory
    even when the model computes on cuda (GPU-box rehearsal of world>1
    with gloo collectives)."""
    if not (dist.is_available() and dist.is_initialized()):
        return compute_device
    if dist.get_backend() == "nccl" and compute_device.type == "cuda":
        return compute_device
    return torch.device("cpu")


def gather_audio_to_rank0(
    pieces: Sequence[np.ndarray],
    indices: Sequence[int],
    n_total: int,
    device: torch.device,
) -> Optional[List[np.ndarray]]:
    """All ranks call with their local (audio, original-index) results;
    rank 0 returns the full list ordered by original index, others None.

    One all_gather round: first the per-rank packed sizes + headers, then
    the packed sample payloads (padded to the max packed size)."""
    if not (dist.is_available() and dist.is_initialized()):
        out: List[Optional[np.ndarray]] = [None] * n_total
        for i, p in zip(indices, pieces):
            out[i] = p
        return [p for p in out if p is not None]

    device = comm_device(device)
    rank = dist.get_rank()
    world = dist.get_world_size()
    # header: [count, (index, length) * count]
    header = [len(pieces)]
    for i, p in zip(indices, pieces):
        header += [int(i), int(len(p))]
    hmax = 1 + 2 * ((n_total + world - 1) // world + 1)
    h = torch.zeros(hmax, dtype=torch.long, device=device)
    h[: len(header)] = torch.tensor(header, dtype=torch.long)
    hs = [torch.zeros_like(h) for _ in range(world)]
    dist.all_gather(hs, h)

    packed = (np.concatenate(pieces) if pieces
              else np.zeros(0, dtype=np.float32)).astype(np.float32)
    sizes = [int(x[1 : 1 + 2 * int(x[0])][1::2].sum().item()) for x in hs]
    pmax = max(max(sizes), 1)
    buf = torch.zeros(pmax, dtype=torch.float32, device=device)
    if len(packed):
        buf[: len(packed)] = torch.from_numpy(packed).to(device)
    bufs = [torch.zeros_like(buf) for _ in range(world)]
    dist.all_gather(bufs, buf)

    if rank != 0:
        return None
    out: List[Optional[np.ndarray]] = [None] * n_total
    for r in range(world):
        hr = hs[r].cpu()
        cnt = int(hr[0])
        data = bufs[r].cpu().numpy()
        off = 0
        for j in range(cnt):
            idx = int(hr[1 + 2 * j])
            ln = int(hr[2 + 2 * j])
            out[idx] = data[off:off + ln]
            off += ln
    return [p if p is not None else np.zeros(0, dtype=np.float32)
            for p in out]


class DistributedSynthesizer:
    """Serve a corpus of utterances across N GPUs: shard round-robin,
    synthesize in padded batches per rank, gather waveforms to rank 0.

    This is the engine behind baseline config #4 (8xMI355X, 512
    concurrent utterances)."""

    def __init__(self, voice, batch_size: int = 64):
        self.voice = voice
        self.batch_size = batch_size
        self.rank, self.world, self.local_rank = (
            dist.get_rank() if dist.is_initialized() else 0,
            dist.get_world_size() if dist.is_initialized() else 1,
            int(os.environ.get("LOCAL_RANK", "0")),
        )

    def synthesize_corpus(
        self, phonemes_list: Sequence[str]
    ) -> Optional[List[np.ndarray]]:
        """phonemes -> waveforms, ordered; rank 0 gets results."""
        n = len(phonemes_list)
        mine = shard_round_robin(n, self.rank, self.world)
        local: List[np.ndarray] = []
        for i in range(0, len(mine), self.batch_size):
            idx = mine[i : i + self.batch_size]
            audios = self.voice.speak_batch([phonemes_list[j] for j in idx])
            local.extend(a.samples for a in audios)
        return gather_audio_to_rank0(
            local, mine, n, self.voice.device
            if self.voice.device.type == "cuda" else torch.device("cpu"))
