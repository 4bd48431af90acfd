"""Adaptive latent-frame chunker for streaming HiFi-GAN decode.

Parity: reference AdaptiveMelChunker (crates/sonata/models/piper/src/
lib.rs:860-913): chunk size grows each step (chunk_size * step), capped at
MAX_CHUNK_SIZE=1024 frames, floor MIN_CHUNK_SIZE=44; interior chunks carry
2*chunk_padding overlap frames and the decoded audio trims
chunk_padding*hop samples per side (:891-911); a final chunk smaller than
MIN_CHUNK_SIZE is absorbed into the previous one (:898-906); one-shot
fallback when num_frames <= chunk_size*2 + chunk_padding*2 (:785,848-853).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Iterator, Optional, Tuple

MIN_CHUNK_SIZE = 44
MAX_CHUNK_SIZE = 1024


@dataclass(frozen=True)
class ChunkSpec:
    """One decoder call: frames [mel_start, mel_end) of z, with the decoded
    audio sliced to [audio_start, audio_end) relative to the chunk's own
    output (padding trim)."""

    mel_start: int
    mel_end: int
    trim_left_frames: int
    trim_right_frames: int
    is_last: bool


def chunk_plan(
    num_frames: int, chunk_size: int, chunk_padding: int
) -> Iterator[ChunkSpec]:
    """Yield decoder chunk specs covering `num_frames` latent frames."""
    if num_frames <= 0:
        return
    chunk_size = max(int(chunk_size), 1)
    chunk_padding = max(int(chunk_padding), 0)
    if num_frames <= chunk_size * 2 + chunk_padding * 2:
        yield ChunkSpec(0, num_frames, 0, 0, True)
        return

    start = 0
    step = 1
    while start < num_frames:
        size = min(chunk_size * step, MAX_CHUNK_SIZE)
        step += 1
        end = min(start + size, num_frames)
        # absorb a too-small tail into this chunk
        if num_frames - end < MIN_CHUNK_SIZE:
            end = num_frames
        pad_l = min(chunk_padding, start)
        pad_r = min(chunk_padding, num_frames - end)
        is_last = end >= num_frames
        yield ChunkSpec(start - pad_l, end + pad_r, pad_l, pad_r, is_last)
        if is_last:
            return
        start = end
