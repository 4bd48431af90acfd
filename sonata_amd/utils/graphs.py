"""hipGraph caching for the streaming decoder.

The realtime path decodes small chunks: at B=1 a chunk is ~150 kernel
launches of a few µs each, so LAUNCH overhead dominates.  hipGraphs
(torch.cuda.CUDAGraph on ROCm) replay the whole chunk as one submission.
The adaptive chunker emits sizes from a small deterministic set
(chunk·step capped at 1024), so a per-(B, C, F) graph cache hits on
every steady-state chunk; unseen shapes fall back to eager.

Opt-in via SONATA_HIPGRAPH=1 (or graph_decode(..., enabled=True)):
graph capture pins input/output buffers per shape, costing HBM per
cached shape — negligible against 288 GB.
"""

from __future__ import annotations

import os
from typing import Callable, Dict, Tuple

import torch


def enabled() -> bool:
    """Decode-chunk graph replay: measured NEUTRAL (chunk decode is
    GPU-time-bound, not launch-bound) — stays opt-in."""
    return os.environ.get("SONATA_HIPGRAPH", "0") not in ("0", "", "false")


def phase1_enabled() -> bool:
    """Encoder phase-1 graph replay (text encoder + SDP): ON by default
    since the sync-free spline made capture legal — replaying the ~200
    small SDP launches as one submission cut B=1 first-chunk latency
    9.1 -> 6.8 ms (profiles/r02_latency_ladder.json).  SONATA_HIPGRAPH=0
    disables all graph use."""
    v = os.environ.get("SONATA_HIPGRAPH", "phase1")
    if v in ("0", "", "false"):
        return False
    return True


class DecodeGraphCache:
    """Caches captured graphs of `fn(z, y_mask) -> audio` per shape."""

    def __init__(self, fn: Callable, max_shapes: int = 32):
        self.fn = fn
        self.max_shapes = max_shapes
        self._graphs: Dict[Tuple, tuple] = {}

    def __call__(self, z: torch.Tensor, y_mask: torch.Tensor) -> torch.Tensor:
        key = (tuple(z.shape), z.dtype)
        entry = self._graphs.get(key)
        if entry is None:
            if len(self._graphs) >= self.max_shapes:
                return self.fn(z, y_mask)  # cache full: eager
            entry = self._capture(z, y_mask, key)
        g, z_buf, m_buf, out_buf = entry
        z_buf.copy_(z)
        m_buf.copy_(y_mask)
        g.replay()
        return out_buf
    def _capture(self, z, y_mask, key):
        z_buf = z.clone()
        m_buf = y_mask.clone()
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                out = self.fn(z_buf, m_buf)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            out_buf = self.fn(z_buf, m_buf)
        entry = (g, z_buf, m_buf, out_buf)
        self._graphs[key] = entry
        return entry


class TupleGraphCache:
    """Graph cache for fn(*tensors) -> tuple(tensors), keyed by input
    shapes/dtypes."""

    def __init__(self, fn: Callable, max_shapes: int = 16):
        self.fn = fn
        self.max_shapes = max_shapes
        self._graphs: Dict[Tuple, tuple] = {}

    def __call__(self, *inputs: torch.Tensor):
        key = tuple((tuple(t.shape), t.dtype) for t in inputs)
        entry = self._graphs.get(key)
        if entry is None:
            if len(self._graphs) >= self.max_shapes:
                return self.fn(*inputs)
            entry = self._capture(inputs, key)
        g, in_bufs, out_bufs = entry
        for buf, t in zip(in_bufs, inputs):
            buf.copy_(t)
        g.replay()
        return out_bufs

    def _capture(self, inputs, key):
        in_bufs = tuple(t.clone() for t in inputs)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.fn(*in_bufs)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            out_bufs = self.fn(*in_bufs)
        entry = (g, in_bufs, tuple(out_bufs))
        self._graphs[key] = entry
        return entry
