"""Per-stage timing (observability).

The reference's only instrumentation is a wall-clock around session.run
surfaced as RTF (piper/src/lib.rs:361-398).  Here every
pipeline stage (phonemize / encode / duration / flow / decode / post)
can be timed; GPU stages bracket with torch.cuda events so the numbers
are device-accurate.  Enabled via SONATA_TRACE=1 (near-zero cost when
off); the per-thread accumulator is retrievable programmatically for
servers that export metrics.

rocprofv3 remains the deep profiler (profiles/ holds committed kernel
stats); this module covers the always-on production counters.
"""

from __future__ import annotations

import os
import threading
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict

_TLS = threading.local()


def _enabled() -> bool:
    return os.environ.get("SONATA_TRACE", "0") not in ("0", "", "false")


class StageTimes:
    """Accumulated per-stage wall milliseconds for the current thread."""

    def __init__(self):
        self.ms: Dict[str, float] = defaultdict(float)
        self.calls: Dict[str, int] = defaultdict(int)

    def add(self, stage: str, ms: float) -> None:
        self.ms[stage] += ms
        self.calls[stage] += 1

    def snapshot(self) -> Dict[str, dict]:
        return {
            s: {"ms": round(self.ms[s], 3), "calls": self.calls[s]}
            for s in sorted(self.ms)
        }

    def reset(self) -> None:
        self.ms.clear()
        self.calls.clear()


def get_stage_times() -> StageTimes:
    st = getattr(_TLS, "times", None)
    if st is None:
        st = _TLS.times = StageTimes()
    return st


@contextmanager
def stage_timer(stage: str, device=None):
    """Time a pipeline stage; synchronizes CUDA when the stage ran on a
    GPU device so the measurement is device-true."""
    if not _enabled():
        yield
        return
    import torch

    sync = device is not None and str(device).startswith("cuda") \
        and torch.cuda.is_available()
    if sync:
        torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if sync:
            torch.cuda.synchronize(device)
        get_stage_times().add(stage, (time.perf_counter() - t0) * 1000.0)
