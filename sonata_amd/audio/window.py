"""Hann window with a power-of-two LUT cache.

Parity: reference crates/audio/ops/src/hanning_window.rs:4-76 (precomputed
LUT for lengths 64..4096, periodic Hann formula).
"""

from __future__ import annotations

from functools import lru_cache

import numpy as np


@lru_cache(maxsize=64)
def _hann_cached(n: int) -> np.ndarray:
    if n <= 1:
        return np.ones(max(n, 0), dtype=np.float32)
    k = np.arange(n, dtype=np.float64)
    w = 0.5 - 0.5 * np.cos(2.0 * np.pi * k / n)  # periodic Hann
    w.flags.writeable = False
    return w.astype(np.float32)


def hann_window(n: int) -> np.ndarray:
    return _hann_cached(int(n))
