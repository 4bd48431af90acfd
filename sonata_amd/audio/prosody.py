"""Prosody post-processing: rate / volume / pitch on raw waveforms.

Parity: the reference delegates this to the `sonic` C library
(crates/sonata/synth/src/lib.rs:55-105: sonicCreateStream -> SetSpeed/
SetVolume/SetPitch -> WriteFloat -> Flush -> Read).  Here it is a fresh
CPU implementation: WSOLA time-stretch + linear-interpolation resampling,
which compose to give speed and pitch control with the same parameter
semantics (speed in (0.5, 5.5), volume in (0, 1], pitch in (0.5, 1.5) —
synth/src/lib.rs:13-15).
"""

from __future__ import annotations

import numpy as np

from .window import hann_window


def resample_linear(x: np.ndarray, ratio: float) -> np.ndarray:
    """Resample by `ratio` (output length = len(x)/ratio) with linear interp."""
    x = np.asarray(x, dtype=np.float32).reshape(-1)
    if x.size == 0 or abs(ratio - 1.0) < 1e-6:
        return x.copy()
    n_out = max(int(round(len(x) / ratio)), 1)
    pos = np.arange(n_out, dtype=np.float64) * (len(x) - 1) / max(n_out - 1, 1)
    i0 = np.floor(pos).astype(np.int64)
    i1 = np.minimum(i0 + 1, len(x) - 1)
    frac = (pos - i0).astype(np.float32)
    return (x[i0] * (1.0 - frac) + x[i1] * frac).astype(np.float32)


def time_stretch_wsola(
    x: np.ndarray,
    speed: float,
    sample_rate: int,
    frame_ms: float = 30.0,
    search_ms: float = 10.0,
) -> np.ndarray:
    """WSOLA time stretch: output duration = input/speed, pitch preserved."""
    x = np.asarray(x, dtype=np.float32).reshape(-1)
    if x.size == 0 or abs(speed - 1.0) < 1e-3:
        return x.copy()
    frame = max(int(sample_rate * frame_ms / 1000.0), 64)
    half = frame // 2
    frame = half * 2
    search = max(int(sample_rate * search_ms / 1000.0), 16)
    syn_hop = half
    ana_hop = syn_hop * speed

    n_out_frames = max(int((len(x) - frame - search) / ana_hop), 1)
    out = np.zeros(n_out_frames * syn_hop + frame, dtype=np.float32)
    norm = np.zeros_like(out)
    win = hann_window(frame)

    prev_tail = None
    for k in range(n_out_frames):
        target = int(k * ana_hop)
        if prev_tail is not None and search > 1:
            lo = max(target - search, 0)
            hi = min(target + search, len(x) - frame)
            if hi > lo:
                # pick the candidate start maximizing correlation with the
                # previous synthesis frame's tail (natural continuation)
                seg = x[lo : hi + half]
                # vectorized cross-correlation over candidate offsets
                n_cand = hi - lo
                idx = np.arange(half)
                cand = seg[np.arange(n_cand)[:, None] + idx[None, :]]
                scores = cand @ prev_tail
                target = lo + int(np.argmax(scores))
        target = min(max(target, 0), len(x) - frame)
        fr = x[target : target + frame] * win
        pos = k * syn_hop
        out[pos : pos + frame] += fr
        norm[pos : pos + frame] += win
        prev_tail = x[target + syn_hop : target + syn_hop + half]

    nz = norm > 1e-6
    out[nz] /= norm[nz]
    n_expect = int(len(x) / speed)
    return out[:n_expect] if len(out) >= n_expect else out


def apply_prosody(
    samples: np.ndarray,
    sample_rate: int,
    speed: float = 1.0,
    volume: float = 1.0,
    pitch: float = 1.0,
) -> np.ndarray:
    """speed: 1.0 = unchanged, 2.0 = twice as fast (duration halved).
    pitch:  1.0 = unchanged, 2.0 = one octave up (duration preserved).
    volume: linear gain."""
    y = np.asarray(samples, dtype=np.float32).reshape(-1)
    if abs(pitch - 1.0) >= 1e-3:
        # shift pitch: resample by pitch (changes duration), then stretch back
        y = resample_linear(y, pitch)
        y = time_stretch_wsola(y, 1.0 / pitch, sample_rate)
    if abs(speed - 1.0) >= 1e-3:
        y = time_stretch_wsola(y, speed, sample_rate)
    if abs(volume - 1.0) >= 1e-6:
        y = y * np.float32(volume)
    return y
