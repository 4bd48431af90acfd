"""Waveform DSP primitives (vectorized numpy).

Parity: reference crates/audio/ops/src/samples.rs — to_i16 peak scaling
(:51-75), merge/normalize (:79-94), hann window apply (:95-101),
overlap_with sine-ramp overlap-add (:102-118), quarter-sine fades and
crossfade (:119-157), amplitude-threshold low/highpass (:158-171),
strip_silence (:172-181), to_decibel (:182-184).  The implementations here
are fresh vectorized designs with the same observable semantics.
"""

from __future__ import annotations

import numpy as np

from .window import hann_window

_I16_MAX = 32767.0


def _f32(x) -> np.ndarray:
    return np.asarray(x, dtype=np.float32).reshape(-1)


def to_i16(samples, peak_normalize: bool = True) -> np.ndarray:
    """float32 -> int16 with peak normalization (scale = 32767/absmax)."""
    s = _f32(samples)
    if s.size == 0:
        return np.zeros(0, dtype=np.int16)
    if peak_normalize:
        peak = float(np.max(np.abs(s)))
        scale = _I16_MAX / peak if peak > 1e-8 else 0.0
    else:
        scale = _I16_MAX
    out = np.clip(s * scale, -32768.0, 32767.0)
    return out.astype(np.int16)


def to_i16_bytes(samples, peak_normalize: bool = True) -> bytes:
    """Little-endian i16 PCM bytes."""
    return to_i16(samples, peak_normalize).astype("<i2").tobytes()


def merge(a, b) -> np.ndarray:
    return np.concatenate([_f32(a), _f32(b)])


def normalize(samples) -> np.ndarray:
    s = _f32(samples)
    peak = float(np.max(np.abs(s))) if s.size else 0.0
    if peak <= 1e-8:
        return s
    return s / peak


def apply_hann_window(samples) -> np.ndarray:
    s = _f32(samples)
    return s * hann_window(len(s))


def _quarter_sine_ramp(n: int) -> np.ndarray:
    """Ramp 0->1 following sin(x) over [0, pi/2]."""
    if n <= 0:
        return np.zeros(0, dtype=np.float32)
    x = np.linspace(0.0, np.pi / 2.0, n, dtype=np.float32)
    return np.sin(x).astype(np.float32)


def fade_in(samples, n: int) -> np.ndarray:
    s = _f32(samples).copy()
    n = min(n, len(s))
    s[:n] *= _quarter_sine_ramp(n)
    return s


def fade_out(samples, n: int) -> np.ndarray:
    s = _f32(samples).copy()
    n = min(n, len(s))
    s[len(s) - n :] *= _quarter_sine_ramp(n)[::-1]
    return s


def crossfade(a, b, n: int) -> np.ndarray:
    """Join a and b with an n-sample equal-power crossfade.

    The last n samples of `a` are mixed with the first n of `b`
    (reference: samples.rs:144-157, used with n=42 at stream-chunk seams).
    """
    a = _f32(a)
    b = _f32(b)
    n = min(n, len(a), len(b))
    if n == 0:
        return np.concatenate([a, b])
    ramp = _quarter_sine_ramp(n)
    mixed = a[len(a) - n :] * ramp[::-1] + b[:n] * ramp
    return np.concatenate([a[: len(a) - n], mixed, b[n:]])


def overlap_with(a, b, n: int) -> np.ndarray:
    """Sine-ramp overlap-add join of two buffers over n samples
    (reference: samples.rs:102-118)."""
    return crossfade(a, b, n)


def lowpass_amplitude(samples, threshold: float) -> np.ndarray:
    """Amplitude-threshold 'lowpass': zero samples above |threshold|
    (the reference's filters are amplitude gates, not spectral —
    samples.rs:158-171)."""
    s = _f32(samples).copy()
    s[np.abs(s) > threshold] = 0.0
    return s


def highpass_amplitude(samples, threshold: float) -> np.ndarray:
    s = _f32(samples).copy()
    s[np.abs(s) < threshold] = 0.0
    return s


def strip_silence(samples, threshold: float = 1e-4) -> np.ndarray:
    """Trim leading/trailing samples under |threshold|."""
    s = _f32(samples)
    nz = np.flatnonzero(np.abs(s) >= threshold)
    if nz.size == 0:
        return np.zeros(0, dtype=np.float32)
    return s[nz[0] : nz[-1] + 1]


def to_decibel(samples) -> np.ndarray:
    s = _f32(samples)
    return (20.0 * np.log10(np.maximum(np.abs(s), 1e-10))).astype(np.float32)


def generate_silence(ms: float, sample_rate: int) -> np.ndarray:
    n = int(round(ms * sample_rate / 1000.0))
    return np.zeros(max(n, 0), dtype=np.float32)
