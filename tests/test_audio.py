"""Audio DSP unit tests (hermetic, numpy only).

Models the reference's audio-ops unit tests (ops/src/samples.rs:282-350).
"""

import numpy as np
import pytest

from sonata_amd.audio import (
    apply_hann_window,
    crossfade,
    fade_in,
    fade_out,
    hann_window,
    highpass_amplitude,
    lowpass_amplitude,
    normalize,
    overlap_with,
    strip_silence,
    to_decibel,
    to_i16,
    to_i16_bytes,
    wav_bytes,
)
from sonata_amd.audio.prosody import apply_prosody, resample_linear, time_stretch_wsola
from sonata_amd.audio.wav import read_wav_file, write_wav_file


def test_to_i16_peak_normalizes():
    x = np.array([0.0, 0.25, -0.5], dtype=np.float32)
    out = to_i16(x)
    assert out.dtype == np.int16
    assert out[2] == -32767
    assert abs(int(out[1]) - 16384) <= 1


def test_to_i16_empty_and_silence():
    assert to_i16(np.zeros(0)).size == 0
    assert np.all(to_i16(np.zeros(10)) == 0)


def test_to_i16_bytes_le():
    b = to_i16_bytes(np.array([1.0], dtype=np.float32))
    assert b == (32767).to_bytes(2, "little")


def test_normalize():
    x = np.array([0.1, -0.2], dtype=np.float32)
    n = normalize(x)
    assert np.isclose(np.abs(n).max(), 1.0)
    assert normalize(np.zeros(4)).max() == 0.0


def test_fades_monotone():
    x = np.ones(100, dtype=np.float32)
    fi = fade_in(x, 50)
    assert fi[0] == 0.0 and np.isclose(fi[49], 1.0, atol=1e-6)
    assert np.all(np.diff(fi[:50]) >= -1e-7)
    fo = fade_out(x, 50)
    assert np.isclose(fo[50], 1.0, atol=1e-6) and fo[-1] == 0.0


def test_crossfade_length_and_continuity():
    a = np.ones(100, dtype=np.float32)
    b = np.ones(100, dtype=np.float32)
    out = crossfade(a, b, 20)
    assert len(out) == 180
    # equal-power-ish: mixed region stays near 1 for identical signals
    assert np.all(out > 0.65)
    assert np.allclose(overlap_with(a, b, 20), out)


def test_crossfade_zero_overlap():
    out = crossfade(np.ones(5), np.zeros(5), 0)
    assert len(out) == 10


def test_hann_window():
    w = hann_window(64)
    assert w[0] == 0.0
    assert np.isclose(w[32], 1.0, atol=1e-6)
    x = np.ones(64, dtype=np.float32)
    assert np.allclose(apply_hann_window(x), w)


def test_amplitude_filters():
    x = np.array([0.1, 0.9, -0.05], dtype=np.float32)
    lo = lowpass_amplitude(x, 0.5)
    assert lo[1] == 0.0 and lo[0] == np.float32(0.1)
    hi = highpass_amplitude(x, 0.5)
    assert hi[0] == 0.0 and hi[1] == np.float32(0.9)


def test_strip_silence():
    x = np.array([0.0, 0.0, 0.5, 0.1, 0.0], dtype=np.float32)
    s = strip_silence(x, 0.05)
    assert np.allclose(s, [0.5, 0.1])
    assert strip_silence(np.zeros(8)).size == 0


def test_to_decibel():
    db = to_decibel(np.array([1.0, 0.1], dtype=np.float32))
    assert np.isclose(db[0], 0.0, atol=1e-5)
    assert np.isclose(db[1], -20.0, atol=1e-4)


def test_wav_roundtrip(tmp_path):
    rng = np.random.default_rng(0)
    x = (rng.standard_normal(1000) * 0.3).astype(np.float32)
    p = str(tmp_path / "t.wav")
    write_wav_file(p, x, 22050)
    y, rate, ch = read_wav_file(p)
    assert rate == 22050 and ch == 1 and len(y) == 1000
    # peak-normalized on write; compare shapes after normalize
    assert np.corrcoef(x, y)[0, 1] > 0.999


def test_wav_bytes_header():
    b = wav_bytes(np.zeros(4, dtype=np.float32), 16000)
    assert b[:4] == b"RIFF" and b[8:12] == b"WAVE"
    assert len(b) == 44 + 8


def test_resample_linear():
    x = np.sin(np.linspace(0, 10, 1000)).astype(np.float32)
    y = resample_linear(x, 2.0)
    assert abs(len(y) - 500) <= 1
    y2 = resample_linear(x, 1.0)
    assert np.allclose(x, y2)


def test_time_stretch_duration():
    sr = 22050
    t = np.arange(sr, dtype=np.float32) / sr
    x = np.sin(2 * np.pi * 220 * t).astype(np.float32)
    y = time_stretch_wsola(x, 2.0, sr)
    assert abs(len(y) - sr // 2) < sr // 20
    y = time_stretch_wsola(x, 0.5, sr)
    assert abs(len(y) - 2 * sr) < sr // 10


def test_apply_prosody_volume():
    x = np.ones(100, dtype=np.float32) * 0.5
    y = apply_prosody(x, 22050, volume=0.5)
    assert np.allclose(y, 0.25)


def test_crossfade_reference_semantics():
    """Quarter-sine crossfade (reference samples.rs:144-157): output
    length = len(a) + len(b) - n; constant-ish power through the seam."""
    import numpy as np

    from sonata_amd.audio.samples import crossfade

    a = np.ones(100, np.float32)
    b = np.ones(80, np.float32) * -1.0
    out = crossfade(a, b, 20)
    assert len(out) == 160
    # ends untouched
    np.testing.assert_array_equal(out[:80], a[:80])
    np.testing.assert_array_equal(out[-60:], b[-60:])
    # seam is a monotonic blend from +1 toward -1
    seam = out[80:100]
    assert seam[0] > 0.8 and seam[-1] < -0.8
    assert (np.diff(seam) <= 1e-6).all()


def test_to_i16_peak_normalization_reference():
    """Reference semantics (samples.rs:51-75): scale = 32767/absmax
    ALWAYS — quiet audio is amplified to full scale, loud audio is
    brought back into range."""
    import numpy as np

    from sonata_amd.audio.samples import to_i16

    quiet = np.array([0.5, -0.25], np.float32)
    out = to_i16(quiet)
    assert out[0] == 32767 and out[1] == -16383  # scaled by 32767/0.5
    loud = np.array([2.0, -1.0], np.float32)
    out = to_i16(loud)
    assert out[0] == 32767 and out[1] == -16383  # scaled by 32767/2
    # normalization can be disabled (raw stream chunks)
    raw = to_i16(quiet, peak_normalize=False)
    assert raw[0] == int(0.5 * 32767)


def test_overlap_with_sine_ramp():
    """overlap_with joins with sine ramps (reference samples.rs:102-118):
    length = len(a) + len(b) - n and the overlap mixes both."""
    import numpy as np

    from sonata_amd.audio.samples import overlap_with

    a = np.full(50, 0.8, np.float32)
    b = np.full(50, 0.2, np.float32)
    out = overlap_with(a, b, 10)
    assert len(out) == 90
    # sine ramps sum slightly over 1 mid-seam (reference semantics);
    # bounded overshoot, ends exact
    assert np.all(out[40:50] <= 0.9) and np.all(out[40:50] >= 0.15)
    assert abs(out[40] - 0.8) < 0.05 and abs(out[49] - 0.2) < 0.05


def test_wav_roundtrip_header():
    """Our WAV writer produces a parseable PCM16 file (wave stdlib)."""
    import io
    import wave

    import numpy as np

    from sonata_amd.audio.wav import wav_bytes

    samples = np.sin(np.linspace(0, 20, 4410)).astype(np.float32) * 0.7
    data = wav_bytes(samples, 22050)
    with wave.open(io.BytesIO(data)) as w:
        assert w.getframerate() == 22050
        assert w.getnchannels() == 1
        assert w.getsampwidth() == 2
        assert w.getnframes() == 4410
