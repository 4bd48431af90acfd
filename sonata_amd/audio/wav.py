"""Minimal RIFF/WAVE writer (16-bit PCM mono/stereo).

Parity: reference crates/audio/ops/src/wave_writer.rs:18-93 (riff-wave).
"""

from __future__ import annotations

import io
import struct

import numpy as np

from .samples import to_i16


def wav_header(data_bytes: int, sample_rate: int, num_channels: int = 1,
               sample_width: int = 2) -> bytes:
    """44-byte RIFF/WAVE header for a PCM payload of `data_bytes`."""
    import struct

    byte_rate = sample_rate * num_channels * sample_width
    block_align = num_channels * sample_width
    return (b"RIFF" + struct.pack("<I", 36 + data_bytes) + b"WAVE"
            + b"fmt " + struct.pack("<IHHIIHH", 16, 1, num_channels,
                                    sample_rate, byte_rate, block_align,
                                    sample_width * 8)
            + b"data" + struct.pack("<I", data_bytes))


def wav_bytes(samples, sample_rate: int, num_channels: int = 1,
              peak_normalize: bool = True) -> bytes:
    pcm = to_i16(samples, peak_normalize=peak_normalize).astype("<i2").tobytes()
    byte_rate = sample_rate * num_channels * 2
    block_align = num_channels * 2
    buf = io.BytesIO()
    buf.write(b"RIFF")
    buf.write(struct.pack("<I", 36 + len(pcm)))
    buf.write(b"WAVE")
    buf.write(b"fmt ")
    buf.write(struct.pack("<IHHIIHH", 16, 1, num_channels, sample_rate,
                          byte_rate, block_align, 16))
    buf.write(b"data")
    buf.write(struct.pack("<I", len(pcm)))
    buf.write(pcm)
    return buf.getvalue()


def write_wav_file(path: str, samples, sample_rate: int,
                   num_channels: int = 1) -> None:
    with open(path, "wb") as f:
        f.write(wav_bytes(samples, sample_rate, num_channels))


def read_wav_file(path: str):
    """Read a 16-bit PCM WAV back to float32 in [-1,1]. Test helper."""
    with open(path, "rb") as f:
        data = f.read()
    assert data[:4] == b"RIFF" and data[8:12] == b"WAVE"
    # walk chunks
    pos = 12
    fmt = None
    pcm = None
    while pos + 8 <= len(data):
        cid = data[pos : pos + 4]
        size = struct.unpack("<I", data[pos + 4 : pos + 8])[0]
        body = data[pos + 8 : pos + 8 + size]
        if cid == b"fmt ":
            fmt = struct.unpack("<HHIIHH", body[:16])
        elif cid == b"data":
            pcm = body
        pos += 8 + size + (size & 1)
    assert fmt is not None and pcm is not None
    _, channels, rate, _, _, bits = fmt
    assert bits == 16
    x = np.frombuffer(pcm, dtype="<i2").astype(np.float32) / 32767.0
    return x, rate, channels
