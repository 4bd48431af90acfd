"""Python side of the C ABI frontend (csrc/capi.cpp).

The C library embeds CPython and calls these helpers; they keep the
C↔Python surface to plain tuples / bytes / generators so the C code does
no attribute navigation.  Mirrors the behavior of the reference C API
implementation (crates/frontends/capi/src/lib.rs): mode dispatch
lazy/parallel/realtime(chunk 72, pad 3), per-chunk i16 wave bytes.
"""

from __future__ import annotations

import os
from typing import Iterator, Optional, Tuple

REALTIME_CHUNK_SIZE = 72  # reference capi/src/lib.rs:407-409
REALTIME_CHUNK_PADDING = 3

MODE_LAZY, MODE_PARALLEL, MODE_REALTIME = 0, 1, 2


def load_voice(config_path: str, device: Optional[str] = None):
    import torch

    from ..models.voice import load_voice as _load
    from ..synth.synthesizer import SonataSpeechSynthesizer

    if not device:
        device = os.environ.get("SONATA_DEVICE") or (
            "cuda:0" if torch.cuda.is_available() else "cpu")
    return SonataSpeechSynthesizer(_load(config_path, device=device))


def get_audio_info(synth) -> Tuple[int, int, int]:
    info = synth.audio_output_info()
    return info.sample_rate, info.num_channels, info.sample_width


def get_synth_config(synth) -> Tuple[int, float, float, float]:
    cfg = synth.get_synthesis_config()
    return (cfg.speaker_id or 0, cfg.length_scale, cfg.noise_scale,
            cfg.noise_w)


def set_synth_config(synth, speaker: int, length_scale: float,
                     noise_scale: float, noise_w: float) -> None:
    cfg = synth.get_synthesis_config()
    cfg.speaker_id = speaker
    cfg.length_scale = length_scale
    cfg.noise_scale = noise_scale
    cfg.noise_w = noise_w
    synth.set_synthesis_config(cfg)


def _output_config(rate: int, volume: int, pitch: int, silence_ms: int):
    from ..synth.synthesizer import AudioOutputConfig

    # 0 means "unset" at the C ABI (u8 percents); reference treats params
    # as optional — map 0 to None.
    cfg = AudioOutputConfig(
        rate=rate or None, volume=volume or None, pitch=pitch or None,
        appended_silence_ms=silence_ms or None,
    )
    if cfg.is_noop and cfg.appended_silence_ms is None:
        return None
    return cfg


def speak_chunks(synth, text: str, mode: int, rate: int, volume: int,
                 pitch: int, silence_ms: int) -> Iterator[bytes]:
    """Generator of i16 wave-byte chunks for libsonataSpeak."""
    from ..audio.samples import to_i16_bytes

    out_cfg = _output_config(rate, volume, pitch, silence_ms)
    if mode == MODE_REALTIME:
        for chunk in synth.synthesize_streamed(
                text, out_cfg, REALTIME_CHUNK_SIZE, REALTIME_CHUNK_PADDING):
            yield to_i16_bytes(chunk)
    elif mode == MODE_LAZY:
        for audio in synth.synthesize_lazy(text, out_cfg):
            yield audio.as_wave_bytes()
    else:
        for audio in synth.synthesize_parallel(text, out_cfg):
            yield audio.as_wave_bytes()


def speak_to_file(synth, text: str, path: str, rate: int, volume: int,
                  pitch: int, silence_ms: int) -> None:
    synth.synthesize_to_file(path, text,
                             _output_config(rate, volume, pitch, silence_ms))
