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


# --------------------------------------------------------------------- #
# Native-engine support (csrc/capi/capi.cpp hot path): the C library
# drives the C++ VitsEngine directly for synthesis; Python is entered
# only for the TEXT front-end (phonemize + tashkeel + per-utterance
# seeds) and optional prosody DSP.  A C caller never holds the GIL
# while the neural graph runs.
# --------------------------------------------------------------------- #
class CFrontend:
    """Lightweight text front-end: voice config + phonemizer (+ tashkeel
    for Arabic voices).  Loads NO model weights."""

    def __init__(self, config_path: str):
        from ..models.config import ModelConfig

        self.config = ModelConfig.from_json_path(config_path)
        self._tashkeel = None
        if self.config.espeak_voice.startswith("ar"):
            from ..text.tashkeel import TashkeelModel

            self._tashkeel = TashkeelModel.default(device="cpu")

    def phonemize(self, text: str):
        from ..text.phonemizer import text_to_phonemes

        if self._tashkeel is not None:
            text = self._tashkeel.diacritize(text)
        return text_to_phonemes(text, voice=self.config.espeak_voice)


def load_frontend(config_path: str) -> CFrontend:
    return CFrontend(config_path)


def phonemize_with_seeds(frontend: CFrontend, text: str, speaker_id: int):
    """[(phonemes, seed), ...] — seeds use the SAME derivation as the
    Python engine path (models/voice.py _utterance_seed) so native and
    Python synthesis of the same text produce identical audio."""
    from ..models.voice import _utterance_seed

    sid = speaker_id if frontend.config.num_speakers > 1 else None
    return [(p, _utterance_seed(p, sid)) for p in frontend.phonemize(text)]


def apply_prosody(f32_bytes: bytes, sample_rate: int, rate: int,
                  volume: int, pitch: int) -> bytes:
    """rate/volume/pitch percents (0=unset) applied to f32 samples."""
    import numpy as np

    cfg = _output_config(rate, volume, pitch, 0)
    s = np.frombuffer(f32_bytes, dtype=np.float32)
    if cfg is None or cfg.is_noop:
        return f32_bytes
    return cfg.apply(s, sample_rate).astype(np.float32).tobytes()


def silence_samples(ms: int, sample_rate: int) -> int:
    return int(sample_rate * ms / 1000.0)
