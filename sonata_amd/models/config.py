"""Voice-pack configuration: Piper-compatible JSON schema + typed runtime
synthesis config.

Parity: reference crates/sonata/models/piper/src/lib.rs:112-158
(`ModelConfig`: key, language, audio{sample_rate,quality}, num_speakers,
speaker_id_map, streaming flag, espeak{voice},
inference{noise_scale,length_scale,noise_w}, num_symbols, phoneme_id_map)
and :160-166 (`PiperSynthesisConfig`: speaker + 3 scales).  The reference
mutates the synthesis config through `Box<dyn Any>` downcasts; here it is
a plain typed dataclass (SURVEY.md §5 flags that pattern as worth
replacing).

Extension: an `architecture` section records the VITS hyper-parameters so
voices we create are self-describing; when absent (a real Piper voice
config), the quality preset supplies them.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field, asdict
from typing import Dict, List, Optional

from ..core import ModelError
from ..text.ids import default_phoneme_id_map, num_symbols as _num_symbols


@dataclass
class VitsArchitecture:
    """VITS hyper-parameters (the net we build for a voice)."""

    inter_channels: int = 192
    hidden_channels: int = 192
    filter_channels: int = 768
    n_heads: int = 2
    n_layers: int = 6
    kernel_size: int = 3
    p_dropout: float = 0.0  # inference engine: dropout is identity
    resblock_kernel_sizes: List[int] = field(default_factory=lambda: [3, 7, 11])
    resblock_dilation_sizes: List[List[int]] = field(
        default_factory=lambda: [[1, 3, 5], [1, 3, 5], [1, 3, 5]]
    )
    upsample_rates: List[int] = field(default_factory=lambda: [8, 8, 2, 2])
    upsample_initial_channel: int = 512
    upsample_kernel_sizes: List[int] = field(default_factory=lambda: [16, 16, 4, 4])
    gin_channels: int = 0  # >0 for multi-speaker
    window_size: int = 4  # relative-attention window

    @property
    def hop_length(self) -> int:
        h = 1
        for r in self.upsample_rates:
            h *= r
        return h


# quality -> (sample_rate, architecture overrides)
# x_low uses a reduced net; low/medium/high share the standard VITS net
# (Piper convention: quality mainly selects sample rate / training scale).
QUALITY_PRESETS: Dict[str, dict] = {
    "x_low": dict(
        sample_rate=16000,
        arch=dict(
            inter_channels=96,
            hidden_channels=96,
            filter_channels=384,
            n_layers=3,
            upsample_initial_channel=256,
        ),
    ),
    "low": dict(sample_rate=16000, arch=dict()),
    "medium": dict(sample_rate=22050, arch=dict()),
    "high": dict(sample_rate=22050, arch=dict()),
}


@dataclass
class SynthesisConfig:
    """Runtime synthesis knobs (reference PiperSynthesisConfig,
    piper/src/lib.rs:160-166)."""

    speaker_id: Optional[int] = None
    noise_scale: float = 0.667
    length_scale: float = 1.0
    noise_w: float = 0.8

    def copy(self) -> "SynthesisConfig":
        return SynthesisConfig(
            self.speaker_id, self.noise_scale, self.length_scale, self.noise_w
        )


@dataclass
class ModelConfig:
    key: str = ""
    language_code: str = "en-us"
    sample_rate: int = 22050
    quality: str = "medium"
    num_speakers: int = 1
    speaker_id_map: Dict[str, int] = field(default_factory=dict)
    streaming: bool = False
    espeak_voice: str = "en-us"
    noise_scale: float = 0.667
    length_scale: float = 1.0
    noise_w: float = 0.8
    num_symbols: int = 0
    phoneme_id_map: Dict[str, List[int]] = field(default_factory=dict)
    architecture: VitsArchitecture = field(default_factory=VitsArchitecture)

    def __post_init__(self):
        if not self.phoneme_id_map:
            self.phoneme_id_map = default_phoneme_id_map()
        if not self.num_symbols:
            self.num_symbols = _num_symbols(self.phoneme_id_map)
        if self.num_speakers > 1 and self.architecture.gin_channels == 0:
            self.architecture.gin_channels = 256

    # ------------------------------------------------------------------ #
    @staticmethod
    def from_json_dict(d: dict) -> "ModelConfig":
        audio = d.get("audio", {})
        inference = d.get("inference", {})
        lang = d.get("language", {})
        lang_code = lang.get("code", "en-us") if isinstance(lang, dict) else str(lang)
        quality = audio.get("quality", "medium") or "medium"
        preset = QUALITY_PRESETS.get(quality, QUALITY_PRESETS["medium"])
        arch_d = dict(preset["arch"])
        arch_d.update(d.get("architecture", {}))
        arch = VitsArchitecture(**arch_d)
        num_speakers = int(d.get("num_speakers", 1))
        if num_speakers > 1 and arch.gin_channels == 0:
            arch.gin_channels = 256
        return ModelConfig(
            key=d.get("key", ""),
            language_code=lang_code,
            sample_rate=int(audio.get("sample_rate", preset["sample_rate"])),
            quality=quality,
            num_speakers=num_speakers,
            speaker_id_map=d.get("speaker_id_map", {}) or {},
            streaming=bool(d.get("streaming", False)),
            espeak_voice=d.get("espeak", {}).get("voice", lang_code),
            noise_scale=float(inference.get("noise_scale", 0.667)),
            length_scale=float(inference.get("length_scale", 1.0)),
            noise_w=float(inference.get("noise_w", 0.8)),
            num_symbols=int(d.get("num_symbols", 0)),
            phoneme_id_map=d.get("phoneme_id_map", {}) or {},
            architecture=arch,
        )

    @staticmethod
    def from_json_path(path: str) -> "ModelConfig":
        try:
            with open(path, "r", encoding="utf-8") as f:
                d = json.load(f)
        except (OSError, json.JSONDecodeError) as e:
            raise ModelError(f"cannot load voice config {path}: {e}") from e
        return ModelConfig.from_json_dict(d)

    def to_json_dict(self) -> dict:
        return {
            "key": self.key,
            "language": {"code": self.language_code},
            "audio": {"sample_rate": self.sample_rate, "quality": self.quality},
            "num_speakers": self.num_speakers,
            "speaker_id_map": self.speaker_id_map,
            "streaming": self.streaming,
            "espeak": {"voice": self.espeak_voice},
            "inference": {
                "noise_scale": self.noise_scale,
                "length_scale": self.length_scale,
                "noise_w": self.noise_w,
            },
            "num_symbols": self.num_symbols,
            "phoneme_id_map": self.phoneme_id_map,
            "architecture": asdict(self.architecture),
        }

    def save_json(self, path: str) -> None:
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w", encoding="utf-8") as f:
            json.dump(self.to_json_dict(), f, ensure_ascii=False, indent=1)

    def default_synthesis_config(self) -> SynthesisConfig:
        sid = 0 if self.num_speakers > 1 else None
        return SynthesisConfig(sid, self.noise_scale, self.length_scale, self.noise_w)
