"""pysonata-compatible Python API.

Parity: reference crates/frontends/python/src/lib.rs — classes
`PiperModel(config_path)` with speaker + scale getters/setters (:241-326),
`Sonata.with_piper(model)` (:328-338), `synthesize`/`synthesize_lazy`/
`synthesize_parallel` -> iterator of `WaveSamples` (:339-367),
`synthesize_streamed(chunk_size=45, chunk_padding=3)` -> iterator of raw
i16 bytes (:369-383), `synthesize_to_file` (:385-394), `language`/
`speakers`/`get_audio_output_info` (:395-406), `AudioOutputConfig`
(:69-96), `WaveSamples` with wave bytes + sample_rate / duration_ms /
inference_ms / real_time_factor / save_to_file (:98-134), standalone
`phonemize_text` (:408-440), `SonataException` (:21-35).

Drop-in import:  `from sonata_amd.frontends import pysonata`
MI355X addition: `PiperModel(config_path, device=...)` loads straight
onto a GPU (bf16) — default cuda:0 when available.
"""

from __future__ import annotations

from typing import Iterator, List, Optional

from ..core import SonataError
from ..synth.synthesizer import AudioOutputConfig as _OutCfg
from ..synth.synthesizer import SonataSpeechSynthesizer


class SonataException(Exception):
    pass


class AudioOutputConfig:
    """Prosody knobs in percent 0-100 (reference python lib.rs:69-96)."""

    def __init__(self, rate: Optional[float] = None,
                 volume: Optional[float] = None,
                 pitch: Optional[float] = None,
                 appended_silence_ms: Optional[float] = None):
        self.rate = rate
        self.volume = volume
        self.pitch = pitch
        self.appended_silence_ms = appended_silence_ms

    def _to_internal(self) -> _OutCfg:
        return _OutCfg(rate=self.rate, volume=self.volume, pitch=self.pitch,
                       appended_silence_ms=self.appended_silence_ms)


class WaveSamples:
    """Synthesized audio chunk (reference python lib.rs:98-134)."""

    def __init__(self, audio):
        self._audio = audio

    def get_wave_bytes(self) -> bytes:
        return self._audio.as_wave_bytes()

    @property
    def sample_rate(self) -> int:
        return self._audio.info.sample_rate

    @property
    def duration_ms(self) -> float:
        return self._audio.duration_ms

    @property
    def inference_ms(self) -> float:
        return self._audio.inference_ms

    @property
    def real_time_factor(self) -> float:
        return self._audio.real_time_factor

    def save_to_file(self, path: str) -> None:
        self._audio.save_to_file(path)


class PiperModel:
    """A loaded Piper voice (reference python lib.rs:241-326)."""

    def __init__(self, config_path: str, device: Optional[str] = None,
                 **kwargs):
        import torch

        if device is None:
            device = "cuda:0" if torch.cuda.is_available() else "cpu"
        try:
            from ..models.voice import load_voice

            self._voice = load_voice(config_path, device=device, **kwargs)
        except SonataError as e:
            raise SonataException(str(e)) from e

    # speaker + scale accessors
    @property
    def speakers(self) -> Optional[dict]:
        return self._voice.get_speakers()

    def get_speaker(self) -> Optional[str]:
        cfg = self._voice.get_synthesis_config()
        speakers = self._voice.get_speakers() or {}
        return speakers.get(cfg.speaker_id)

    def set_speaker(self, name: str) -> None:
        speakers = self._voice.get_speakers() or {}
        by_name = {v: k for k, v in speakers.items()}
        if name not in by_name:
            raise SonataException(f"unknown speaker: {name}")
        cfg = self._voice.get_synthesis_config()
        cfg.speaker_id = by_name[name]
        self._voice.set_synthesis_config(cfg)

    def _get_scale(self, name: str) -> float:
        return getattr(self._voice.get_synthesis_config(), name)

    def _set_scale(self, name: str, value: float) -> None:
        cfg = self._voice.get_synthesis_config()
        setattr(cfg, name, float(value))
        self._voice.set_synthesis_config(cfg)

    @property
    def length_scale(self) -> float:
        return self._get_scale("length_scale")

    @length_scale.setter
    def length_scale(self, v: float) -> None:
        self._set_scale("length_scale", v)

    @property
    def noise_scale(self) -> float:
        return self._get_scale("noise_scale")

    @noise_scale.setter
    def noise_scale(self, v: float) -> None:
        self._set_scale("noise_scale", v)

    @property
    def noise_w(self) -> float:
        return self._get_scale("noise_w")

    @noise_w.setter
    def noise_w(self, v: float) -> None:
        self._set_scale("noise_w", v)


class Sonata:
    """Speech synthesizer facade (reference python lib.rs:328-406)."""

    def __init__(self, model: PiperModel):
        self._model = model
        self._synth = SonataSpeechSynthesizer(model._voice)

    @staticmethod
    def with_piper(model: PiperModel) -> "Sonata":
        return Sonata(model)

    # --- synthesis ---------------------------------------------------- #
    def synthesize(self, text: str,
                   audio_output_config: Optional[AudioOutputConfig] = None
                   ) -> Iterator[WaveSamples]:
        return self.synthesize_parallel(text, audio_output_config)

    def synthesize_lazy(self, text: str,
                        audio_output_config: Optional[AudioOutputConfig]
                        = None) -> Iterator[WaveSamples]:
        cfg = audio_output_config._to_internal() if audio_output_config else None
        for a in self._synth.synthesize_lazy(text, cfg):
            yield WaveSamples(a)

    def synthesize_parallel(self, text: str,
                            audio_output_config: Optional[AudioOutputConfig]
                            = None) -> Iterator[WaveSamples]:
        cfg = audio_output_config._to_internal() if audio_output_config else None
        for a in self._synth.synthesize_parallel(text, cfg):
            yield WaveSamples(a)

    def synthesize_streamed(self, text: str,
                            audio_output_config: Optional[AudioOutputConfig]
                            = None, chunk_size: int = 45,
                            chunk_padding: int = 3) -> Iterator[bytes]:
        from ..audio.samples import to_i16_bytes

        cfg = audio_output_config._to_internal() if audio_output_config else None
        for chunk in self._synth.synthesize_streamed(text, cfg, chunk_size,
                                                     chunk_padding):
            yield to_i16_bytes(chunk)

    def synthesize_to_file(self, filename: str, text: str,
                           audio_output_config: Optional[AudioOutputConfig]
                           = None) -> None:
        cfg = audio_output_config._to_internal() if audio_output_config else None
        self._synth.synthesize_to_file(filename, text, cfg)

    # --- info --------------------------------------------------------- #
    @property
    def language(self) -> Optional[str]:
        return self._synth.model.language

    @property
    def speakers(self) -> Optional[dict]:
        return self._synth.model.get_speakers()

    def get_audio_output_info(self):
        return self._synth.audio_output_info()


def phonemize_text(text: str, language: str = "en-us",
                   separator: Optional[str] = None,
                   remove_lang_switch_flags: bool = False,
                   remove_stress: bool = False,
                   use_tashkeel: bool = False) -> List[str]:
    """Standalone phonemizer (reference python lib.rs:408-440)."""
    from ..text.phonemizer import text_to_phonemes

    if use_tashkeel:
        from ..text.tashkeel import TashkeelModel

        text = TashkeelModel.default().diacritize(text)
    return text_to_phonemes(text, voice=language, separator=separator,
                            remove_lang_switch=remove_lang_switch_flags,
                            remove_stress=remove_stress)
