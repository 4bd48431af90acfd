"""Command-line frontend.

Parity: reference crates/frontends/cli/src/main.rs — positional voice
config path; `-f` input file or JSON-lines stdin loop (:78-124); `-o`
output file auto-enumerated `name-N.ext` per request (:234-247); flags
mode / speaker-id / length-scale / noise-scale / noise-w / rate / pitch /
volume / silence / chunk-size / chunk-padding (:32-76); lazy / parallel /
realtime dispatch with realtime default chunk 100 / pad 3 (:126-182);
raw WAV bytes to stdout when no output file is given (:160-176);
`SONATA_LOG` env controls logging (:113-116).

MI355X addition: `--device` (default cuda:0 when available) selects the
GPU serving path.
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import sys
from dataclasses import dataclass
from typing import Optional

log = logging.getLogger("sonata")


@dataclass
class SynthesisRequest:
    """One JSON-stdin request (reference SynthesisRequest, cli main.rs:78-111)."""

    text: str
    speaker_id: Optional[int] = None
    length_scale: Optional[float] = None
    noise_scale: Optional[float] = None
    noise_w: Optional[float] = None
    rate: Optional[float] = None
    pitch: Optional[float] = None
    volume: Optional[float] = None
    appended_silence_ms: Optional[float] = None

    @staticmethod
    def from_json(line: str) -> "SynthesisRequest":
        d = json.loads(line)
        return SynthesisRequest(
            text=d["text"],
            speaker_id=d.get("speaker_id"),
            length_scale=d.get("length_scale"),
            noise_scale=d.get("noise_scale"),
            noise_w=d.get("noise_w"),
            rate=d.get("rate"),
            pitch=d.get("pitch"),
            volume=d.get("volume"),
            appended_silence_ms=d.get("appended_silence_ms"),
        )


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="sonata",
        description="MI355X-native Piper/VITS text-to-speech",
    )
    p.add_argument("config", help="voice config path (<stem>.json)")
    p.add_argument("-f", "--input-file",
                   help="read text from file; default: JSON-lines stdin loop")
    p.add_argument("-o", "--output-file",
                   help="output WAV path (auto-enumerated per request); "
                        "default: raw WAV bytes to stdout")
    p.add_argument("-m", "--mode", default="parallel",
                   choices=["lazy", "parallel", "realtime"])
    p.add_argument("-s", "--speaker-id", type=int, default=None)
    p.add_argument("--length-scale", type=float, default=None)
    p.add_argument("--noise-scale", type=float, default=None)
    p.add_argument("--noise-w", type=float, default=None)
    p.add_argument("-r", "--rate", type=float, default=None,
                   help="speaking rate percent 0-100")
    p.add_argument("-p", "--pitch", type=float, default=None)
    p.add_argument("-v", "--volume", type=float, default=None)
    p.add_argument("--silence", type=float, default=None,
                   help="appended silence ms after each sentence")
    p.add_argument("--chunk-size", type=int, default=100,
                   help="realtime first-chunk mel frames (reference default)")
    p.add_argument("--chunk-padding", type=int, default=3)
    p.add_argument("--device", default=None,
                   help="torch device (default: cuda:0 if available)")
    return p


def _enumerate_path(path: str, n: int) -> str:
    """out.wav -> out-1.wav for request n>0 (reference main.rs:234-247)."""
    if n == 0:
        return path
    stem, ext = os.path.splitext(path)
    return f"{stem}-{n}{ext}"


def _make_output_config(args, req: SynthesisRequest):
    from ..synth.synthesizer import AudioOutputConfig

    rate = req.rate if req.rate is not None else args.rate
    pitch = req.pitch if req.pitch is not None else args.pitch
    volume = req.volume if req.volume is not None else args.volume
    silence = (req.appended_silence_ms if req.appended_silence_ms is not None
               else args.silence)
    if rate is None and pitch is None and volume is None and silence is None:
        return None
    return AudioOutputConfig(rate=rate, volume=volume, pitch=pitch,
                             appended_silence_ms=silence)


def _apply_synth_config(voice, args, req: SynthesisRequest) -> None:
    cfg = voice.get_synthesis_config()
    sid = req.speaker_id if req.speaker_id is not None else args.speaker_id
    if sid is not None:
        cfg.speaker_id = sid
    for name in ("length_scale", "noise_scale", "noise_w"):
        v = getattr(req, name)
        if v is None:
            v = getattr(args, name)
        if v is not None:
            setattr(cfg, name, v)
    voice.set_synthesis_config(cfg)


def process_request(synth, args, req: SynthesisRequest, n: int,
                    stdout=None) -> int:
    """Synthesize one request; returns number of audio samples produced."""
    import numpy as np

    from ..audio.wav import wav_bytes
    from ..core import Audio

    _apply_synth_config(synth.model, args, req)
    out_cfg = _make_output_config(args, req)
    info = synth.audio_output_info()
    total = 0
    if args.output_file:
        path = _enumerate_path(args.output_file, n)
        audio = synth.synthesize_to_file(path, req.text, out_cfg)
        total = len(audio.samples)
        log.info("wrote %s (%.1f ms audio, rtf %.4f)", path,
                 audio.duration_ms, audio.real_time_factor)
    else:
        stdout = stdout if stdout is not None else sys.stdout.buffer
        if args.mode == "realtime":
            # stream chunks to stdout AS THEY ARRIVE (reference cli
            # main.rs:160-176): WAV header first (size patched only when
            # stdout is seekable), then i16 PCM per chunk
            from ..audio.samples import to_i16_bytes
            from ..audio.wav import wav_header

            header_pos = stdout.tell() if stdout.seekable() else None
            stdout.write(wav_header(0, info.sample_rate))
            for chunk in synth.synthesize_streamed(
                    req.text, out_cfg, args.chunk_size, args.chunk_padding):
                stdout.write(to_i16_bytes(chunk, peak_normalize=False))
                stdout.flush()
                total += len(chunk)
            if header_pos is not None:
                end = stdout.tell()
                stdout.seek(header_pos)
                stdout.write(wav_header(total * 2, info.sample_rate))
                stdout.seek(end)
            stdout.flush()
            return total
        else:
            it = (synth.synthesize_lazy(req.text, out_cfg)
                  if args.mode == "lazy"
                  else synth.synthesize_parallel(req.text, out_cfg))
            parts = [a.samples for a in it]
            samples = (np.concatenate(parts) if parts
                       else np.zeros(0, dtype=np.float32))
        total = len(samples)
        stdout.write(wav_bytes(samples, info.sample_rate))
        stdout.flush()
    return total


def main(argv=None, stdin=None, stdout=None) -> int:
    args = build_parser().parse_args(argv)
    logging.basicConfig(
        level=os.environ.get("SONATA_LOG", "INFO").upper(),
        format="%(levelname)s %(name)s: %(message)s",
    )
    import torch

    from ..models.voice import load_voice
    from ..synth.synthesizer import SonataSpeechSynthesizer

    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    voice = load_voice(args.config, device=device)
    synth = SonataSpeechSynthesizer(voice)

    if args.input_file:
        with open(args.input_file, "r", encoding="utf-8") as f:
            text = f.read()
        process_request(synth, args, SynthesisRequest(text=text), 0, stdout)
        return 0

    # JSON-lines stdin loop (reference main.rs:78-124): one request per
    # line; malformed JSON is logged and skipped (:255).
    stdin = stdin if stdin is not None else sys.stdin
    n = 0
    for line in stdin:
        line = line.strip()
        if not line:
            continue
        try:
            req = SynthesisRequest.from_json(line)
        except (json.JSONDecodeError, KeyError) as e:
            log.error("bad request line: %s", e)
            continue
        process_request(synth, args, req, n, stdout)
        n += 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
