"""sonata_grpc server.

Parity: reference crates/frontends/grpc/src/main.rs — default port 49314
on 127.0.0.1, `SONATA_GRPC_SERVER_PORT` override (:17,437-440); voice
registry keyed by a hash of the canonical config path, idempotent
LoadVoice (:76-108); speaker set by NAME through SynthesisOptions
(:211-255); SynthesizeUtterance streams one SynthesisResult (WAV bytes +
RTF) per sentence (:320-355); SynthesizeUtteranceRealtime streams
WaveSamples chunks via synthesize_streamed(text, cfg, 55, 3) (:356-410);
errors map to gRPC status codes (:47-59).

MI355X-native: voices load straight onto the GPU (bf16) and all sentence
synthesis inside one utterance runs as a true padded batch on-device.
Without generated stubs (no protoc offline), the service is registered
through grpc generic handlers over dynamically-built descriptors
(proto.py).
"""

from __future__ import annotations

import hashlib
import logging
import os
import threading
from concurrent import futures
from typing import Dict, Optional

import grpc

from ...core import SonataError
from ...synth.batcher import DynamicBatcher
from ...synth.synthesizer import AudioOutputConfig, SonataSpeechSynthesizer
from .proto import (MESSAGES, MODE_LAZY, QUALITY_VALUES, RPCS, SERVICE_NAME)

log = logging.getLogger("sonata.grpc")

DEFAULT_PORT = 49314
REALTIME_CHUNK_SIZE = 55   # reference main.rs:383
REALTIME_CHUNK_PADDING = 3

__version__ = "0.1.0"


def _voice_id_for(config_path: str) -> str:
    """Deterministic voice id from the canonical config path (reference
    uses xxh3_64(path) truncated, main.rs:82-95)."""
    canon = os.path.realpath(config_path)
    h = hashlib.blake2b(canon.encode("utf-8"), digest_size=8).digest()
    return str(int.from_bytes(h, "little") % 10**13)


class _Voice:
    def __init__(self, voice_id: str, synth: SonataSpeechSynthesizer):
        self.voice_id = voice_id
        self.synth = synth
        # dynamic batcher: concurrent RPCs coalesce into padded GPU
        # batches (per-utterance seeding makes batching invisible)
        self.batcher = DynamicBatcher(synth.model)


class SonataGrpcService:
    """Implementation behind the generic handlers; one instance per server."""

    def __init__(self, device: Optional[str] = None):
        import torch

        self.device = device or (
            "cuda:0" if torch.cuda.is_available() else "cpu")
        self._voices: Dict[str, _Voice] = {}
        self._lock = threading.RLock()

    # ------------------------------------------------------------------ #
    # helpers
    # ------------------------------------------------------------------ #
    def _get(self, voice_id: str, context) -> _Voice:
        with self._lock:
            v = self._voices.get(voice_id)
        if v is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"voice not loaded: {voice_id}")
        return v

    def _voice_info(self, v: _Voice):
        model = v.synth.model
        info = model.audio_output_info()
        msg = MESSAGES["VoiceInfo"](
            voice_id=v.voice_id,
            synth_options=self._options_msg(v),
            audio=MESSAGES["AudioInfo"](
                sample_rate=info.sample_rate,
                num_channels=info.num_channels,
                sample_width=info.sample_width,
            ),
        )
        speakers = model.get_speakers() or {}
        for sid, name in speakers.items():
            msg.speakers[sid] = name
        if model.language:
            msg.language = model.language
        q = QUALITY_VALUES.get(getattr(model.config, "quality", ""), 0)
        if q:
            msg.quality = q
        msg.supports_streaming_output = bool(model.supports_streaming_output)
        return msg

    def _options_msg(self, v: _Voice):
        cfg = v.synth.get_synthesis_config()
        msg = MESSAGES["SynthesisOptions"](
            length_scale=cfg.length_scale,
            noise_scale=cfg.noise_scale,
            noise_w=cfg.noise_w,
        )
        speakers = v.synth.model.get_speakers() or {}
        if cfg.speaker_id is not None and cfg.speaker_id in speakers:
            msg.speaker = speakers[cfg.speaker_id]
        return msg

    @staticmethod
    def _speech_args_to_config(args) -> Optional[AudioOutputConfig]:
        if args is None:
            return None
        cfg = AudioOutputConfig(
            rate=args.rate if args.HasField("rate") else None,
            volume=args.volume if args.HasField("volume") else None,
            pitch=args.pitch if args.HasField("pitch") else None,
            appended_silence_ms=(args.appended_silence_ms
                                 if args.HasField("appended_silence_ms")
                                 else None),
        )
        if cfg.is_noop and cfg.appended_silence_ms is None:
            return None
        return cfg

    # ------------------------------------------------------------------ #
    # RPC implementations
    # ------------------------------------------------------------------ #
    def GetSonataVersion(self, request, context):
        return MESSAGES["Version"](version=__version__)

    def LoadVoice(self, request, context):
        voice_id = _voice_id_for(request.config_path)
        with self._lock:
            if voice_id in self._voices:  # idempotent (main.rs:96-108)
                return self._voice_info(self._voices[voice_id])
        try:
            from ...models.voice import load_voice

            voice = load_voice(request.config_path, device=self.device)
        except (SonataError, OSError, ValueError) as e:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          f"failed to load voice: {e}")
        v = _Voice(voice_id, SonataSpeechSynthesizer(voice))
        # warm graph captures in the background: the first real request
        # shouldn't pay the one-time capture cost (~36 ms measured)
        threading.Thread(target=voice.warmup, daemon=True).start()
        with self._lock:
            winner = self._voices.setdefault(voice_id, v)
        if winner is not v:
            # concurrent LoadVoice for the same config: release the
            # loser's batcher worker thread and let its weights be GC'd
            v.batcher.close()
            v = winner
        log.info("loaded voice %s from %s on %s", voice_id,
                 request.config_path, self.device)
        return self._voice_info(v)

    def GetVoiceInfo(self, request, context):
        return self._voice_info(self._get(request.voice_id, context))

    def GetSynthesisOptions(self, request, context):
        return self._options_msg(self._get(request.voice_id, context))

    def SetSynthesisOptions(self, request, context):
        v = self._get(request.voice_id, context)
        opts = request.synthesis_options
        cfg = v.synth.get_synthesis_config()
        if opts.HasField("speaker"):
            speakers = v.synth.model.get_speakers() or {}
            by_name = {name: sid for sid, name in speakers.items()}
            if opts.speaker in by_name:
                cfg.speaker_id = by_name[opts.speaker]
            else:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                              f"unknown speaker: {opts.speaker}")
        if opts.HasField("length_scale"):
            cfg.length_scale = opts.length_scale
        if opts.HasField("noise_scale"):
            cfg.noise_scale = opts.noise_scale
        if opts.HasField("noise_w"):
            cfg.noise_w = opts.noise_w
        v.synth.set_synthesis_config(cfg)
        return self._options_msg(v)

    def SynthesizeUtterance(self, request, context):
        v = self._get(request.voice_id, context)
        out_cfg = self._speech_args_to_config(
            request.speech_args if request.HasField("speech_args") else None)
        mode = request.synthesis_mode
        if mode == MODE_LAZY:
            it = v.synth.synthesize_lazy(request.text, out_cfg)
            for audio in it:
                yield MESSAGES["SynthesisResult"](
                    wav_samples=audio.as_wave_bytes(),
                    rtf=float(audio.real_time_factor),
                )
            return
        # default (parallel/batched): sentences go through the dynamic
        # batcher so CONCURRENT RPCs share padded GPU batches
        sentences = list(v.synth.model.phonemize_text(request.text))
        futures = [v.batcher.submit(sent) for sent in sentences]
        for f in futures:
            audio = f.result()
            if out_cfg is not None:
                audio = v.synth._post(audio, out_cfg)
            yield MESSAGES["SynthesisResult"](
                wav_samples=audio.as_wave_bytes(),
                rtf=float(audio.real_time_factor),
            )

    def SynthesizeUtteranceRealtime(self, request, context):
        v = self._get(request.voice_id, context)
        out_cfg = self._speech_args_to_config(
            request.speech_args if request.HasField("speech_args") else None)
        from ...audio.samples import to_i16_bytes

        for chunk in v.synth.synthesize_streamed(
                request.text, out_cfg,
                REALTIME_CHUNK_SIZE, REALTIME_CHUNK_PADDING):
            yield MESSAGES["WaveSamples"](wav_samples=to_i16_bytes(chunk))


def _generic_handler(service: SonataGrpcService) -> grpc.GenericRpcHandler:
    handlers = {}
    for name, (req_t, resp_t, streaming) in RPCS.items():
        req_cls, resp_cls = MESSAGES[req_t], MESSAGES[resp_t]
        method = getattr(service, name)
        if streaming:
            h = grpc.unary_stream_rpc_method_handler(
                method,
                request_deserializer=req_cls.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            )
        else:
            h = grpc.unary_unary_rpc_method_handler(
                method,
                request_deserializer=req_cls.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            )
        handlers[name] = h
    return grpc.method_handlers_generic_handler(SERVICE_NAME, handlers)


def create_server(
    port: Optional[int] = None,
    device: Optional[str] = None,
    max_workers: int = 16,
    reuse_port: bool = False,
):
    """Build a grpc.Server bound to 127.0.0.1 (reference binds loopback,
    main.rs:437-445).  Returns (server, bound_port, service).

    `reuse_port=True` sets SO_REUSEPORT so several server PROCESSES can
    share one port (the kernel load-balances connections): the escape
    from the single-process GIL ceiling on concurrent serving."""
    service = SonataGrpcService(device=device)
    opts = [("grpc.so_reuseport", 1 if reuse_port else 0)]
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers),
                         options=opts)
    server.add_generic_rpc_handlers((_generic_handler(service),))
    if port is None:
        port = int(os.environ.get("SONATA_GRPC_SERVER_PORT", DEFAULT_PORT))
    bound = server.add_insecure_port(f"127.0.0.1:{port}")
    return server, bound, service


def worker_device(i: int, device: Optional[str], n_gpus: int
                  ) -> Optional[str]:
    """GPU pin for multi-process worker i: round-robin across visible
    GPUs unless the caller pinned an explicit device."""
    if device is not None and device not in ("cuda", "auto"):
        return device  # explicit pin wins (e.g. "cuda:3", "cpu")
    if n_gpus > 0:
        return f"cuda:{i % n_gpus}"
    return device if device != "auto" else None


def serve(port: Optional[int] = None, device: Optional[str] = None,
          processes: int = 1) -> None:
    logging.basicConfig(
        level=os.environ.get("SONATA_GRPC", "INFO").upper())
    if processes > 1:
        # N identical server processes share the port via SO_REUSEPORT;
        # each holds its own voice copies (cheap next to 288 GB HBM) and
        # its own GIL, so handler throughput scales ~linearly.  On a
        # multi-GPU node the workers are pinned round-robin across the
        # visible GPUs (worker i -> cuda:{i % n_gpus}), turning the front
        # into the DP serving tier of baseline config #4: the kernel
        # load-balances connections, each GPU serves its own engines.
        import multiprocessing as mp

        import torch

        if port is None:
            port = int(os.environ.get("SONATA_GRPC_SERVER_PORT",
                                      DEFAULT_PORT))
        n_gpus = torch.cuda.device_count()
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_serve_one,
                             args=(port, worker_device(i, device, n_gpus)))
                 for i in range(1, processes)]
        for p in procs:
            p.start()
        try:
            _serve_one(port, worker_device(0, device, n_gpus))
        finally:
            for p in procs:
                p.terminate()
        return
    server, bound, _ = create_server(port=port, device=device)
    server.start()
    log.info("sonata_grpc serving on 127.0.0.1:%d", bound)
    server.wait_for_termination()


def _serve_one(port: int, device: Optional[str]) -> None:
    server, bound, _ = create_server(port=port, device=device,
                                     reuse_port=True)
    server.start()
    log.info("sonata_grpc worker serving on 127.0.0.1:%d (pid shared port)",
             bound)
    server.wait_for_termination()


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=None)
    ap.add_argument("--device", default=None)
    ap.add_argument("--processes", type=int, default=1)
    a = ap.parse_args()
    serve(port=a.port, device=a.device, processes=a.processes)
