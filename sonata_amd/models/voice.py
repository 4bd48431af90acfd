"""VitsVoice: a loaded voice pack implementing the SonataModel interface.

Parity: reference crates/sonata/models/piper/src/lib.rs —
`from_config_path` (:88-110), `VitsModel`/`VitsStreamingModel` (:291,480),
trait `VitsModelCommons` (:168-289: phonemize, id-encode, speaker maps,
fallback synthesis config), `SpeechStreamer` chunked decoding with
42-sample crossfade (:765-858).

Voice-pack format: `<name>.json` (Piper-compatible config schema) +
`<name>.safetensors` (weights).  A real Piper `.onnx` voice can be
converted via sonata_amd.models.onnx_import (weights-only importer).
"""

from __future__ import annotations

import os
import threading
import time
from typing import Dict, Iterator, List, Optional, Sequence

import numpy as np
import torch

from ..audio.samples import crossfade
from ..utils.trace import stage_timer
from ..core import Audio, AudioInfo, ModelError, Phonemes, SonataModel
from ..text.ids import phonemes_to_ids
from ..text.phonemizer import text_to_phonemes
from .chunker import chunk_plan
from .config import ModelConfig, SynthesisConfig
from .vits import VitsModel

CROSSFADE_SAMPLES = 42  # reference: piper/src/lib.rs:838


def _utterance_seed(phonemes: str, sid: Optional[int]) -> int:
    """Deterministic per-utterance seed: same text -> same audio on any
    rank (SURVEY.md §7 hard part 7: seed per utterance, not per rank)."""
    import hashlib

    h = hashlib.sha256(
        (phonemes + "|" + str(sid if sid is not None else -1)).encode("utf-8")
    ).digest()
    return int.from_bytes(h[:8], "little") & 0x7FFFFFFFFFFFFFFF


class VitsVoice(SonataModel):
    def __init__(
        self,
        config: ModelConfig,
        net: VitsModel,
        device: str = "cpu",
        dtype: torch.dtype = torch.float32,
        engine=None,
    ):
        self.config = config
        self.net = net.eval().to(device=device, dtype=dtype)
        self.device = torch.device(device)
        self.dtype = dtype
        # optional C++ VitsEngine runtime (csrc/engine): when attached,
        # speak_batch/stream_synthesis execute through it (same kernels,
        # same per-utterance seeds -> identical audio; no Python in the
        # graph loop).  NOTE: the engine holds its own weight copy — it
        # attaches at load time, after which parallel.broadcast_module
        # only affects self.net (ranks load identical packs anyway).
        self._engine = engine
        self._synth_config = config.default_synthesis_config()
        self._cfg_lock = threading.Lock()
        # Serializes GPU inference on THIS voice: the hipGraph caches
        # (capture + replay via shared capture buffers), the pinned
        # staging buffers of the stream pipeline, and the C++ engine's
        # lazily-built weight cache are all per-voice mutable state.
        # Concurrent callers on one voice otherwise race (a 4-minute
        # mixed-load soak surfaced rare hipErrorInvalidConfiguration).
        # Cross-voice concurrency is unaffected; the gRPC batcher
        # already funnels per-voice work through one thread.
        self._infer_lock = threading.Lock()
        self._tashkeel = None
        if config.espeak_voice.startswith("ar"):
            from ..text.tashkeel import TashkeelModel

            self._tashkeel = TashkeelModel.default(device="cpu")

    # ------------------------------------------------------------------ #
    # SonataModel interface
    # ------------------------------------------------------------------ #
    def audio_output_info(self) -> AudioInfo:
        return AudioInfo(sample_rate=self.config.sample_rate)

    @property
    def language(self) -> Optional[str]:
        return self.config.language_code

    def get_speakers(self) -> Optional[Dict[int, str]]:
        if self.config.num_speakers <= 1:
            return None
        return {v: k for k, v in self.config.speaker_id_map.items()}

    def get_synthesis_config(self) -> SynthesisConfig:
        with self._cfg_lock:
            return self._synth_config.copy()

    def set_synthesis_config(self, config: SynthesisConfig) -> None:
        with self._cfg_lock:
            self._synth_config = config.copy()

    def phonemize_text(self, text: str) -> Phonemes:
        with stage_timer("phonemize"):
            if self._tashkeel is not None:
                text = self._tashkeel.diacritize(text)
            sentences = text_to_phonemes(text, voice=self.config.espeak_voice)
            return Phonemes(sentences)

    # ------------------------------------------------------------------ #
    # inference
    # ------------------------------------------------------------------ #
    def _encode_ids(self, phonemes: str) -> List[int]:
        return phonemes_to_ids(phonemes, self.config.phoneme_id_map)

    def _generators(
        self, phonemes_batch: Sequence[str], sid: Optional[int]
    ) -> List[torch.Generator]:
        """One deterministic generator per utterance — same text gives the
        same audio regardless of batch composition or rank."""
        gens = []
        for p in phonemes_batch:
            gen = torch.Generator(device=self.device)
            gen.manual_seed(_utterance_seed(p, sid))
            gens.append(gen)
        return gens

    def _sid_tensor(self, batch: int, sid: Optional[int]):
        if self.config.num_speakers <= 1:
            return None
        s = sid if sid is not None else 0
        return torch.full((batch,), s, dtype=torch.long, device=self.device)

    @torch.no_grad()
    def speak_one_sentence(self, phonemes: str) -> Audio:
        return self.speak_batch([phonemes])[0]

    @torch.no_grad()
    def speak_batch(self, phonemes_batch: Sequence[str]) -> List[Audio]:
        """True padded [B, T] batching (the reference loops batch=1:
        piper/src/lib.rs:425-437 — batching is a headline improvement)."""
        cfg = self.get_synthesis_config()
        id_lists = [self._encode_ids(p) for p in phonemes_batch]
        B = len(id_lists)
        T = max((len(i) for i in id_lists), default=1)
        ids = torch.zeros((B, T), dtype=torch.long)
        lengths = torch.zeros((B,), dtype=torch.long)
        for b, il in enumerate(id_lists):
            ids[b, : len(il)] = torch.tensor(il, dtype=torch.long)
            lengths[b] = len(il)
        ids = ids.to(self.device)
        lengths = lengths.to(self.device)
        gens = self._generators(phonemes_batch, cfg.speaker_id)

        t0 = time.perf_counter()
        if self._engine is not None:
            seeds = [_utterance_seed(p, cfg.speaker_id)
                     for p in phonemes_batch]
            sid_t = self._sid_tensor(B, cfg.speaker_id)
            with self._infer_lock, stage_timer("infer", self.device):
                audio, audio_lengths = self._engine.infer(
                    ids, lengths, sid_t, cfg.noise_scale, cfg.length_scale,
                    cfg.noise_w, seeds)
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
            infer_ms = (time.perf_counter() - t0) * 1000.0
            info = self.audio_output_info()
            out: List[Audio] = []
            audio = audio.float().cpu().numpy()
            audio_lengths = audio_lengths.cpu().numpy()
            for b in range(B):
                n = int(audio_lengths[b])
                out.append(Audio(audio[b, 0, :n], info,
                                 inference_ms=infer_ms / B))
            return out
        with self._infer_lock, stage_timer("infer", self.device):
            audio, audio_lengths = self.net.infer(
            ids,
            lengths,
            sid=self._sid_tensor(B, cfg.speaker_id),
            noise_scale=cfg.noise_scale,
            length_scale=cfg.length_scale,
            noise_w=cfg.noise_w,
            generators=gens,
            )
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        infer_ms = (time.perf_counter() - t0) * 1000.0

        info = self.audio_output_info()
        out: List[Audio] = []
        audio = audio.float().cpu().numpy()
        audio_lengths = audio_lengths.cpu().numpy()
        for b in range(B):
            n = int(audio_lengths[b])
            out.append(Audio(audio[b, 0, :n], info, inference_ms=infer_ms / B))
        return out

    # ------------------------------------------------------------------ #
    # streaming decode (encoder once, HiFi-GAN chunked)
    # ------------------------------------------------------------------ #
    @property
    def supports_streaming_output(self) -> bool:
        return True

    @torch.no_grad()
    def stream_synthesis(
        self, phonemes: str, chunk_size: int = 45, chunk_padding: int = 3
    ) -> Iterator[np.ndarray]:
        """Yield waveform chunks: encoder runs once, the HiFi-GAN decoder
        runs per adaptive chunk with overlap-discard + crossfade seams
        (reference SpeechStreamer, piper/src/lib.rs:765-858)."""
        from ..utils.graphs import phase1_enabled

        cfg = self.get_synthesis_config()
        ids_l = self._encode_ids(phonemes)
        ids = torch.tensor([ids_l], dtype=torch.long, device=self.device)
        lengths = torch.tensor([len(ids_l)], dtype=torch.long, device=self.device)
        # the lock spans the whole stream (graph caches + pinned staging
        # are per-voice state); released when the generator finishes or
        # is closed/GC'd
        with self._infer_lock:
            if (phase1_enabled() and self.device.type == "cuda"
                    and self.config.num_speakers <= 1):
                # hipGraph-replayed encoder phase 1 (default ON: 9.1 ->
                # 6.8 ms first chunk at B=1; the sync-free spline made
                # capture legal).  Gated to single-speaker voices: the
                # captured phase-1 closure runs with g=None, which would
                # drop speaker conditioning.
                yield from self._stream_graphed(
                    phonemes, cfg, ids, lengths, chunk_size, chunk_padding)
                return
            yield from self._stream_eager(phonemes, cfg, ids, lengths,
                                          chunk_size, chunk_padding)

    @torch.no_grad()
    def _stream_eager(self, phonemes, cfg, ids, lengths, chunk_size,
                      chunk_padding):
        with stage_timer("encode", self.device):
            if self._engine is not None:
                # C++ engine encoder: same kernels/seeds, no per-launch
                # Python overhead on the first-chunk latency path
                z, y_mask, g = self._engine.infer_encoder(
                    ids, lengths, self._sid_tensor(1, cfg.speaker_id),
                    cfg.noise_scale, cfg.length_scale, cfg.noise_w,
                    [_utterance_seed(phonemes, cfg.speaker_id)])
                if g is not None and (not g.numel()):
                    g = None
            else:
                gens = self._generators([phonemes], cfg.speaker_id)
                z, y_mask, g = self.net.infer_encoder(
                    ids, lengths, sid=self._sid_tensor(1, cfg.speaker_id),
                    noise_scale=cfg.noise_scale,
                    length_scale=cfg.length_scale,
                    noise_w=cfg.noise_w, generators=gens,
                )
        yield from self._stream_decode(z, y_mask, g, chunk_size,
                                       chunk_padding)

    @torch.no_grad()
    def _stream_graphed(self, phonemes, cfg, ids, lengths, chunk_size,
                        chunk_padding):
        """Realtime path with hipGraph-captured encoder phase 1 (text
        encoder + duration predictor) per padded-T bucket; phase 2 (frame
        expansion, flow) stays eager (F is data-dependent); chunk decode
        replays per-shape graphs.  Padding invariance is exact (masked
        everywhere), so bucketing ids costs nothing numerically."""
        from ..utils.graphs import TupleGraphCache
        from .vits import masked_noise_rows

        T = ids.shape[1]
        Tpad = (T + 31) // 32 * 32
        if Tpad != T:
            ids = torch.nn.functional.pad(ids, (0, Tpad - T))
        gens = self._generators([phonemes], cfg.speaker_id)
        noise = masked_noise_rows(1, 2, Tpad, lengths, gens,
                                  self.device, self.dtype)
        noise = noise * cfg.noise_w  # pre-scale: graph runs noise_w=1
        if not hasattr(self, "_phase1_graphs"):
            self._phase1_graphs = TupleGraphCache(
                lambda i, l, n: self.net.encode_phase1(i, l, None, 1.0, n))
        with stage_timer("encode_graph", self.device):
            x, m_p, logs_p, x_mask, logw = self._phase1_graphs(
                ids, lengths, noise)
            z, y_mask, g = self.net.encode_phase2(
                m_p, logs_p, x_mask, logw, None, cfg.noise_scale,
                cfg.length_scale, gens)
        yield from self._stream_decode(z, y_mask, g, chunk_size,
                                       chunk_padding)

    @torch.no_grad()
    def warmup(self) -> None:
        """Pay one-time costs (hipGraph captures for the default stream
        path, kernel/module caches) at LOAD time so the first real
        request doesn't (cold first request measured ~36 ms of capture;
        profiles/r02_modes_final.json).  Safe no-op on CPU."""
        if self.device.type != "cuda":
            return
        try:
            for _ in self.stream_synthesis("wˈɔːm ˈʌp sˈɛntəns.", 45, 3):
                pass
            self.speak_one_sentence("wˈɔːm.")
        except Exception:  # warmup must never break loading
            pass

    def _stream_decode(self, z, y_mask, g, chunk_size: int,
                       chunk_padding: int) -> Iterator[np.ndarray]:
        from ..utils.graphs import DecodeGraphCache, enabled as graphs_on

        graph_cache = None
        if (graphs_on() and self.device.type == "cuda" and g is None):
            # hipGraph replay per chunk shape (launch-bound at small B);
            # decode is capture-safe (no host syncs inside)
            if not hasattr(self, "_decode_graphs"):
                if self._engine is not None:
                    self._decode_graphs = DecodeGraphCache(
                        lambda zc, mc: self._engine.decode(zc, mc, None,
                                                           None))
                else:
                    self._decode_graphs = DecodeGraphCache(
                        lambda zc, mc: self.net.decode(zc, mc, None))
            graph_cache = self._decode_graphs
        hop = self.net.arch.hop_length
        num_frames = z.shape[-1]
        # Overlap-crossfade at seams without changing the timeline: each
        # interior chunk keeps `ext` extra decoded samples (taken from its
        # right padding region); the next chunk's first `ext` samples cover
        # the same timeline window, so the sine-ramp mix preserves total
        # length exactly (streamed output == one-shot length).
        tail: Optional[np.ndarray] = None
        prev_ext = 0

        def decode_one(spec):
            z_c = z[:, :, spec.mel_start : spec.mel_end].contiguous()
            m_c = y_mask[:, :, spec.mel_start : spec.mel_end].contiguous()
            with stage_timer("decode_chunk", self.device):
                if graph_cache is not None:
                    return graph_cache(z_c, m_c)
                if self._engine is not None:
                    return self._engine.decode(z_c, m_c, g, None)
                return self.net.decode(z_c, m_c, g)

        def trim_and_emit(spec, wav):
            nonlocal tail, prev_ext
            lo = spec.trim_left_frames * hop
            hi = len(wav) - spec.trim_right_frames * hop
            ext = 0 if spec.is_last else min(
                CROSSFADE_SAMPLES, spec.trim_right_frames * hop
            )
            cur = wav[lo : hi + ext]
            if tail is not None:
                cur = crossfade(tail, cur, prev_ext)
            cut = len(cur) - ext
            tail_next = cur[cut:] if ext else None
            out = cur[:cut].astype(np.float32)
            tail = tail_next
            prev_ext = ext
            return out

        if self.device.type != "cuda":
            for spec in chunk_plan(num_frames, chunk_size, chunk_padding):
                wav = decode_one(spec)[0, 0].float().cpu().numpy()
                yield trim_and_emit(spec, wav)
                if spec.is_last:
                    return
            return

        # GPU: 1-deep pipeline — chunk i+1's decode is enqueued before
        # chunk i's host copy is consumed, so GPU decode overlaps the
        # D2H transfer + Python trim/crossfade/emit of the previous
        # chunk.  Double-buffered pinned staging; stream order makes the
        # graph-replay output safe (the copy is enqueued before the next
        # replay can overwrite its capture buffer).
        if not hasattr(self, "_pin_bufs"):
            max_samples = (1024 + 2 * 16) * hop
            self._pin_bufs = [
                torch.empty(max_samples, dtype=torch.float32,
                            pin_memory=True) for _ in range(2)]
            self._pin_events = [torch.cuda.Event(), torch.cuda.Event()]
        pending = None  # (spec, buf_idx, n_samples)
        which = 0
        first = True
        for spec in chunk_plan(num_frames, chunk_size, chunk_padding):
            audio = decode_one(spec)
            n = audio.shape[-1]
            buf = self._pin_bufs[which]
            if n > buf.shape[0]:  # defensive: unexpected chunk size
                buf = torch.empty(n, dtype=torch.float32, pin_memory=True)
                self._pin_bufs[which] = buf
            buf[:n].copy_(audio[0, 0].float(), non_blocking=True)
            self._pin_events[which].record()
            if first:
                # latency priority: emit chunk 0 before enqueuing chunk
                # 1's ~150 eager launches; pipeline from chunk 1 on
                self._pin_events[which].synchronize()
                wav = self._pin_bufs[which][:n].numpy().copy()
                yield trim_and_emit(spec, wav)
                first = False
            else:
                if pending is not None:
                    pspec, pwhich, pn = pending
                    self._pin_events[pwhich].synchronize()
                    wav = self._pin_bufs[pwhich][:pn].numpy().copy()
                    yield trim_and_emit(pspec, wav)
                pending = (spec, which, n)
            which ^= 1
            if spec.is_last:
                break
        if pending is not None:
            pspec, pwhich, pn = pending
            self._pin_events[pwhich].synchronize()
            wav = self._pin_bufs[pwhich][:pn].numpy().copy()
            yield trim_and_emit(pspec, wav)


# --------------------------------------------------------------------------- #
# loading / creation
# --------------------------------------------------------------------------- #
def _weights_path_for(config_path: str) -> str:
    stem = config_path
    if stem.endswith(".json"):
        stem = stem[: -len(".json")]
    if stem.endswith(".onnx"):
        stem = stem[: -len(".onnx")]
    return stem + ".safetensors"


def load_voice(
    config_path: str, device: str = "cpu",
    dtype: Optional[torch.dtype] = None, engine: str = "auto"
) -> VitsVoice:
    """Load a voice pack: `<stem>.json` + `<stem>.safetensors`.

    Mirrors the reference loader dispatch (piper/src/lib.rs:88-110); the
    `streaming` config key only changes default synthesis mode — the same
    net serves one-shot and streaming here (encoder/decoder split is a
    method boundary, not two files)."""
    config = ModelConfig.from_json_path(config_path)
    if dtype is None:
        dtype = torch.bfloat16 if device.startswith("cuda") else torch.float32
    net = VitsModel(config.num_symbols, config.architecture,
                    n_speakers=max(config.num_speakers, 1))
    wpath = _weights_path_for(config_path)
    if os.path.exists(wpath):
        from safetensors.torch import load_file

        state = load_file(wpath)
        missing, unexpected = net.load_state_dict(state, strict=False)
        if missing or unexpected:
            raise ModelError(
                f"voice weights mismatch: missing={missing[:5]} "
                f"unexpected={unexpected[:5]}"
            )
    else:
        raise ModelError(f"voice weights not found: {wpath}")
    # serving runtime: the C++ VitsEngine by default (same kernels, same
    # per-utterance seeds -> identical audio); SONATA_ENGINE=python or
    # engine="python" keeps the torch-module path.
    #
    # warmup(): see VitsVoice.warmup — capture/compile costs paid at
    # load time instead of the first request.
    eng = None
    choice = os.environ.get("SONATA_ENGINE", engine)
    if choice == "auto" and not device.startswith("cuda"):
        choice = "python"  # CPU stays on the torch oracle path
    if choice in ("auto", "cpp"):
        try:
            from ..ops import hip_ext

            ext = hip_ext(required=False)
            if ext is not None and hasattr(ext, "VitsEngine"):
                eng = ext.VitsEngine(
                    config_path, device,
                    "bf16" if (dtype == torch.bfloat16) else "f32")
        except Exception:
            if choice == "cpp":
                raise
            eng = None
    return VitsVoice(config, net, device=device, dtype=dtype, engine=eng)


def create_random_voice(
    out_dir: str,
    name: str = "test_voice",
    quality: str = "medium",
    language: str = "en-us",
    num_speakers: int = 1,
    seed: int = 0,
) -> str:
    """Create a random-init voice pack (tests / synthetic benches — there is
    no network for real checkpoints).  Returns the config path."""
    from .config import QUALITY_PRESETS, VitsArchitecture

    preset = QUALITY_PRESETS[quality]
    arch = VitsArchitecture(**preset["arch"])
    if num_speakers > 1:
        arch.gin_channels = 256
    config = ModelConfig(
        key=name,
        language_code=language,
        sample_rate=preset["sample_rate"],
        quality=quality,
        num_speakers=num_speakers,
        speaker_id_map={f"spk{i}": i for i in range(num_speakers)}
        if num_speakers > 1 else {},
        espeak_voice=language,
        architecture=arch,
    )
    torch.manual_seed(seed)
    net = VitsModel(config.num_symbols, arch, n_speakers=max(num_speakers, 1))
    os.makedirs(out_dir, exist_ok=True)
    cfg_path = os.path.join(out_dir, f"{name}.json")
    config.save_json(cfg_path)
    from safetensors.torch import save_file

    save_file(net.state_dict(), _weights_path_for(cfg_path))
    return cfg_path
