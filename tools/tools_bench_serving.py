"""Baseline-ladder measurements beyond the flagship bench.py:

  #2  single-utterance RTF, medium, bf16, 1 GPU
  #3  batch=64 streaming chunked HiFi-GAN decode (high quality), 1 GPU:
      time-to-first-audio + steady-state audio-sec/s
  (#4 8-GPU scaling is the driver's SCALE run of bench.py)

Prints one JSON line per config.  Synthetic phonemes, random-init voice
(no network for checkpoints)."""
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from sonata_amd.models.config import ModelConfig, QUALITY_PRESETS, VitsArchitecture
from sonata_amd.models.vits import VitsModel
from sonata_amd.models.chunker import chunk_plan
from sonata_amd.text.ids import default_phoneme_id_map, num_symbols

dev = "cuda:0" if torch.cuda.is_available() else "cpu"
dt = torch.bfloat16 if dev.startswith("cuda") else torch.float32


def make_net(quality):
    preset = QUALITY_PRESETS[quality]
    arch = VitsArchitecture(**preset["arch"])
    torch.manual_seed(0)
    net = VitsModel(num_symbols(default_phoneme_id_map()), arch).eval()
    return net.to(dev, dt), preset["sample_rate"], arch


def make_ids(batch, seq, seed=0):
    g = torch.Generator().manual_seed(seed)
    nv = num_symbols(default_phoneme_id_map())
    ids = torch.zeros(batch, seq, dtype=torch.long)
    ids[:, 1:-1:2] = torch.randint(3, nv, (batch, (seq - 1) // 2), generator=g)
    ids[:, 0] = 1
    ids[:, -1] = 2
    return ids.to(dev), torch.full((batch,), seq, dtype=torch.long, device=dev)


def sync():
    if dev.startswith("cuda"):
        torch.cuda.synchronize()


# ---- config #2: single-utterance RTF, medium --------------------------- #
net, sr, arch = make_net("medium")
ids, lens = make_ids(1, 128, seed=1)
with torch.no_grad():
    for _ in range(3):
        audio, alens = net.infer(ids, lens)
    sync()
    t0 = time.perf_counter()
    N = 20
    total_audio = 0.0
    for _ in range(N):
        audio, alens = net.infer(ids, lens)
        total_audio += float(alens.sum()) / sr
    sync()
    el = time.perf_counter() - t0
print(json.dumps({
    "config": "#2 single-utterance medium bf16 1GPU",
    "rtf": round(el / total_audio, 6),
    "audio_sec_per_s": round(total_audio / el, 1),
    "ms_per_utt": round(el * 1000 / N, 2),
}))

# ---- config #3: batch=64 streaming chunked decode, high ---------------- #
net, sr, arch = make_net("high")
B = 64
ids, lens = make_ids(B, 256, seed=2)
hop = arch.hop_length
with torch.no_grad():
    z, y_mask, g = net.infer_encoder(ids, lens)
    sync()
    # streamed: decode adaptive chunks of z across the whole batch
    for warm in range(2):
        t0 = time.perf_counter()
        first_chunk_s = None
        total = 0
        F = z.shape[-1]
        for spec in chunk_plan(F, 45, 3):
            zc = z[:, :, spec.mel_start:spec.mel_end]
            mc = y_mask[:, :, spec.mel_start:spec.mel_end]
            a = net.decode(zc, mc, g)
            lo = spec.trim_left_frames * hop
            hi = a.shape[-1] - spec.trim_right_frames * hop
            total += (hi - lo) * B
            sync()
            if first_chunk_s is None:
                first_chunk_s = time.perf_counter() - t0
        el = time.perf_counter() - t0
audio_sec = total / sr
print(json.dumps({
    "config": "#3 batch=64 streaming chunked decode, high, 1GPU",
    "time_to_first_audio_ms": round(first_chunk_s * 1000, 2),
    "decode_audio_sec_per_s": round(audio_sec / el, 1),
    "chunks": len(list(chunk_plan(z.shape[-1], 45, 3))),
    "frames": int(z.shape[-1]),
}))

# ---- one-shot batch=64 high (for comparison) --------------------------- #
with torch.no_grad():
    for _ in range(2):
        audio, alens = net.infer(ids, lens)
    sync()
    t0 = time.perf_counter()
    N = 5
    tot = 0.0
    for _ in range(N):
        audio, alens = net.infer(ids, lens)
        tot += float(alens.sum()) / sr
    sync()
    el = time.perf_counter() - t0
print(json.dumps({
    "config": "#3b batch=64 one-shot high 1GPU",
    "audio_sec_per_s": round(tot / el, 1),
    "ms_per_step": round(el * 1000 / N, 1),
}))
