"""Stream-mode benchmark harness — the counterpart of the reference's
divan benches (crates/sonata/synth/src/benchmarks.rs): full-stream wall
time and FIRST-CHUNK latency for lazy / parallel / realtime modes, plus
prosody post-processing cost.  Prints one JSON line per measurement."""
import json
import sys
import tempfile
import time

sys.path.insert(0, ".")

import torch

from sonata_amd.models import create_random_voice
from sonata_amd.models.voice import load_voice
from sonata_amd.synth.synthesizer import AudioOutputConfig, SonataSpeechSynthesizer

dev = "cuda:0" if torch.cuda.is_available() else "cpu"
d = tempfile.mkdtemp()
voice = load_voice(create_random_voice(d, "modes", quality="medium"
                                       if dev.startswith("cuda") else "x_low"),
                   device=dev)
synth = SonataSpeechSynthesizer(voice)
TEXT = ("Hello there everyone. This is a longer paragraph of text. "
        "It contains several sentences of varying length. "
        "The streaming modes chunk it differently. Goodbye now.")

# warmup: parallel AND one full realtime pass (graph captures are a
# one-time first-request cost, reported separately as cold_first below)
list(synth.synthesize_parallel(TEXT))
t_cold = time.perf_counter()
cold_first = None
for _c in synth.synthesize_streamed(TEXT, chunk_size=45, chunk_padding=3):
    if cold_first is None:
        cold_first = (time.perf_counter() - t_cold) * 1000
print(json.dumps({"mode": "realtime-cold-first-request",
                  "first_chunk_ms": round(cold_first or 0, 2)}))

N = 5
for mode in ["lazy", "parallel", "realtime"]:
    t0 = time.perf_counter()
    first = None
    total_audio = 0.0
    for _ in range(N):
        t1 = time.perf_counter()
        if mode == "lazy":
            it = synth.synthesize_lazy(TEXT)
        elif mode == "parallel":
            it = synth.synthesize_parallel(TEXT)
        else:
            it = synth.synthesize_streamed(TEXT, chunk_size=45,
                                           chunk_padding=3)
        got_first = False
        for item in it:
            if not got_first:
                got_first = True
                if first is None:
                    first = time.perf_counter() - t1
            samples = item if mode == "realtime" else item.samples
            total_audio += len(samples) / voice.config.sample_rate
    el = time.perf_counter() - t0
    print(json.dumps({
        "mode": mode, "device": dev,
        "stream_wall_ms": round(el * 1000 / N, 2),
        "first_chunk_ms": round((first or 0) * 1000, 2),
        "audio_sec_per_s": round(total_audio / el, 1),
    }))

# prosody post-processing (sonic-equivalent WSOLA) cost
cfg = AudioOutputConfig(rate=65, volume=80, pitch=55)
t0 = time.perf_counter()
for _ in range(N):
    list(synth.synthesize_parallel(TEXT, cfg))
el = time.perf_counter() - t0
print(json.dumps({"mode": "parallel+prosody", "device": dev,
                  "stream_wall_ms": round(el * 1000 / N, 2)}))
