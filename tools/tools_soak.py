"""Mixed-load stability soak: batcher + streaming + prosody across
co-resident voices for N seconds; reports op counts, errors, HBM peak."""
import json
import sys
import tempfile
import threading
import time

import torch

sys.path.insert(0, ".")
from sonata_amd.models import create_random_voice
from sonata_amd.models.voice import load_voice
from sonata_amd.synth.batcher import DynamicBatcher
from sonata_amd.synth.synthesizer import (AudioOutputConfig,
                                          SonataSpeechSynthesizer)

SECONDS = int(sys.argv[1]) if len(sys.argv) > 1 else 180
dev = "cuda:0"
d = tempfile.mkdtemp()
voices = [load_voice(create_random_voice(d, f"s{i}", quality=q, seed=i),
                     device=dev)
          for i, q in enumerate(["medium", "x_low", "medium"])]
for v in voices:
    v.warmup()
ops = [0, 0, 0]
errs = []
stop = time.time() + SECONDS


def worker(idx, fn):
    while time.time() < stop:
        try:
            fn()
            ops[idx] += 1
        except Exception as e:  # noqa: BLE001
            errs.append(repr(e))
            if len(errs) > 5:
                return


b = DynamicBatcher(voices[0])
synth = SonataSpeechSynthesizer(voices[2])
cfg = AudioOutputConfig(rate=60, volume=80)
threads = (
    [threading.Thread(target=worker, args=(0, lambda: b.synthesize(
        "ðɪs ɪz ə sˈoʊk tˈɛst sˈɛntəns.").samples.sum())) for _ in range(8)]
    + [threading.Thread(target=worker, args=(1, lambda: sum(
        len(c) for c in voices[1].stream_synthesis("sˈoʊkɪŋ ðə stɹˈim.",
                                                   45, 3)))) for _ in range(2)]
    + [threading.Thread(target=worker, args=(2, lambda: list(
        synth.synthesize_parallel("A prosody soak sentence here.", cfg))))
       for _ in range(2)]
)
t0 = time.time()
for t in threads:
    t.start()
for t in threads:
    t.join(timeout=SECONDS + 120)
print(json.dumps({
    "seconds": round(time.time() - t0, 1),
    "batcher_ops": ops[0], "stream_ops": ops[1], "prosody_ops": ops[2],
    "errors": errs[:5],
    "hbm_peak_gib": round(torch.cuda.max_memory_allocated() / 2**30, 2),
}))
b.close()
