"""B=1 streaming first-chunk latency with per-stage breakdown
(baseline ladder evidence)."""
import json
import sys
import tempfile
import time

import torch

sys.path.insert(0, ".")
from sonata_amd.models import create_random_voice
from sonata_amd.models.voice import load_voice
from sonata_amd.utils.trace import get_stage_times

d = tempfile.mkdtemp()
pack = create_random_voice(d, "lat", quality="medium")
v = load_voice(pack, device="cuda:0")
ph = ("ðɪs ɪz ə tˈɛst ʌv ðə stɹˈimɪŋ lˈeɪtənsi pˈæθweɪ wɪθ ə "
      "lˈɔŋɡɚ sˈɛntəns tu dɪkˈoʊd.")
for _ in range(3):
    for _ in v.stream_synthesis(ph, 45, 3):
        pass
get_stage_times().reset()
firsts, totals = [], []
for _ in range(10):
    t0 = time.perf_counter()
    chunks = []
    it = v.stream_synthesis(ph, 45, 3)
    chunks.append(next(it))
    firsts.append((time.perf_counter() - t0) * 1e3)
    for c in it:
        chunks.append(c)
    totals.append((time.perf_counter() - t0) * 1e3)
n_samp = sum(len(c) for c in chunks)
firsts.sort()
print(json.dumps({
    "first_chunk_ms_median": round(firsts[len(firsts) // 2], 2),
    "first_chunk_ms_min": round(firsts[0], 2),
    "total_ms_median": round(sorted(totals)[len(totals) // 2], 2),
    "audio_ms": round(n_samp / v.config.sample_rate * 1e3, 1),
    "stages": get_stage_times().snapshot(),
}))
