"""Baseline ladder config #5: multi-voice serving — ar (tashkeel) + de +
a fleet of en voices co-resident in HBM, served interleaved through the
gRPC frontend on loopback.  Prints one JSON summary line."""
import json
import sys
import tempfile
import time

sys.path.insert(0, ".")

import torch

from sonata_amd.frontends.grpc import create_server
from sonata_amd.frontends.grpc.client import SonataGrpcClient
from sonata_amd.frontends.grpc.proto import MESSAGES
from sonata_amd.models import create_random_voice

dev = "cuda:0" if torch.cuda.is_available() else "cpu"
d = tempfile.mkdtemp()
N_EN = 14
packs = [create_random_voice(d, "ar_JO_v", quality="medium", language="ar"),
         create_random_voice(d, "de_DE_v", quality="medium", language="de")]
packs += [create_random_voice(d, f"en_v{i}", quality="medium", seed=i)
          for i in range(N_EN)]

server, port, _ = create_server(port=0, device=dev)
server.start()
client = SonataGrpcClient(f"127.0.0.1:{port}")
t0 = time.perf_counter()
vids = [client.LoadVoice(MESSAGES["VoicePath"](config_path=p)).voice_id
        for p in packs]
load_s = time.perf_counter() - t0

texts = {
    "ar": "مرحبا بالعالم هذا اختبار طويل نسبيا للنظام.",
    "de": "Hallo Welt, das ist ein längerer Testsatz für das System.",
    "en": "Hello world, this is a moderately long test sentence for the system.",
}


def text_for(i):
    if i == 0:
        return texts["ar"]
    if i == 1:
        return texts["de"]
    return texts["en"]


# warmup one pass
for i, vid in enumerate(vids):
    list(client.SynthesizeUtterance(MESSAGES["Utterance"](
        voice_id=vid, text=text_for(i))))

t0 = time.perf_counter()
total_bytes = 0
ROUNDS = 4
for _ in range(ROUNDS):
    for i, vid in enumerate(vids):
        for r in client.SynthesizeUtterance(MESSAGES["Utterance"](
                voice_id=vid, text=text_for(i))):
            total_bytes += len(r.wav_samples)
el = time.perf_counter() - t0
audio_sec = total_bytes / 2 / 22050
mem = (torch.cuda.max_memory_allocated() / 2**30
       if dev.startswith("cuda") else 0)
print(json.dumps({
    "config": "#5 multi-voice gRPC: ar(tashkeel)+de+%d en co-resident" % N_EN,
    "voices": len(vids), "device": dev,
    "load_s": round(load_s, 2),
    "utterances": ROUNDS * len(vids),
    "audio_sec_per_s": round(audio_sec / el, 1),
    "hbm_peak_gib": round(mem, 2),
}))
client.close()
server.stop(grace=None)
