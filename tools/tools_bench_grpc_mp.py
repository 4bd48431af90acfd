"""Concurrent gRPC serving throughput: 1 vs N server processes sharing a
port (SO_REUSEPORT).  Spawn-safe (file-based __main__)."""
import json
import multiprocessing as mp
import sys
import tempfile
import time
from concurrent.futures import ThreadPoolExecutor

sys.path.insert(0, ".")

import grpc  # noqa: E402

from sonata_amd.frontends.grpc.client import SonataGrpcClient  # noqa: E402
from sonata_amd.frontends.grpc.proto import MESSAGES  # noqa: E402

TEXT = "Hello world, this is a moderately long test sentence for the system."


def worker(port, device):
    from sonata_amd.frontends.grpc.server import _serve_one

    _serve_one(port, device)


def measure(nproc: int, device: str, pack: str, port: int):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=worker, args=(port, device), daemon=True)
             for _ in range(nproc)]
    for p in procs:
        p.start()
    try:
        deadline = time.time() + 240
        while time.time() < deadline:
            try:
                c = SonataGrpcClient(f"127.0.0.1:{port}")
                c.LoadVoice(MESSAGES["VoicePath"](config_path=pack))
                c.close()
                break
            except grpc.RpcError:
                time.sleep(3)
        time.sleep(5)  # let remaining workers finish binding
        clients = [SonataGrpcClient(f"127.0.0.1:{port}") for _ in range(64)]
        vid = None
        for c in clients:
            try:
                vid = c.LoadVoice(
                    MESSAGES["VoicePath"](config_path=pack)).voice_id
            except grpc.RpcError:
                pass
        assert vid

        def one(i):
            tot = 0
            for r in clients[i % len(clients)].SynthesizeUtterance(
                    MESSAGES["Utterance"](voice_id=vid, text=TEXT)):
                tot += len(r.wav_samples)
            return tot

        with ThreadPoolExecutor(max_workers=64) as ex:
            list(ex.map(one, range(128)))  # warm every worker
        t0 = time.perf_counter()
        N = 512
        with ThreadPoolExecutor(max_workers=128) as ex:
            sizes = list(ex.map(one, range(N)))
        el = time.perf_counter() - t0
        print(json.dumps({
            "grpc_processes": nproc, "concurrency": 128, "utts": N,
            "audio_sec_per_s": round(sum(sizes) / 2 / 22050 / el, 1),
        }), flush=True)
        for c in clients:
            c.close()
    finally:
        for p in procs:
            p.terminate()
        time.sleep(2)


if __name__ == "__main__":
    import torch

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    from sonata_amd.models import create_random_voice

    d = tempfile.mkdtemp()
    pack = create_random_voice(
        d, "srv", quality="medium" if device.startswith("cuda") else "x_low")
    for nproc in [1, 4]:
        measure(nproc, device, pack, 49920 + nproc)
