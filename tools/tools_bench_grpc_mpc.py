"""Serving capacity with MULTI-PROCESS clients (the single-process
client harness was itself the bottleneck: one GIL deserializing 113 MB
of responses).  N server procs (SO_REUSEPORT) x M client procs."""
import json
import multiprocessing as mp
import sys
import tempfile
import time

sys.path.insert(0, ".")


def client_proc(port, vid, n_utts, conc, q, bar):
    import grpc  # noqa: E402

    from sonata_amd.frontends.grpc.client import SonataGrpcClient
    from sonata_amd.frontends.grpc.proto import MESSAGES
    from concurrent.futures import ThreadPoolExecutor

    TEXT = ("Hello world, this is a moderately long test sentence "
            "for the system.")
    clients = [SonataGrpcClient(f"127.0.0.1:{port}") for _ in range(conc)]

    def one(i):
        tot = 0
        for r in clients[i % conc].SynthesizeUtterance(
                MESSAGES["Utterance"](voice_id=vid, text=TEXT)):
            tot += len(r.wav_samples)
        return tot

    with ThreadPoolExecutor(max_workers=conc) as ex:
        list(ex.map(one, range(conc * 2)))  # warm
    bar.wait()  # all clients warm before anyone starts the timed run
    t0 = time.perf_counter()
    with ThreadPoolExecutor(max_workers=conc) as ex:
        sizes = list(ex.map(one, range(n_utts)))
    el = time.perf_counter() - t0
    q.put((sum(sizes), el))
    for c in clients:
        c.close()


def main():
    import torch

    from sonata_amd.models import create_random_voice

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    d = tempfile.mkdtemp()
    pack = create_random_voice(d, "srv", quality="medium")
    port = 49931
    ctx = mp.get_context("spawn")

    from sonata_amd.frontends.grpc.server import _serve_one
    import os
    n_srv = int(os.environ.get("SRV_PROCS", "1"))
    servers = [ctx.Process(target=_serve_one, args=(port, device),
                           daemon=True) for _ in range(n_srv)]
    for p in servers:
        p.start()
    time.sleep(8)
    import grpc

    from sonata_amd.frontends.grpc.client import SonataGrpcClient
    from sonata_amd.frontends.grpc.proto import MESSAGES

    vid = None
    for _ in range(40):
        try:
            c = SonataGrpcClient(f"127.0.0.1:{port}")
            vid = c.LoadVoice(
                MESSAGES["VoicePath"](config_path=pack)).voice_id
            c.close()
        except grpc.RpcError:
            time.sleep(3)
            continue
        break
    # make sure every reuseport worker has the voice
    for _ in range(8):
        c = SonataGrpcClient(f"127.0.0.1:{port}")
        c.LoadVoice(MESSAGES["VoicePath"](config_path=pack))
        c.close()
    assert vid
    for n_cli in (1, 4):
        q = ctx.SimpleQueue()
        bar = ctx.Barrier(n_cli + 1)
        per = 512 // n_cli
        procs = [ctx.Process(target=client_proc,
                             args=(port, vid, per, 32, q, bar))
                 for _ in range(n_cli)]
        for p in procs:
            p.start()
        bar.wait()  # released once every client finished its warmup
        t0 = time.perf_counter()
        results = [q.get() for _ in range(n_cli)]
        for p in procs:
            p.join(timeout=300)
        wall = time.perf_counter() - t0
        total_bytes = sum(r[0] for r in results)
        print(json.dumps({
            "client_procs": n_cli, "server_procs": n_srv,
            "utts": per * n_cli,
            "audio_sec_per_s": round(total_bytes / 2 / 22050 / wall, 1),
        }), flush=True)
    for p in servers:
        p.terminate()


if __name__ == "__main__":
    main()
