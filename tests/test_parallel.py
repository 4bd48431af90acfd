"""Multi-process data-parallel tests (gloo backend, world_size=2, CPU) —
the distributed path must be correct by construction here; the driver
runs the real RCCL/8-GPU scaling bench (SURVEY.md §4 additions)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _run_dp(rank, world, port, pack, q):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    try:
        import torch.distributed as dist

        from sonata_amd.models.voice import load_voice
        from sonata_amd.parallel import (DistributedSynthesizer,
                                         broadcast_module,
                                         gather_audio_to_rank0,
                                         init_distributed,
                                         shard_round_robin)

        init_distributed(backend="gloo")
        voice = load_voice(pack, device="cpu")

        # perturb non-zero ranks' weights, then broadcast from rank 0:
        # all ranks must end bit-identical
        if rank != 0:
            with torch.no_grad():
                for p in voice.net.parameters():
                    p.add_(1.0)
        broadcast_module(voice.net, src=0)
        checksum = float(sum(p.double().sum() for p in voice.net.parameters()))

        # sharding is a partition
        idx0 = shard_round_robin(5, 0, world)
        idx1 = shard_round_robin(5, 1, world)
        assert sorted(idx0 + idx1) == [0, 1, 2, 3, 4]

        # corpus synthesis: rank 0 gets all 5 utterances in order
        ds = DistributedSynthesizer(voice, batch_size=2)
        phon = ["wˈʌn.", "tˈuː θɹˈiː.", "fˈoːɹ.", "fˈaɪv sˈɪks.", "sˈɛvən."]
        out = ds.synthesize_corpus(phon)
        if rank == 0:
            assert out is not None and len(out) == 5
            assert all(len(o) > 100 for o in out)
            # determinism: utterance 0 synthesized locally matches the
            # distributed result regardless of which rank produced it
            ref = voice.speak_one_sentence(phon[1]).samples
            np.testing.assert_allclose(out[1], ref, atol=1e-5)
        else:
            assert out is None

        # gather handles an empty rank (n_total < world contributions)
        single = gather_audio_to_rank0(
            [np.ones(3, np.float32)] if rank == 0 else [],
            [0] if rank == 0 else [], 1, torch.device("cpu"))
        if rank == 0:
            assert len(single) == 1 and single[0].shape == (3,)

        dist.barrier()
        dist.destroy_process_group()
        q.put(("ok", rank, checksum))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put(("err", rank, traceback.format_exc() + str(e)))


def test_dp_world2(tmp_path):
    from sonata_amd.models import create_random_voice

    pack = create_random_voice(str(tmp_path), "dp_voice", quality="x_low")
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29511
    procs = [ctx.Process(target=_run_dp, args=(r, 2, port, pack, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
    errs = [r for r in results if r[0] != "ok"]
    assert not errs, errs
    # broadcast made weights identical
    assert abs(results[0][2] - results[1][2]) < 1e-6


def test_dp_world4_gather(tmp_path):
    """World-size-4 gather correctness (uneven shards, bigger world)."""
    import torch.multiprocessing as mp2

    from sonata_amd.models import create_random_voice

    pack = create_random_voice(str(tmp_path), "dp4", quality="x_low")
    ctx = mp2.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_run_dp4, args=(r, 4, 29621, pack, q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(4)]
    for p in procs:
        p.join(timeout=120)
    errs = [r for r in results if r[0] != "ok"]
    assert not errs, errs


def _run_dp4(rank, world, port, pack, q):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    try:
        import torch.distributed as dist

        from sonata_amd.models.voice import load_voice
        from sonata_amd.parallel import (DistributedSynthesizer,
                                         init_distributed)

        init_distributed(backend="gloo")
        voice = load_voice(pack, device="cpu")
        ds = DistributedSynthesizer(voice, batch_size=3)
        phon = [f"wˈʌn {'tˈuː ' * (1 + i % 3)}." for i in range(7)]
        out = ds.synthesize_corpus(phon)
        if rank == 0:
            assert out is not None and len(out) == 7
            assert all(len(o) > 100 for o in out)
        else:
            assert out is None
        dist.barrier()
        dist.destroy_process_group()
        q.put(("ok", rank))
    except Exception:
        import traceback

        q.put(("err", rank, traceback.format_exc()))


def test_shard_round_robin_uneven():
    from sonata_amd.parallel import shard_round_robin

    # partition property for any world size / count
    for n, world in [(5, 3), (1, 8), (0, 2), (17, 4)]:
        all_idx = sorted(sum((shard_round_robin(n, r, world)
                              for r in range(world)), []))
        assert all_idx == list(range(n))


def _run_w8(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    try:
        import torch.distributed as dist

        from sonata_amd.parallel import (broadcast_module,
                                         gather_audio_to_rank0,
                                         init_distributed,
                                         shard_round_robin)

        init_distributed(backend="gloo")

        # mixed-dtype module: int64 buffer with value > 2^24 must survive
        # broadcast bit-exactly (regression: f32 round-trip corrupted it)
        mod = torch.nn.Linear(4, 4)
        mod.register_buffer("step", torch.tensor([2 ** 40 + 12345],
                                                 dtype=torch.long))
        mod.register_buffer("scale64", torch.tensor([1.0 + 2 ** -40],
                                                    dtype=torch.float64))
        if rank != 0:
            with torch.no_grad():
                for p in mod.parameters():
                    p.mul_(0.0)
            mod.step.fill_(-1)
            mod.scale64.fill_(0.0)
        broadcast_module(mod, src=0)
        assert int(mod.step[0]) == 2 ** 40 + 12345, int(mod.step[0])
        assert float(mod.scale64[0]) == 1.0 + 2 ** -40

        # 512-utterance-shaped gather (tiny payloads, full header math):
        # every rank contributes ceil/floor shards, rank 0 reassembles
        n = 37
        mine = shard_round_robin(n, rank, world)
        pieces = [np.full(3 + (i % 5), float(i), np.float32) for i in mine]
        out = gather_audio_to_rank0(pieces, mine, n, torch.device("cpu"))
        if rank == 0:
            assert len(out) == n
            for i, o in enumerate(out):
                assert len(o) == 3 + (i % 5) and float(o[0]) == float(i)
        else:
            assert out is None
        dist.barrier()
        dist.destroy_process_group()
        q.put(("ok", rank))
    except Exception:
        import traceback

        q.put(("err", rank, traceback.format_exc()))


def test_dp_world8_collectives():
    """World-size-8 rehearsal of the exact collective sequence the 8-GPU
    bench/serving path issues (broadcast buckets + header/payload
    all_gather), gloo on CPU."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_run_w8, args=(r, 8, 29631, q))
             for r in range(8)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(8)]
    for p in procs:
        p.join(timeout=180)
    errs = [r for r in results if r[0] != "ok"]
    assert not errs, errs


def test_grpc_worker_device_round_robin():
    from sonata_amd.frontends.grpc.server import worker_device

    # 8 workers, 8 GPUs: one each
    assert [worker_device(i, None, 8) for i in range(8)] == [
        f"cuda:{i}" for i in range(8)]
    # 16 workers, 8 GPUs: two per GPU
    assert worker_device(9, "cuda", 8) == "cuda:1"
    # explicit pin wins; CPU-only box stays None/cpu
    assert worker_device(3, "cuda:5", 8) == "cuda:5"
    assert worker_device(3, "cpu", 8) == "cpu"
    assert worker_device(3, None, 0) is None
    assert worker_device(3, "auto", 0) is None


def _run_gpu_w2(rank, world, port, pack, q):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "LOCAL_RANK": "0",  # one physical GPU: both ranks share cuda:0
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    try:
        import torch.distributed as dist

        from sonata_amd.models.voice import load_voice
        from sonata_amd.parallel import (DistributedSynthesizer,
                                         init_distributed)

        init_distributed(backend="gloo")
        voice = load_voice(pack, device="cuda:0")
        ds = DistributedSynthesizer(voice, batch_size=4)
        phon = [f"wˈʌn {'tˈuː ' * (1 + i % 3)}." for i in range(9)]
        out = ds.synthesize_corpus(phon)
        if rank == 0:
            assert out is not None and len(out) == 9
            assert all(len(o) > 100 for o in out)
            ref = voice.speak_one_sentence(phon[2]).samples
            np.testing.assert_allclose(out[2], ref, atol=1e-3)
        else:
            assert out is None
        dist.barrier()
        dist.destroy_process_group()
        q.put(("ok", rank))
    except Exception:
        import traceback

        q.put(("err", rank, traceback.format_exc()))


@pytest.mark.gpu
def test_dp_world2_cuda_compute_gloo_comm(tmp_path):
    """World-2 rehearsal of the multi-GPU serving path on ONE physical
    GPU: both ranks run the full per-rank pipeline (engine + cuda
    compute), collectives bridge through gloo/host (comm_device).  This
    exercises device pinning, per-rank engine instantiation and the
    gather size math the 8-GPU run depends on."""
    from sonata_amd.models import create_random_voice

    pack = create_random_voice(str(tmp_path), "gpu_dp", quality="x_low")
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_run_gpu_w2, args=(r, 2, 29741, pack, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(2)]
    for p in procs:
        p.join(timeout=300)
    errs = [r for r in results if r[0] != "ok"]
    assert not errs, errs


@pytest.mark.gpu
def test_rccl_world1_roundtrip():
    """RCCL (nccl backend) initializes and runs the bench's collectives
    (barrier + all_gather) at world_size 1 on a real GPU — the minimal
    RCCL exercise possible on a single-GPU box; world>1 runs via the
    driver's scaling bench."""
    import os

    import torch
    import torch.distributed as dist

    os.environ.update({"RANK": "0", "WORLD_SIZE": "1", "LOCAL_RANK": "0",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29719"})
    assert torch.cuda.is_available()
    dist.init_process_group(backend="nccl")
    try:
        t = torch.tensor([3.5, 7.0], dtype=torch.float64, device="cuda:0")
        out = [torch.zeros_like(t)]
        dist.all_gather(out, t)
        assert torch.equal(out[0], t)
        dist.barrier()
    finally:
        dist.destroy_process_group()
