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
