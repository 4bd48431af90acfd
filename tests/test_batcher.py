"""Dynamic batcher: correctness under concurrency + batch-invariance."""

import threading

import numpy as np
import pytest

from sonata_amd.models import create_random_voice
from sonata_amd.models.voice import load_voice
from sonata_amd.synth import DynamicBatcher


@pytest.fixture(scope="module")
def voice(tmp_path_factory):
    d = tmp_path_factory.mktemp("batcher")
    return load_voice(create_random_voice(str(d), "b", quality="x_low"),
                      device="cpu")


def test_concurrent_submissions_match_serial(voice):
    batcher = DynamicBatcher(voice, max_batch=8, max_wait_ms=20)
    phons = [f"wˈʌn tˈuː {'θɹˈiː ' * (1 + i % 4)}." for i in range(24)]
    serial = [voice.speak_one_sentence(p).samples for p in phons]

    results = [None] * len(phons)

    def worker(i):
        results[i] = batcher.synthesize(phons[i]).samples

    threads = [threading.Thread(target=worker, args=(i,))
               for i in range(len(phons))]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    batcher.close()
    for i, (got, ref) in enumerate(zip(results, serial)):
        assert got is not None, i
        assert len(got) == len(ref), i
        np.testing.assert_allclose(got, ref, atol=1e-5)


def test_batcher_propagates_errors(voice):
    class Broken:
        def speak_batch(self, p):
            raise RuntimeError("boom")

    b = DynamicBatcher(Broken(), max_wait_ms=1)
    with pytest.raises(RuntimeError, match="boom"):
        b.synthesize("x")
    b.close()


@pytest.mark.gpu
def test_batcher_gpu_concurrent():
    """Dynamic batcher on the GPU serving path (engine-backed voice):
    concurrent results equal serial ones."""
    import tempfile

    import torch

    assert torch.cuda.is_available()
    with tempfile.TemporaryDirectory() as d:
        v = load_voice(create_random_voice(d, "bg", quality="x_low"),
                       device="cuda:0")
        batcher = DynamicBatcher(v, max_batch=16, max_wait_ms=10)
        phons = [f"wˈʌn {'tˈuː ' * (1 + i % 5)}." for i in range(24)]
        serial = {p: v.speak_one_sentence(p).samples for p in set(phons)}
        futures = [batcher.submit(p) for p in phons]
        for p, f in zip(phons, futures):
            got = f.result(timeout=120).samples
            ref = serial[p]
            assert len(got) == len(ref)
            assert float(np.abs(got - ref).max()) < 1e-3
        batcher.close()


def test_length_bucketing_preserves_results(voice):
    """Mixed short/long submissions split into length buckets; every
    result still equals serial synthesis (per-utterance seeding)."""
    batcher = DynamicBatcher(voice, max_batch=32, max_wait_ms=30)
    phons = (["wˈʌn."] * 6
             + ["wˈʌn tˈuː θɹˈiː fˈoːɹ fˈaɪv sˈɪks sˈɛvən ˈeɪt "
                "nˈaɪn tˈɛn ˈeɪt nˈaɪn tˈɛn." ] * 6)
    futures = [batcher.submit(p) for p in phons]
    for p, f in zip(phons, futures):
        got = f.result(timeout=120).samples
        ref = voice.speak_one_sentence(p).samples
        assert len(got) == len(ref)
        np.testing.assert_allclose(got, ref, atol=1e-5)
    batcher.close()
