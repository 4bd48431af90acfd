"""Synthesizer facade tests: lazy/parallel/realtime modes + prosody config.

Models the reference's synth integration tests (synth/src/tests.rs:1-28:
lazy/parallel/realtime end-to-end with real voices — here with random-init
voice packs).
"""

import numpy as np
import pytest

from sonata_amd.audio.wav import read_wav_file
from sonata_amd.synth import (
    AudioOutputConfig,
    PITCH_RANGE,
    RATE_RANGE,
    SonataSpeechSynthesizer,
)

TEXT = "Hello world. This is a test."


@pytest.fixture(scope="module")
def synth(xlow_voice):
    return SonataSpeechSynthesizer(xlow_voice)


def test_lazy_stream(synth):
    audios = list(synth.synthesize_lazy(TEXT))
    assert len(audios) == 2
    for a in audios:
        assert len(a.samples) > 0
        assert a.info.sample_rate == 16000


def test_parallel_stream_matches_lazy(synth):
    lazy = list(synth.synthesize_lazy(TEXT))
    par = list(synth.synthesize_parallel(TEXT))
    assert len(lazy) == len(par)
    for a, b in zip(lazy, par):
        # lengths bit-identical (per-utterance noise); audio matches except
        # the unmasked-decoder tail of padded batch members
        assert len(a.samples) == len(b.samples)
        assert np.corrcoef(a.samples, b.samples)[0, 1] > 0.99


def test_realtime_stream(synth):
    chunks = list(synth.synthesize_streamed(TEXT, chunk_size=45,
                                            chunk_padding=3))
    assert len(chunks) >= 2
    total = sum(len(c) for c in chunks)
    ref = sum(len(a.samples) for a in synth.synthesize_lazy(TEXT))
    assert abs(total - ref) <= 4 * 42  # crossfade trims per sentence


def test_realtime_propagates_errors(synth):
    class Boom(Exception):
        pass

    orig = synth.model.stream_synthesis

    def bad(*a, **k):
        raise Boom("producer failed")

    synth.model.stream_synthesis = bad
    try:
        with pytest.raises(Boom):
            list(synth.synthesize_streamed("one."))
    finally:
        synth.model.stream_synthesis = orig


def test_appended_silence(synth):
    cfg = AudioOutputConfig(appended_silence_ms=100)
    a = list(synth.synthesize_lazy("one.", cfg))[0]
    b = list(synth.synthesize_lazy("one."))[0]
    extra = len(a.samples) - len(b.samples)
    assert abs(extra - 1600) <= 2  # 100ms @ 16kHz


def test_rate_changes_duration(synth):
    # rate=10% -> speed = 0.5 + 5*0.1 = 1.0 (no change); rate 50% -> 3.0
    fast_cfg = AudioOutputConfig(rate=50)
    base = list(synth.synthesize_lazy("a longer sentence for this test.",))[0]
    fast = list(
        synth.synthesize_lazy("a longer sentence for this test.", fast_cfg)
    )[0]
    speed = RATE_RANGE[0] + (RATE_RANGE[1] - RATE_RANGE[0]) * 0.5
    assert len(fast.samples) < len(base.samples) / (speed / 2)


def test_volume(synth):
    loud = list(synth.synthesize_lazy("one.", AudioOutputConfig(volume=100)))[0]
    quiet = list(synth.synthesize_lazy("one.", AudioOutputConfig(volume=10)))[0]
    assert np.abs(quiet.samples).max() < np.abs(loud.samples).max()


def test_pitch_preserves_duration(synth):
    base = list(synth.synthesize_lazy("a sentence to shift."))[0]
    hi = list(
        synth.synthesize_lazy("a sentence to shift.", AudioOutputConfig(pitch=90))
    )[0]
    assert abs(len(hi.samples) - len(base.samples)) < 0.1 * len(base.samples)


def test_synthesize_to_file(synth, tmp_path):
    p = str(tmp_path / "out.wav")
    audio = synth.synthesize_to_file(p, TEXT)
    assert len(audio.samples) > 0
    y, rate, ch = read_wav_file(p)
    assert rate == 16000 and len(y) == len(audio.samples)


def test_synth_config_delegation(synth):
    cfg = synth.get_synthesis_config()
    cfg.length_scale = 1.5
    synth.set_synthesis_config(cfg)
    assert synth.get_synthesis_config().length_scale == 1.5
    cfg.length_scale = 1.0
    synth.set_synthesis_config(cfg)


def test_concurrent_streams_one_voice(tmp_path):
    """Regression for the soak-found race: concurrent streams + batch
    calls on ONE voice must serialize cleanly through the per-voice
    inference lock (incl. a consumer that abandons its generator)."""
    import threading

    from sonata_amd.models import create_random_voice
    from sonata_amd.models.voice import load_voice

    pack = create_random_voice(str(tmp_path), "cc", quality="x_low")
    v = load_voice(pack, device="cpu")
    errs = []

    def stream_worker():
        try:
            for _ in range(3):
                total = sum(len(c) for c in
                            v.stream_synthesis("wˈʌn tˈuː θɹˈiː.", 45, 3))
                assert total > 100
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    def abandon_worker():
        try:
            for _ in range(3):
                it = v.stream_synthesis("fˈoːɹ fˈaɪv.", 45, 3)
                next(it)
                it.close()  # abandon mid-stream: lock must release
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    def batch_worker():
        try:
            for _ in range(3):
                v.speak_batch(["sˈɪks sˈɛvən.", "ˈeɪt."])
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=f) for f in
               (stream_worker, stream_worker, abandon_worker, batch_worker)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=300)
    assert not errs, errs
    assert not any(t.is_alive() for t in threads), "deadlock"
