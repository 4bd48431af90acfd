"""pysonata-compatible API tests (CPU) — class/method surface parity with
the reference Python frontend (crates/frontends/python/src/lib.rs)."""

import pytest

from sonata_amd.frontends import pysonata
from sonata_amd.models import create_random_voice


@pytest.fixture(scope="module")
def model(tmp_path_factory):
    d = tmp_path_factory.mktemp("pys")
    pack = create_random_voice(str(d), "pys_voice", quality="x_low",
                               num_speakers=2)
    return pysonata.PiperModel(pack, device="cpu")


def test_model_accessors(model):
    assert model.speakers == {0: "spk0", 1: "spk1"}
    model.set_speaker("spk1")
    assert model.get_speaker() == "spk1"
    with pytest.raises(pysonata.SonataException):
        model.set_speaker("nope")
    model.length_scale = 1.2
    assert abs(model.length_scale - 1.2) < 1e-6
    model.noise_scale = 0.5
    model.noise_w = 0.7
    assert abs(model.noise_scale - 0.5) < 1e-6


def test_synthesize_modes(model, tmp_path):
    tts = pysonata.Sonata.with_piper(model)
    assert tts.language == "en-us"
    assert tts.get_audio_output_info().sample_rate == 16000

    waves = list(tts.synthesize("hˈɛloʊ. wˈɜːld."))
    assert len(waves) == 2
    w = waves[0]
    assert len(w.get_wave_bytes()) > 500
    assert w.sample_rate == 16000
    assert w.duration_ms > 0 and w.inference_ms > 0
    assert w.real_time_factor > 0

    lazy = list(tts.synthesize_lazy("wˈʌn."))
    assert len(lazy) == 1

    chunks = list(tts.synthesize_streamed("hˈɛloʊ ˈɛvɹiwˌʌn tˈʊdeɪ.",
                                          chunk_size=20, chunk_padding=2))
    assert len(chunks) >= 1 and all(isinstance(c, bytes) for c in chunks)

    out = tmp_path / "p.wav"
    tts.synthesize_to_file(str(out), "sˈɛntəns.",
                           pysonata.AudioOutputConfig(rate=60))
    assert out.read_bytes()[:4] == b"RIFF"

    w.save_to_file(str(tmp_path / "w.wav"))
    assert (tmp_path / "w.wav").exists()


def test_phonemize_text_standalone():
    sents = pysonata.phonemize_text("Hello there. How are you?",
                                    language="en-us")
    assert len(sents) == 2
    stressed = pysonata.phonemize_text("hello", remove_stress=False)
    unstressed = pysonata.phonemize_text("hello", remove_stress=True)
    assert "ˈ" not in unstressed[0]
    assert stressed != unstressed or "ˈ" not in stressed[0]
