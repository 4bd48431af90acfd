"""gRPC frontend tests (CPU, loopback): version, voice load (idempotent),
voice info, synthesis options round-trip, streamed synthesis — mirroring
the reference server behavior (crates/frontends/grpc/src/main.rs)."""

import grpc
import pytest

from sonata_amd.frontends.grpc import create_server
from sonata_amd.frontends.grpc.client import SonataGrpcClient
from sonata_amd.frontends.grpc.proto import MESSAGES
from sonata_amd.models import create_random_voice


@pytest.fixture(scope="module")
def voice_pack(tmp_path_factory):
    d = tmp_path_factory.mktemp("grpc_voice")
    return create_random_voice(str(d), "grpc_voice", quality="x_low",
                               num_speakers=2)


@pytest.fixture(scope="module")
def running(voice_pack):
    server, port, service = create_server(port=0, device="cpu")
    server.start()
    client = SonataGrpcClient(f"127.0.0.1:{port}")
    yield client, voice_pack
    client.close()
    server.stop(grace=None)


def test_version(running):
    client, _ = running
    v = client.GetSonataVersion(MESSAGES["Empty"]())
    assert v.version


def test_load_voice_idempotent(running):
    client, pack = running
    a = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack))
    b = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack))
    assert a.voice_id == b.voice_id
    assert a.audio.sample_rate == 16000  # x_low preset
    assert a.supports_streaming_output
    assert dict(a.speakers) == {0: "spk0", 1: "spk1"}
    info = client.GetVoiceInfo(
        MESSAGES["VoiceIdentifier"](voice_id=a.voice_id))
    assert info.voice_id == a.voice_id


def test_unknown_voice_not_found(running):
    client, _ = running
    with pytest.raises(grpc.RpcError) as e:
        client.GetVoiceInfo(MESSAGES["VoiceIdentifier"](voice_id="nope"))
    assert e.value.code() == grpc.StatusCode.NOT_FOUND


def test_synthesis_options_roundtrip(running):
    client, pack = running
    vid = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack)).voice_id
    opts = MESSAGES["SynthesisOptions"](speaker="spk1", length_scale=1.3)
    got = client.SetSynthesisOptions(MESSAGES["VoiceSynthesisOptions"](
        voice_id=vid, synthesis_options=opts))
    assert got.speaker == "spk1"
    assert abs(got.length_scale - 1.3) < 1e-6
    back = client.GetSynthesisOptions(
        MESSAGES["VoiceIdentifier"](voice_id=vid))
    assert back.speaker == "spk1"


def test_synthesize_utterance_stream(running):
    client, pack = running
    vid = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack)).voice_id
    results = list(client.SynthesizeUtterance(MESSAGES["Utterance"](
        voice_id=vid, text="wˈʌn. tˈuː.")))
    assert len(results) == 2  # one per sentence
    for r in results:
        assert len(r.wav_samples) > 500
        assert r.rtf > 0


def test_synthesize_realtime_stream(running):
    client, pack = running
    vid = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack)).voice_id
    chunks = list(client.SynthesizeUtteranceRealtime(MESSAGES["Utterance"](
        voice_id=vid, text="hˈɛloʊ ðˈɛr ˈɛvɹiwˌʌn.")))
    assert len(chunks) >= 1
    assert all(len(c.wav_samples) > 0 for c in chunks)


def test_concurrent_rpcs_batch_together(running):
    """Concurrent SynthesizeUtterance RPCs coalesce in the dynamic
    batcher and all complete correctly."""
    from concurrent.futures import ThreadPoolExecutor

    client, pack = running
    vid = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack)).voice_id

    def one(i):
        res = list(client.SynthesizeUtterance(MESSAGES["Utterance"](
            voice_id=vid, text=f"nˈʌmbɚ {'wˈʌn tˈuː '*(1+i%3)}.")))
        assert len(res) == 1 and len(res[0].wav_samples) > 400
        return len(res[0].wav_samples)

    with ThreadPoolExecutor(max_workers=16) as ex:
        sizes = list(ex.map(one, range(32)))
    assert len(sizes) == 32


def _mp_worker(port, device):
    from sonata_amd.frontends.grpc.server import _serve_one

    _serve_one(port, device)


def test_multiprocess_reuseport_serving(voice_pack):
    """SO_REUSEPORT: two server processes share one port; requests land on
    both and all succeed."""
    import multiprocessing as mp
    import time

    from concurrent.futures import ThreadPoolExecutor

    port = 49917
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_mp_worker, args=(port, "cpu"), daemon=True)
             for _ in range(2)]
    for p in procs:
        p.start()
    try:
        # wait for bind (children import torch: allow a while)
        deadline = time.time() + 120
        client = None
        while time.time() < deadline:
            try:
                c = SonataGrpcClient(f"127.0.0.1:{port}")
                c.LoadVoice(MESSAGES["VoicePath"](config_path=voice_pack))
                client = c
                break
            except grpc.RpcError:
                time.sleep(2)
        assert client is not None, "no worker bound the port"
        # many fresh connections spread across workers; load + synthesize
        def one(i):
            c = SonataGrpcClient(f"127.0.0.1:{port}")
            vid = c.LoadVoice(
                MESSAGES["VoicePath"](config_path=voice_pack)).voice_id
            res = list(c.SynthesizeUtterance(MESSAGES["Utterance"](
                voice_id=vid, text="wˈʌn tˈuː.")))
            c.close()
            return len(res[0].wav_samples)

        with ThreadPoolExecutor(max_workers=8) as ex:
            sizes = list(ex.map(one, range(16)))
        assert all(s > 300 for s in sizes)
        client.close()
    finally:
        for p in procs:
            p.terminate()


def test_options_partial_update(running):
    """SetSynthesisOptions only touches the fields present (proto3
    explicit presence via optional)."""
    client, pack = running
    vid = client.LoadVoice(MESSAGES["VoicePath"](config_path=pack)).voice_id
    base = client.SetSynthesisOptions(MESSAGES["VoiceSynthesisOptions"](
        voice_id=vid,
        synthesis_options=MESSAGES["SynthesisOptions"](
            length_scale=1.5, noise_scale=0.4, noise_w=0.6)))
    assert abs(base.noise_scale - 0.4) < 1e-6
    # update ONLY length_scale; others must persist
    got = client.SetSynthesisOptions(MESSAGES["VoiceSynthesisOptions"](
        voice_id=vid,
        synthesis_options=MESSAGES["SynthesisOptions"](length_scale=0.9)))
    assert abs(got.length_scale - 0.9) < 1e-6
    assert abs(got.noise_scale - 0.4) < 1e-6
    assert abs(got.noise_w - 0.6) < 1e-6
