"""Frontend error-path tests: failures surface as typed errors / status
codes, never crashes (reference error mapping)."""

import grpc
import pytest


def test_grpc_load_bad_voice_invalid_argument():
    from sonata_amd.frontends.grpc import create_server
    from sonata_amd.frontends.grpc.client import SonataGrpcClient
    from sonata_amd.frontends.grpc.proto import MESSAGES

    server, port, _ = create_server(port=0, device="cpu")
    server.start()
    try:
        client = SonataGrpcClient(f"127.0.0.1:{port}")
        with pytest.raises(grpc.RpcError) as e:
            client.LoadVoice(MESSAGES["VoicePath"](
                config_path="/nonexistent/voice.json"))
        assert e.value.code() == grpc.StatusCode.INVALID_ARGUMENT
        client.close()
    finally:
        server.stop(grace=None)


def test_cli_missing_config_raises(tmp_path):
    from sonata_amd.core import ModelError
    from sonata_amd.frontends import cli

    with pytest.raises((ModelError, OSError)):
        cli.main(["/nonexistent/voice.json", "-f", "/dev/null",
                  "--device", "cpu"])


def test_load_voice_missing_weights(tmp_path):
    import json

    from sonata_amd.core import ModelError
    from sonata_amd.models.voice import load_voice

    cfg = tmp_path / "v.json"
    cfg.write_text(json.dumps({"audio": {"quality": "x_low"}}))
    with pytest.raises(ModelError, match="weights"):
        load_voice(str(cfg), device="cpu")


def test_phonemizer_unknown_language():
    from sonata_amd.core import PhonemizationError
    from sonata_amd.text.phonemizer import text_to_phonemes

    with pytest.raises(PhonemizationError):
        text_to_phonemes("hello", voice="xx-unknown")


def test_engine_cpp_explicit_requires_extension(tmp_path):
    """engine='cpp' must fail loudly if asked for explicitly and broken;
    'auto' falls back silently."""
    from sonata_amd.models import create_random_voice
    from sonata_amd.models.voice import load_voice

    pack = create_random_voice(str(tmp_path), "e", quality="x_low")
    v = load_voice(pack, device="cpu", engine="python")
    assert v._engine is None  # python path, no engine attached
