"""CLI frontend tests (CPU): file mode, stdin JSON-lines protocol, output
enumeration, WAV-to-stdout — mirroring the reference CLI behavior
(crates/frontends/cli/src/main.rs)."""

import io
import json
import os

import pytest

from sonata_amd.frontends import cli
from sonata_amd.models import create_random_voice


@pytest.fixture(scope="module")
def voice_pack(tmp_path_factory):
    d = tmp_path_factory.mktemp("voice")
    return create_random_voice(str(d), "cli_voice", quality="x_low")


def test_file_mode_writes_wav(voice_pack, tmp_path):
    inp = tmp_path / "in.txt"
    inp.write_text("həˈloʊ wˈɜːld.")
    out = tmp_path / "out.wav"
    rc = cli.main([voice_pack, "-f", str(inp), "-o", str(out),
                   "--device", "cpu"])
    assert rc == 0
    data = out.read_bytes()
    assert data[:4] == b"RIFF" and len(data) > 1000


def test_stdin_loop_enumerates_outputs(voice_pack, tmp_path):
    out = tmp_path / "req.wav"
    lines = io.StringIO(
        json.dumps({"text": "wˈʌn."}) + "\n"
        + "this is not json\n"  # must be skipped, not fatal
        + json.dumps({"text": "tˈuː.", "rate": 60}) + "\n"
    )
    rc = cli.main([voice_pack, "-o", str(out), "--device", "cpu"],
                  stdin=lines)
    assert rc == 0
    assert out.exists()
    assert (tmp_path / "req-1.wav").exists()


def test_stdout_wav_bytes(voice_pack):
    buf = io.BytesIO()
    lines = io.StringIO(json.dumps({"text": "hˈaɪ."}) + "\n")
    rc = cli.main([voice_pack, "--device", "cpu", "-m", "realtime",
                   "--chunk-size", "20", "--chunk-padding", "2"],
                  stdin=lines, stdout=buf)
    assert rc == 0
    assert buf.getvalue()[:4] == b"RIFF"


def test_scales_flags_apply(voice_pack, tmp_path):
    inp = tmp_path / "in.txt"
    inp.write_text("tˈɛst.")
    out = tmp_path / "s.wav"
    rc = cli.main([voice_pack, "-f", str(inp), "-o", str(out),
                   "--length-scale", "1.4", "--noise-scale", "0.1",
                   "--device", "cpu"])
    assert rc == 0
    assert out.exists()


def test_cli_speaker_flag(tmp_path):
    from sonata_amd.frontends import cli
    from sonata_amd.models import create_random_voice

    pack = create_random_voice(str(tmp_path), "spkcli", quality="x_low",
                               num_speakers=3)
    inp = tmp_path / "in.txt"
    inp.write_text("hˈɛloʊ.")
    out = tmp_path / "s.wav"
    rc = cli.main([pack, "-f", str(inp), "-o", str(out), "-s", "2",
                   "--device", "cpu"])
    assert rc == 0 and out.exists()
