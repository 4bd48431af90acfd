"""Observability: per-stage timers (SONATA_TRACE), SURVEY.md §5."""

import os

import pytest


def test_stage_timers(tmp_path, monkeypatch):
    monkeypatch.setenv("SONATA_TRACE", "1")
    from sonata_amd.models import create_random_voice
    from sonata_amd.models.voice import load_voice
    from sonata_amd.utils import get_stage_times

    get_stage_times().reset()
    pack = create_random_voice(str(tmp_path), "tr", quality="x_low")
    v = load_voice(pack, device="cpu")
    v.speak_one_sentence("hˈɛloʊ.")
    snap = get_stage_times().snapshot()
    assert "infer" in snap and snap["infer"]["calls"] == 1
    assert snap["infer"]["ms"] > 0


def test_trace_off_is_noop(tmp_path, monkeypatch):
    monkeypatch.delenv("SONATA_TRACE", raising=False)
    from sonata_amd.utils import get_stage_times
    from sonata_amd.utils.trace import stage_timer

    get_stage_times().reset()
    with stage_timer("x"):
        pass
    assert get_stage_times().snapshot() == {}
