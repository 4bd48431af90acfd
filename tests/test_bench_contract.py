"""Driver-contract rehearsal: bench.py runs single-rank and under
torch.distributed.run (the exact launch the driver uses for N>1),
printing one valid JSON line from rank 0."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _check_json(line: str, world: int):
    d = json.loads(line)
    assert d["metric"] == "audio_seconds_per_second"
    assert d["value"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["config"]["parallelism"] == f"dp{world}"
    assert d["data"] == "synthetic"
    return d


@pytest.mark.timeout(600)
def test_bench_single_rank_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "2", "--seq-len", "64", "--quality", "x_low"],
        cwd=ROOT, capture_output=True, text=True, timeout=550)
    assert r.returncode == 0, r.stderr[-2000:]
    _check_json(r.stdout.strip().splitlines()[-1], 1)


@pytest.mark.timeout(600)
def test_bench_torchrun_world2_cpu():
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "2",
         "--seq-len", "64", "--quality", "x_low"],
        cwd=ROOT, capture_output=True, text=True, timeout=550)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1  # only rank 0 prints
    _check_json(lines[-1], 2)
