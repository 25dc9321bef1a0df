"""The benchmark harness itself is under test (VERDICT r1 "What's weak"
#1): n_gpus must come from the measured world size, a mis-launched bench
must refuse to report, and the --via-manager path must produce the
BASELINE.json headline pair (tokens/s + job-to-Running latency).

All CPU (gloo, tiny model); the same code paths run on MI355X with RCCL.
"""
from __future__ import annotations

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")

TINY = ["--model", "llama-tiny", "--micro-batch", "1", "--seq-len", "32",
        "--attn", "sdpa", "--steps", "2", "--warmup", "1",
        "--device", "cpu"]


def _clean_env():
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT", "TORCHELASTIC_RUN_ID", "GROUP_RANK"):
        env.pop(k, None)
    return env


def _last_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout}")


def test_bench_self_spawns_n_ranks():
    """`python bench.py --gpus 2` with no torchrun wrapper must launch 2
    real ranks (self-exec under torch.distributed.run) and report the
    MEASURED world size."""
    r = subprocess.run(
        [sys.executable, BENCH, "--gpus", "2"] + TINY,
        capture_output=True, text=True, timeout=420, env=_clean_env(),
        cwd=REPO)
    assert r.returncode == 0, f"stdout={r.stdout}\nstderr={r.stderr}"
    out = _last_json_line(r.stdout)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["value"] > 0


def test_bench_refuses_world_size_mismatch():
    """WORLD_SIZE=1 but --gpus 4: must exit 2 without reporting (the r1
    8x-inflation trap)."""
    env = _clean_env()
    env.update({"WORLD_SIZE": "1", "RANK": "0"})
    r = subprocess.run(
        [sys.executable, BENCH, "--gpus", "4"] + TINY,
        capture_output=True, text=True, timeout=180, env=env, cwd=REPO)
    assert r.returncode == 2, f"rc={r.returncode} stdout={r.stdout}"
    assert "refusing" in r.stderr
    assert "{" not in r.stdout  # no JSON metric line emitted


@pytest.mark.parametrize("n", [1, 2])
def test_bench_via_manager(n):
    """The headline path: gang-scheduled TorchJob through the manager,
    tokens/s measured inside the job, job-to-Running reported."""
    r = subprocess.run(
        [sys.executable, BENCH, "--gpus", str(n), "--via-manager",
         "--timeout", "300"] + TINY,
        capture_output=True, text=True, timeout=420, env=_clean_env(),
        cwd=REPO)
    assert r.returncode == 0, f"stdout={r.stdout}\nstderr={r.stderr}"
    out = _last_json_line(r.stdout)
    assert out["n_gpus"] == n  # measured INSIDE the job
    assert out["config"]["via_manager"] is True
    assert out["config"]["p50_job_to_running_s"] is not None
    assert out["config"]["p50_job_to_running_s"] >= 0
    assert out["value"] > 0
