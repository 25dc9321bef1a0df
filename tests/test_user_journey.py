"""Full user journey through the REAL entry points — the reference's
`kubectl apply` + operator workflow end to end on one node:

  manager daemon (python -m torch_on_k8s_amd.manager, leader lock) ←
  client CLI (python -m torch_on_k8s_amd.client) apply → wait →
  describe → logs → models → extract → delete.
"""
from __future__ import annotations

import json
import os
import signal
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _cli(workdir, *args):
    return subprocess.run(
        [sys.executable, "-m", "torch_on_k8s_amd.client",
         "--workdir", str(workdir)] + list(args),
        capture_output=True, text=True, timeout=120,
        env=dict(os.environ, PYTHONPATH=ROOT))


def test_user_journey(tmp_path):
    mgr = subprocess.Popen(
        [sys.executable, "-m", "torch_on_k8s_amd.manager",
         "--workdir", str(tmp_path), "--num-gpus", "0",
         "--metrics-addr", "0", "--sync-period", "0.05",
         "--priority-class", "gold=50"],
        env=dict(os.environ, PYTHONPATH=ROOT),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        spec = tmp_path / "job.yaml"
        spec.write_text(
            "kind: TorchJob\n"
            "metadata: {name: journey}\n"
            "spec:\n"
            "  schedulingPolicy: {priorityClassName: gold}\n"
            "  modelName: journey-model\n"
            "  tasks:\n"
            "    master: {replicas: 1, gpusPerTask: 0, env: {\n"
            "      TOK_BACKEND: gloo, TOK_TRAIN_STEPS: '2',\n"
            "      TOK_TRAINER_CONFIG: '"
            '{"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}'
            "'}}\n")
        r = _cli(tmp_path, "apply", str(spec))
        assert r.returncode == 0 and "torchjob/journey applied" in r.stdout

        r = _cli(tmp_path, "wait", "journey")
        assert r.returncode == 0, r.stdout + r.stderr
        st = json.loads(r.stdout)
        assert st["phase"] == "Succeeded", st

        r = _cli(tmp_path, "describe", "journey")
        assert "Phase:       Succeeded" in r.stdout
        assert "ModelVersion: mv-journey" in r.stdout  # packaged version

        r = _cli(tmp_path, "logs", "journey", "--tail", "5")
        assert "step=" in r.stdout  # trainer progress lines

        r = _cli(tmp_path, "models")
        models = json.loads(r.stdout)
        assert "journey-model" in models
        latest = models["journey-model"]["latest"]
        assert latest

        dest = tmp_path / "extracted"
        r = _cli(tmp_path, "extract", f"journey-model:{latest}", str(dest))
        assert r.returncode == 0
        root = r.stdout.strip()
        assert os.path.exists(os.path.join(root, "final", "model.pt"))

        r = _cli(tmp_path, "delete", "journey")
        assert "deleted" in r.stdout
        deadline = time.time() + 30
        while time.time() < deadline:
            if _cli(tmp_path, "get", "journey").stdout.startswith("not found"):
                break
            time.sleep(0.2)
        assert _cli(tmp_path, "get", "journey").stdout.startswith("not found")
    finally:
        if mgr.poll() is None:
            os.kill(mgr.pid, signal.SIGTERM)
            try:
                mgr.wait(timeout=30)
            except subprocess.TimeoutExpired:
                mgr.kill()
