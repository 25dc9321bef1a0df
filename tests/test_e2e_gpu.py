"""GPU end-to-end: a gang-scheduled TorchJob through the manager ON
REAL MI355X HARDWARE — the full path (spool -> coordinator -> gang ->
LocalProcessRuntime -> HIP-kernel training -> SIGTERM checkpoint ->
failover resume -> Succeeded -> OCI ModelVersion). The CPU e2e suite
covers the logic; this proves it on the metal each round (tiny model:
the realistic-shape kernel coverage lives in test_ops_gpu.py /
test_attention_gpu.py — here the subject is the PATH, and round-end
driver time is budgeted)."""
from __future__ import annotations

import json
import os
import signal
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(360)
def test_manager_job_with_preemption_on_gpu(tmp_path):
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane.api import (JobConditionType,
                                                   TaskType)

    mgr = Manager(str(tmp_path), num_gpus=1, sync_period=0.05)
    spec = {
        "kind": "TorchJob",
        "metadata": {"name": "gpu-e2e"},
        "spec": {
            "modelName": "served",
            "tasks": {"master": {
                "replicas": 1, "gpusPerTask": 1,
                "env": {
                    "TOK_TRAIN_STEPS": "12",
                    # tiny steps are ~ms on GPU: pace them so the
                    # preemption below deterministically lands mid-run
                    "TOK_STEP_DELAY": "0.3",
                    "TOK_TRAINER_CONFIG": json.dumps(
                        {"model": "llama-tiny", "micro_batch": 2,
                         "seq_len": 256}),
                    "PYTHONPATH": ROOT,
                },
            }},
        },
    }
    import yaml
    with open(os.path.join(mgr.spool, "gpu-e2e.yaml"), "w") as f:
        yaml.safe_dump(spec, f)

    # wait for some progress, then preempt with SIGTERM (spot semantics:
    # trainer checkpoints and exits 143 -> controller restarts -> resume)
    mpath = tmp_path / "jobs" / "gpu-e2e" / "metrics.json"
    deadline = time.time() + 180
    job = None
    preempted = False
    while time.time() < deadline:
        mgr.step()
        job = mgr.controller.jobs.get("gpu-e2e")
        if not preempted and mpath.exists():
            try:
                if json.load(open(mpath))["step"] >= 3:
                    h = mgr.controller.handles["gpu-e2e"][
                        ("gpu-e2e", TaskType.MASTER, 0)]
                    os.kill(h.proc.pid, signal.SIGTERM)
                    preempted = True
            except (ValueError, KeyError):
                pass
        if job is not None and job.status.phase in (
                JobConditionType.SUCCEEDED, JobConditionType.FAILED):
            break
        time.sleep(0.05)
    assert preempted, "never reached step 3"
    assert job is not None and \
        job.status.phase == JobConditionType.SUCCEEDED, \
        (job.status.phase if job else None,
         [(e.reason, e.message) for e in mgr.controller.events_for("gpu-e2e")])
    assert job.status.restart_count >= 1  # the preemption round-tripped
    # resumed from checkpoint, not from scratch
    log = (tmp_path / "jobs" / "gpu-e2e" /
           "gpu-e2e-master-0.log").read_text()
    assert "resumed at step" in log
    # model packaged as an OCI ModelVersion
    assert job.status.model_version is not None
    mv = mgr.registry.get_version("served", job.status.model_version)
    assert mv is not None and mv.build_phase == "Succeeded"
    assert os.path.exists(os.path.join(mv.image_ref, "index.json"))
