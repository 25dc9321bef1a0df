"""Metric-driven autoscaling end-to-end THROUGH THE MANAGER: a running
elastic job's real trainer metrics (metrics.json, the structured
replacement for the reference's worker-0 log regex,
torchelastic/observation.go:40-106) feed manager.autoscale_pass, which
doubles the worker count (computeNewReplicas x2, torchelastic/job.go:102-104)
and drives the 2-stage checkpoint transaction + fast-rejoin scale-out.
"""
import json
import os
import time

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(420)
def test_manager_autoscales_on_metrics(tmp_path):
    import yaml
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane.api import JobConditionType

    mgr = Manager(str(tmp_path), num_gpus=0, sync_period=0.05)
    steps = 80
    env = {
        "TOK_BACKEND": "gloo",
        "TOK_TRAIN_STEPS": str(steps),
        "TOK_STEP_DELAY": "0.2",  # slow enough that metricWindow samples
        # land before the run ends even on a contended box (xdist -n 4)
        "TOK_TRAINER_CONFIG": json.dumps(
            {"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}),
        "PYTHONPATH": ROOT,
    }
    doc = {
        "kind": "TorchJob",
        "metadata": {"name": "auto-e2e"},
        "spec": {
            "elasticPolicy": {"minReplicas": 1, "maxReplicas": 2,
                              "metricWindow": 3},
            "tasks": {
                "master": {"replicas": 1, "gpusPerTask": 0, "env": env},
                "worker": {"replicas": 1, "gpusPerTask": 0, "env": env},
            },
        },
    }
    with open(os.path.join(mgr.spool, "auto-e2e.yaml"), "w") as f:
        yaml.safe_dump(doc, f)

    job = None
    scaled = False
    deadline = time.time() + 360
    try:
        while time.time() < deadline:
            mgr.step()
            job = mgr.controller.jobs.get("auto-e2e")
            if job is not None:
                if not scaled and \
                        job.tasks[list(job.tasks)[1]].replicas == 2:
                    scaled = True  # autoscaler doubled the workers
                if job.status.phase in (JobConditionType.SUCCEEDED,
                                        JobConditionType.FAILED):
                    break
            time.sleep(0.05)
        assert job is not None
        from torch_on_k8s_amd.controlplane.api import TaskType
        assert job.tasks[TaskType.WORKER].replicas == 2, \
            "autoscaler never doubled the workers"
        est = job.status.elastic
        assert est is not None and est.last_replicas == 1
        # the decision surfaced as a job event (describe/status visible)
        assert any(e.reason.startswith("Elastic")
                   for e in mgr.controller.events_for("auto-e2e"))
        assert job.status.phase == JobConditionType.SUCCEEDED, (
            job.status.phase, mgr.controller.events_for("auto-e2e"))
        # the gang really ran at world 3 after the scale (master + 2)
        mpath = tmp_path / "jobs" / "auto-e2e" / "metrics.json"
        assert json.load(open(mpath))["world_size"] == 3
    finally:
        if job is not None:
            mgr.controller.delete_job(job.name)
