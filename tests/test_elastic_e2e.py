"""Elastic scale-out end-to-end with REAL processes (CPU, gloo):
master + 1 worker training; mid-run scale to 2 workers runs the 2-stage
checkpoint transaction through the state-file bridge (controller
requests -> rank-0 agent checkpoints + acks -> controller restarts the
gang at the new WORLD_SIZE) and the job completes with all 3 ranks.

This is BASELINE config 4's shape ("elastic min->max mid-run scale-up")
at CPU scale.
"""
import json
import os
import time

import pytest

from torch_on_k8s_amd.controlplane.api import (ANN_CKPT_REQUESTED,
                                               ElasticPolicy,
                                               JobConditionType, TaskSpec,
                                               TaskType, TorchJob)
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.elastic import ElasticScaler
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import LocalProcessRuntime

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def task_env(steps):
    return {
        "TOK_BACKEND": "gloo",
        "TOK_TRAIN_STEPS": str(steps),
        "TOK_TRAINER_CONFIG": json.dumps(
            {"model": "llama-tiny", "micro_batch": 1, "seq_len": 32,
             "lr": 1e-3}),
        "PYTHONPATH": ROOT,
    }


@pytest.mark.timeout(420)
def test_elastic_scale_out_e2e(tmp_path):
    node = NodeState(num_gpus=0)
    rt = LocalProcessRuntime(str(tmp_path / "work"))
    ctl = JobController(node, rt,
                        ControllerConfig(enable_gang_scheduling=False),
                        elastic=ElasticScaler())
    steps = 60
    job = TorchJob(
        name="elastic-e2e",
        tasks={
            TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
            TaskType.WORKER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
        },
        elastic=ElasticPolicy(min_replicas=1, max_replicas=2),
    )
    ctl.create_job(job)
    try:
        _run_scenario(ctl, rt, job, tmp_path, steps)
    finally:
        # never leave orphaned gangs holding rendezvous ports for later
        # tests (a failed run's processes previously caused cross-test
        # port collisions)
        ctl.delete_job(job.name)


def _run_scenario(ctl, rt, job, tmp_path, steps):
    # wait until training is underway
    mpath = tmp_path / "work" / "elastic-e2e" / "metrics.json"
    t0 = time.time()
    while time.time() - t0 < 120:
        ctl.reconcile(job)
        if mpath.exists() and json.load(open(mpath))["step"] >= 3:
            break
        time.sleep(0.3)
    assert mpath.exists(), "training never started"
    assert json.load(open(mpath))["world_size"] == 2

    # scale out 1 -> 2 workers (generation bump -> checkpoint transaction)
    ElasticScaler.scale(job, 2)
    saw_ckpt_request = False
    t0 = time.time()
    while time.time() - t0 < 240:
        ctl.reconcile(job)
        saw_ckpt_request = saw_ckpt_request or \
            ANN_CKPT_REQUESTED in job.annotations
        if job.status.phase in (JobConditionType.SUCCEEDED,
                                JobConditionType.FAILED):
            break
        time.sleep(0.3)

    if job.status.phase != JobConditionType.SUCCEEDED:
        logdir = tmp_path / "work" / "elastic-e2e"
        logs = "\n".join(f"== {p.name}\n{p.read_text()[-1500:]}"
                         for p in logdir.glob("*.log"))
        raise AssertionError(
            f"phase={job.status.phase} events="
            f"{[(e.reason, e.message) for e in ctl.events]}\n{logs}")

    assert saw_ckpt_request, "checkpoint transaction never ran"
    final = json.load(open(mpath))
    assert final["step"] == steps
    assert final["world_size"] == 3  # master + 2 workers after scale
    # the restarted gang resumed from the checkpoint, not step 0
    agent = json.load(open(tmp_path / "work" / "elastic-e2e" / "agent.json"))
    assert agent["ckpt-completed-version"]["version"] == job.generation


@pytest.mark.timeout(420)
def test_fast_rejoin_keeps_processes_resident(tmp_path):
    """Scale event with fast-rejoin: the surviving master/worker keep
    their PIDs (model/optimizer state resident; only the process group
    re-inits) and training resumes at the new world without a
    checkpoint reload. Measures the scale-event downtime (r1 VERDICT
    next-#9)."""
    node = NodeState(num_gpus=0)
    rt = LocalProcessRuntime(str(tmp_path / "work"))
    ctl = JobController(node, rt,
                        ControllerConfig(enable_gang_scheduling=False),
                        elastic=ElasticScaler())
    steps = 200  # long enough to still be running after the scale
    job = TorchJob(
        name="rejoin-e2e",
        tasks={
            TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
            TaskType.WORKER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
        },
        elastic=ElasticPolicy(min_replicas=1, max_replicas=2),
    )
    ctl.create_job(job)
    mpath = tmp_path / "work" / "rejoin-e2e" / "metrics.json"
    try:
        t0 = time.time()
        while time.time() - t0 < 120:
            ctl.reconcile(job)
            if mpath.exists() and json.load(open(mpath))["step"] >= 3:
                break
            time.sleep(0.2)
        assert mpath.exists(), "training never started"
        hs = ctl.handles["rejoin-e2e"]
        master_pid = hs[("rejoin-e2e", TaskType.MASTER, 0)].proc.pid
        worker_pid = hs[("rejoin-e2e", TaskType.WORKER, 0)].proc.pid

        t_scale = time.time()
        ElasticScaler.scale(job, 2)
        # wait until training runs at world 3
        downtime = None
        t0 = time.time()
        while time.time() - t0 < 240:
            ctl.reconcile(job)
            if mpath.exists():
                rec = json.load(open(mpath))
                if rec.get("world_size") == 3:
                    downtime = time.time() - t_scale
                    break
            time.sleep(0.2)
        assert downtime is not None, \
            (job.status.phase, [(e.reason, e.message) for e in ctl.events])
        hs = ctl.handles["rejoin-e2e"]
        # survivors kept their processes
        assert hs[("rejoin-e2e", TaskType.MASTER, 0)].proc.pid == master_pid
        assert hs[("rejoin-e2e", TaskType.WORKER, 0)].proc.pid == worker_pid
        # the new worker is a different process
        assert ("rejoin-e2e", TaskType.WORKER, 1) in hs
        # survivors did NOT reload from checkpoint (no "resumed" in log)
        mlog = (tmp_path / "work" / "rejoin-e2e" /
                "rejoin-e2e-master-0.log").read_text()
        assert "fast-rejoin: world=3" in mlog
        assert "resumed at step" not in mlog
        print(f"scale-event downtime (trigger -> first step at new world): "
              f"{downtime:.1f}s")
    finally:
        ctl.delete_job(job.name)


@pytest.mark.timeout(420)
def test_fast_rejoin_scale_in(tmp_path):
    """Scale-IN with fast-rejoin: 3 ranks -> 2; the victim worker exits
    cleanly on its own, survivors keep their PIDs and continue at the
    smaller world, and the job runs to completion."""
    node = NodeState(num_gpus=0)
    rt = LocalProcessRuntime(str(tmp_path / "work"))
    ctl = JobController(node, rt,
                        ControllerConfig(enable_gang_scheduling=False),
                        elastic=ElasticScaler())
    steps = 40
    job = TorchJob(
        name="shrink-e2e",
        tasks={
            TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
            TaskType.WORKER: TaskSpec(replicas=2, gpus_per_task=0,
                                      env=task_env(steps)),
        },
        elastic=ElasticPolicy(min_replicas=1, max_replicas=2),
    )
    ctl.create_job(job)
    mpath = tmp_path / "work" / "shrink-e2e" / "metrics.json"
    try:
        t0 = time.time()
        while time.time() - t0 < 120:
            ctl.reconcile(job)
            if mpath.exists() and json.load(open(mpath))["step"] >= 3:
                break
            time.sleep(0.2)
        assert mpath.exists(), "training never started"
        assert json.load(open(mpath))["world_size"] == 3
        master_pid = ctl.handles["shrink-e2e"][
            ("shrink-e2e", TaskType.MASTER, 0)].proc.pid

        ElasticScaler.scale(job, 1)  # 2 workers -> 1 (world 3 -> 2)
        t0 = time.time()
        done = False
        while time.time() - t0 < 240:
            ctl.reconcile(job)
            if job.status.phase in (JobConditionType.SUCCEEDED,
                                    JobConditionType.FAILED):
                done = True
                break
            time.sleep(0.2)
        assert done and job.status.phase == JobConditionType.SUCCEEDED, \
            (job.status.phase, [(e.reason, e.message) for e in ctl.events])
        # master survived the shrink in place
        mlog = (tmp_path / "work" / "shrink-e2e" /
                "shrink-e2e-master-0.log").read_text()
        assert "fast-rejoin: world=2" in mlog
        hs_master = ctl.handles.get("shrink-e2e", {}).get(
            ("shrink-e2e", TaskType.MASTER, 0))
        if hs_master is not None:  # handle kept until cleanup
            assert hs_master.proc.pid == master_pid
        # the victim (worker-1) exited cleanly by itself (exit 0 logged
        # as task success, not a failover)
        assert job.status.restart_count == 0
    finally:
        ctl.delete_job(job.name)
