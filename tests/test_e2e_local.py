"""End-to-end on CPU: a real TorchJob (master=1, worker=1) through the
control plane with LocalProcessRuntime — two OS processes training a
tiny Llama over gloo, checkpoint agent protocol, model packaging.

This is the plumbing config of BASELINE.json (config 1: CPU distributed
job, no GPU) run without a cluster.
"""
import json
import os
import time

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

from torch_on_k8s_amd.controlplane.api import (JobConditionType, TaskSpec,
                                               TaskType, TorchJob)
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.modelregistry import (ModelRegistry,
                                                         StorageProvider)
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import LocalProcessRuntime

TRAINER_CFG = {
    "model": "llama-tiny",
    "micro_batch": 2,
    "seq_len": 32,
    "lr": 1e-3,
}


def make_env(steps=3):
    return {
        "TOK_BACKEND": "gloo",
        "TOK_TRAIN_STEPS": str(steps),
        "TOK_TRAINER_CONFIG": json.dumps(TRAINER_CFG),
        "MASTER_ADDR": "127.0.0.1",
        # child procs must see the repo
        "PYTHONPATH": os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))),
    }


def drive(ctl, job, timeout=180):
    t0 = time.time()
    while time.time() - t0 < timeout:
        ctl.reconcile(job)
        if job.status.phase in (JobConditionType.SUCCEEDED,
                                JobConditionType.FAILED):
            return
        time.sleep(0.3)
    raise TimeoutError(f"job stuck in {job.status.phase}; "
                       f"events={[e.reason for e in ctl.events]}")


@pytest.mark.timeout(300)
def test_master_worker_job_runs_to_success(tmp_path):
    node = NodeState(num_gpus=0)
    rt = LocalProcessRuntime(str(tmp_path / "work"))
    store = StorageProvider(str(tmp_path / "store"))
    reg = ModelRegistry(store)
    # CPU job: no gang (no GPUs to reserve)
    ctl = JobController(node, rt,
                        ControllerConfig(enable_gang_scheduling=False),
                        model_registry=reg)
    job = TorchJob(
        name="e2e-job",
        tasks={
            TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=make_env()),
            TaskType.WORKER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=make_env()),
        },
        model_name="tiny-llama",
    )
    ctl.create_job(job)
    drive(ctl, job)
    if job.status.phase != JobConditionType.SUCCEEDED:
        logdir = tmp_path / "work" / "e2e-job"
        logs = "\n".join(
            f"== {p.name}\n{p.read_text()[-2000:]}"
            for p in logdir.glob("*.log"))
        raise AssertionError(f"job failed: {logs}")

    # training happened on both ranks and stayed in sync
    metrics = json.load(open(tmp_path / "work" / "e2e-job" / "metrics.json"))
    assert metrics["step"] == 3
    assert metrics["world_size"] == 2

    # model artifact packaged from the final checkpoint
    assert job.status.model_version is not None
    mv = reg.get_version("tiny-llama", job.status.model_version)
    assert mv is not None and mv.build_phase == "Succeeded"
    assert os.path.exists(mv.image_ref)


@pytest.mark.timeout(300)
def test_worker_process_failure_recovers(tmp_path):
    """Kill the worker mid-run with SIGTERM (retryable 143): the
    controller restarts it and the job still completes."""
    node = NodeState(num_gpus=0)
    rt = LocalProcessRuntime(str(tmp_path / "work"))
    ctl = JobController(node, rt,
                        ControllerConfig(enable_gang_scheduling=False))
    job = TorchJob(
        name="e2e-failover",
        tasks={
            # single master, no worker: master restart must resume from
            # the checkpoint agent's checkpoint
            TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=make_env(steps=30)),
        },
    )
    ctl.create_job(job)
    # wait until it has made some steps (metrics file appears)
    mpath = tmp_path / "work" / "e2e-failover" / "metrics.json"
    t0 = time.time()
    while not mpath.exists() and time.time() - t0 < 120:
        ctl.reconcile(job)
        time.sleep(0.2)
    assert mpath.exists(), "trainer never produced metrics"
    # SIGTERM the master (simulated preemption)
    h = ctl.handles["e2e-failover"][("e2e-failover", TaskType.MASTER, 0)]
    rt.kill(h, grace=True)
    rt.wait(h, timeout=60)
    ctl.reconcile(job)  # observes exit 143 -> retryable -> restart
    assert job.status.restart_count == 1
    drive(ctl, job)
    assert job.status.phase == JobConditionType.SUCCEEDED
    final = json.load(open(mpath))
    assert final["step"] == 30


@pytest.mark.timeout(120)
def test_arbitrary_command_task(tmp_path):
    """Parity with the reference's opaque-container model: a TaskSpec may
    run ANY command, not just the framework entrypoint."""
    node = NodeState(num_gpus=0)
    rt = LocalProcessRuntime(str(tmp_path / "work"))
    ctl = JobController(node, rt,
                        ControllerConfig(enable_gang_scheduling=False))
    import sys as _sys
    job = TorchJob(
        name="custom-cmd",
        tasks={
            TaskType.MASTER: TaskSpec(
                replicas=1, gpus_per_task=0,
                command=[_sys.executable, "-c",
                         "import os; print('RANK', os.environ['RANK']); "
                         "open(os.environ['TOK_STATE_DIR'] + '/ok', 'w')"
                         ".write('done')"]),
        },
    )
    ctl.create_job(job)
    drive(ctl, job, timeout=60)
    assert job.status.phase == JobConditionType.SUCCEEDED
    assert (tmp_path / "work" / "custom-cmd" / "ok").read_text() == "done"


@pytest.mark.timeout(300)
def test_aimaster_sidecar_and_periodic_ckpt(tmp_path):
    """AIMaster supervision sidecar journals training metrics while the
    master trains with TOK_CKPT_EVERY periodic async checkpoints; the
    checkpoint exists mid-run and the job completes."""
    import json as _json
    from torch_on_k8s_amd.controlplane.api import (JobConditionType,
                                                   TaskSpec, TaskType,
                                                   TorchJob)
    from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                          JobController)
    from torch_on_k8s_amd.controlplane.node import NodeState
    from torch_on_k8s_amd.controlplane.runtime import LocalProcessRuntime

    rt = LocalProcessRuntime(str(tmp_path / "work"))
    ctl = JobController(NodeState(num_gpus=0), rt,
                        ControllerConfig(enable_gang_scheduling=False))
    env = {
        "TOK_BACKEND": "gloo",
        "TOK_TRAIN_STEPS": "14",
        "TOK_CKPT_EVERY": "5",
        "TOK_STEP_DELAY": "0.15",
        "TOK_TRAINER_CONFIG": _json.dumps(
            {"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}),
        "PYTHONPATH": ROOT,
    }
    job = TorchJob(name="aim-e2e", tasks={
        TaskType.AIMASTER: TaskSpec(replicas=1, gpus_per_task=0, env=env),
        TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0, env=env),
    })
    ctl.create_job(job)
    state = tmp_path / "work" / "aim-e2e"
    saw_midrun_ckpt = False
    t0 = time.time()
    try:
        while time.time() - t0 < 240:
            ctl.reconcile(job)
            if (state / "ckpt" / "meta.json").exists() and \
                    job.status.phase == JobConditionType.RUNNING:
                saw_midrun_ckpt = True
            if job.status.phase in (JobConditionType.SUCCEEDED,
                                    JobConditionType.FAILED):
                break
            time.sleep(0.2)
        assert job.status.phase == JobConditionType.SUCCEEDED, \
            (job.status.phase, [(e.reason, e.message) for e in ctl.events])
        assert saw_midrun_ckpt, "periodic checkpoint never published mid-run"
        meta = _json.load(open(state / "ckpt" / "meta.json"))
        assert meta["step"] % 5 == 0 and meta["step"] >= 5
        # aimaster journaled structured metrics
        journal = (state / "aimaster.log").read_text().strip().splitlines()
        assert len(journal) >= 2
        assert all("loss" in ln for ln in journal)
    finally:
        ctl.delete_job("aim-e2e")
