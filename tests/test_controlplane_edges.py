"""Control-plane edge behaviors: TTL cleanup, clean-task policies,
coordinator background loop, runtime env contract details."""
import time

import pytest

from torch_on_k8s_amd.controlplane.api import (CleanPodPolicy,
                                               JobConditionType, RunPolicy,
                                               TaskPhase, TaskSpec, TaskType,
                                               TorchJob, set_defaults,
                                               ANN_WORLD_SIZE)
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.coordinator import Coordinator
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import (FakeRuntime, cluster_env,
                                                   task_name)


def mk_ctl(num_gpus=8, **cfg):
    node = NodeState(num_gpus=num_gpus)
    rt = FakeRuntime()
    return JobController(node, rt, ControllerConfig(**cfg)), node, rt


def finish_job(ctl, rt, job):
    ctl.reconcile(job)
    ctl.reconcile(job)
    for key in list(rt.tasks):
        if key[0] == job.name:
            rt.set_phase(key, TaskPhase.SUCCEEDED, 0)
    ctl.reconcile(job)


def test_ttl_cleanup_removes_finished_job():
    ctl, node, rt = mk_ctl()
    job = TorchJob(name="ttl-job",
                   tasks={TaskType.MASTER: TaskSpec(replicas=1)},
                   run_policy=RunPolicy(ttl_seconds_after_finished=0.01))
    ctl.create_job(job)
    finish_job(ctl, rt, job)
    assert job.status.phase == JobConditionType.SUCCEEDED
    time.sleep(0.05)
    ctl.reconcile(job)
    assert "ttl-job" not in ctl.jobs  # TTL-deleted (job.go:510-539 parity)


def test_clean_policy_none_keeps_running_workers():
    ctl, node, rt = mk_ctl()
    job = TorchJob(
        name="keep-job",
        tasks={TaskType.MASTER: TaskSpec(replicas=1),
               TaskType.WORKER: TaskSpec(replicas=2)},
        run_policy=RunPolicy(clean_task_policy=CleanPodPolicy.NONE))
    ctl.create_job(job)
    ctl.reconcile(job)
    ctl.reconcile(job)
    # master + workers succeed -> job done; with NONE nothing is killed
    for key in list(rt.tasks):
        rt.set_phase(key, TaskPhase.SUCCEEDED, 0)
    ctl.reconcile(job)
    assert job.status.phase == JobConditionType.SUCCEEDED
    assert not rt.killed


def test_clean_policy_running_kills_stragglers():
    ctl, node, rt = mk_ctl()
    job = TorchJob(
        name="clean-job",
        tasks={TaskType.MASTER: TaskSpec(replicas=1),
               TaskType.WORKER: TaskSpec(replicas=2)})
    ctl.create_job(job)
    ctl.reconcile(job)
    ctl.reconcile(job)
    # master + worker-0 succeed; worker-1 still running -> job is NOT done
    rt.set_phase((job.name, TaskType.MASTER, 0), TaskPhase.SUCCEEDED, 0)
    rt.set_phase((job.name, TaskType.WORKER, 0), TaskPhase.SUCCEEDED, 0)
    ctl.reconcile(job)
    assert job.status.phase != JobConditionType.SUCCEEDED
    rt.set_phase((job.name, TaskType.WORKER, 1), TaskPhase.SUCCEEDED, 0)
    ctl.reconcile(job)
    assert job.status.phase == JobConditionType.SUCCEEDED


def test_coordinator_background_loop():
    admitted = []
    coord = Coordinator(dequeue_fn=lambda j: admitted.append(j.name),
                        default_quota=100)
    job = set_defaults(TorchJob(name="loop-job",
                                tasks={TaskType.MASTER: TaskSpec()}))
    coord.enqueue_or_update(job)
    coord.run()
    t0 = time.time()
    while not admitted and time.time() - t0 < 5:
        time.sleep(0.05)
    coord.stop()
    assert admitted == ["loop-job"]


def test_cluster_env_contract():
    job = set_defaults(TorchJob(
        name="envjob",
        tasks={TaskType.AIMASTER: TaskSpec(replicas=1),
               TaskType.MASTER: TaskSpec(replicas=1),
               TaskType.WORKER: TaskSpec(replicas=3)}))
    env_m = cluster_env(job, TaskType.MASTER, 0)
    env_w = cluster_env(job, TaskType.WORKER, 2)
    env_a = cluster_env(job, TaskType.AIMASTER, 0)
    # WORLD_SIZE excludes the AIMaster (torchjob_controller.go:350)
    assert env_m["WORLD_SIZE"] == "4"
    assert env_m["RANK"] == "0"
    assert env_w["RANK"] == "3"     # worker idx+1 (master holds rank 0)
    assert env_a["RANK"] == "-1"    # not in the process group
    # elastic WORLD_SIZE override via annotation
    job.annotations[ANN_WORLD_SIZE] = "7"
    assert cluster_env(job, TaskType.WORKER, 0)["WORLD_SIZE"] == "7"
    assert task_name("j", TaskType.WORKER, 2) == "j-worker-2"


def test_master_port_stable_across_restart():
    """The rendezvous port must survive a failover restart (the
    reference patches the service targetPort, service.go:288-303)."""
    ctl, node, rt = mk_ctl()
    job = TorchJob(name="pjob", tasks={TaskType.MASTER: TaskSpec(replicas=1)})
    ctl.create_job(job)
    port1 = rt.tasks[("pjob", TaskType.MASTER, 0)].env["MASTER_PORT"]
    rt.set_phase(("pjob", TaskType.MASTER, 0), TaskPhase.FAILED, 137)
    ctl.reconcile(job)  # restart
    port2 = rt.tasks[("pjob", TaskType.MASTER, 0)].env["MASTER_PORT"]
    assert port1 == port2


def test_restart_reuses_gpu_slots():
    """In-place-restart parity: a retryable failure restarts the task on
    the SAME GPU slots (locality; no gang reshuffle)."""
    ctl, node, rt = mk_ctl()
    job = TorchJob(name="slotjob",
                   tasks={TaskType.WORKER: TaskSpec(replicas=2,
                                                    gpus_per_task=2)})
    ctl.create_job(job)
    ctl.reconcile(job)
    h = ctl.handles["slotjob"][("slotjob", TaskType.WORKER, 1)]
    slots_before = h.gpu_slots
    assert len(slots_before) == 2
    rt.set_phase(h.key, TaskPhase.FAILED, exit_code=137)
    ctl.reconcile(job)
    h2 = ctl.handles["slotjob"][("slotjob", TaskType.WORKER, 1)]
    assert h2 is not h
    assert h2.gpu_slots == slots_before
    assert h2.restart_count == 1


def test_manager_spec_update_scales_replicas(tmp_path):
    """Editing a spooled job YAML (replica change) bumps the generation
    (OnOwnerUpdate analog -> elastic scale path)."""
    import os
    import time as _time
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane import features as feat
    mgr = Manager(str(tmp_path), num_gpus=8,
                  gates=feat.FeatureGates({"JobCoordinator": False}))
    spec = """
kind: TorchJob
metadata: {name: upd-job}
spec:
  tasks:
    worker: {replicas: 2, gpusPerTask: 1, command: ["/bin/sleep", "300"]}
"""
    path = os.path.join(mgr.spool, "upd.yaml")
    with open(path, "w") as f:
        f.write(spec)
    mgr.step()
    job = mgr.controller.jobs["upd-job"]
    gen0 = job.generation
    assert job.tasks[TaskType.WORKER].replicas == 2

    _time.sleep(0.02)
    with open(path, "w") as f:
        f.write(spec.replace("replicas: 2", "replicas: 4"))
    os.utime(path)  # ensure a fresh mtime
    mgr.step()
    assert job.tasks[TaskType.WORKER].replicas == 4
    assert job.generation == gen0 + 1
    # cleanup child sleep processes
    mgr.controller.delete_job("upd-job")
