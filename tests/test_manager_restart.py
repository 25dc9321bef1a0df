"""Manager-restart durability (reference adopt/claim, pod.go:717-745 +
the API-server-backed state the operator rebuilds from): a new
controller over the same workdir must ADOPT the running gang — same
PIDs, no duplicate processes, same rendezvous port, preserved
generation — and drive the job to completion."""
from __future__ import annotations

import json
import os
import time

import pytest

from torch_on_k8s_amd.controlplane.api import (JobConditionType, TaskSpec,
                                               TaskType, TorchJob)
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import LocalProcessRuntime

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def task_env(steps):
    return {
        "TOK_BACKEND": "gloo",
        "TOK_TRAIN_STEPS": str(steps),
        "TOK_TRAINER_CONFIG": json.dumps(
            {"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}),
        "PYTHONPATH": ROOT,
    }


def mk_job(steps):
    return TorchJob(
        name="restart-e2e",
        tasks={
            TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
            TaskType.WORKER: TaskSpec(replicas=1, gpus_per_task=0,
                                      env=task_env(steps)),
        })


@pytest.mark.timeout(420)
def test_manager_restart_adopts_running_gang(tmp_path):
    work = str(tmp_path / "work")
    steps = 60
    rt_a = LocalProcessRuntime(work)
    ctl_a = JobController(NodeState(num_gpus=0), rt_a,
                          ControllerConfig(enable_gang_scheduling=False))
    job_a = ctl_a.create_job(mk_job(steps))
    mpath = tmp_path / "work" / "restart-e2e" / "metrics.json"
    t0 = time.time()
    while time.time() - t0 < 120:
        ctl_a.reconcile(job_a)
        if mpath.exists() and json.load(open(mpath))["step"] >= 2:
            break
        time.sleep(0.2)
    assert mpath.exists(), "training never started"
    pids_a = {k: h.proc.pid for k, h in ctl_a.handles["restart-e2e"].items()}
    port_a = ctl_a._master_port(job_a)
    try:
        # "manager crash": drop controller A without touching processes
        del ctl_a

        rt_b = LocalProcessRuntime(work)
        ctl_b = JobController(NodeState(num_gpus=0), rt_b,
                              ControllerConfig(enable_gang_scheduling=False))
        job_b = ctl_b.create_job(mk_job(steps))
        hs_b = ctl_b.handles["restart-e2e"]
        # adopted, not recreated: same pids, no extra processes
        assert {k: h.pid for k, h in hs_b.items()} == pids_a, \
            (pids_a, {k: (h.pid, h.proc) for k, h in hs_b.items()})
        assert all(h.proc is None for h in hs_b.values())
        assert any(e.reason == "TaskAdopted" for e in ctl_b.events)
        # rendezvous port restored from the persisted job view
        assert ctl_b._master_port(job_b) == port_a

        t0 = time.time()
        while time.time() - t0 < 240:
            ctl_b.reconcile(job_b)
            if job_b.status.phase in (JobConditionType.SUCCEEDED,
                                      JobConditionType.FAILED):
                break
            time.sleep(0.2)
        assert job_b.status.phase == JobConditionType.SUCCEEDED, \
            (job_b.status.phase,
             [(e.reason, e.message) for e in ctl_b.events])
        assert job_b.status.restart_count == 0
    finally:
        # belt-and-braces: no orphans left behind
        ctl = locals().get("ctl_b") or locals().get("ctl_a")
        if ctl is not None:
            ctl.delete_job("restart-e2e")


def test_adopted_exit_marker_grades_completion(tmp_path):
    """A framework task that exits while the manager is down leaves an
    exit marker; the next manager grades it instead of re-running a
    finished task as failed."""
    from torch_on_k8s_amd.controlplane.runtime import TaskHandle, task_name
    rt = LocalProcessRuntime(str(tmp_path))
    h = TaskHandle("j", TaskType.MASTER, 0)
    h.pid = 2**22 + os.getpid()  # definitely not alive
    from torch_on_k8s_amd.controlplane.api import TaskPhase
    h.phase = TaskPhase.RUNNING
    d = tmp_path / "j" / "tasks"
    d.mkdir(parents=True)
    (d / f"{task_name('j', TaskType.MASTER, 0)}.exit").write_text("0")
    rt.poll(h)
    assert h.phase == TaskPhase.SUCCEEDED and h.exit_code == 0
    # without a marker: conservative failure with AdoptedExit reason
    h2 = TaskHandle("j", TaskType.WORKER, 0)
    h2.pid = 2**22 + os.getpid()
    h2.phase = TaskPhase.RUNNING
    rt.poll(h2)
    assert h2.phase == TaskPhase.FAILED and h2.reason == "AdoptedExit"


def test_manager_graceful_shutdown_keeps_tasks(tmp_path):
    """SIGTERM to the manager daemon: it exits cleanly WITHOUT killing
    running gangs (the next manager adopts them)."""
    import signal
    import subprocess
    import sys
    mgr = subprocess.Popen(
        [sys.executable, "-m", "torch_on_k8s_amd.manager",
         "--workdir", str(tmp_path), "--num-gpus", "0",
         "--metrics-addr", "0", "--sync-period", "0.05"],
        env=dict(os.environ, PYTHONPATH=ROOT),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        spec = ("kind: TorchJob\nmetadata: {name: gj}\nspec:\n  tasks:\n"
                "    master: {replicas: 1, gpusPerTask: 0, env: {"
                "TOK_BACKEND: gloo, TOK_TRAIN_STEPS: '500', "
                "TOK_STEP_DELAY: '0.2', TOK_TRAINER_CONFIG: '"
                '{"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}'
                "'}}\n")
        os.makedirs(tmp_path / "spool", exist_ok=True)
        with open(tmp_path / "spool" / "gj.yaml", "w") as f:
            f.write(spec)
        # wait for the task pid record
        rec = tmp_path / "jobs" / "gj" / "tasks" / "gj-master-0.json"
        t0 = time.time()
        while time.time() - t0 < 60 and not rec.exists():
            time.sleep(0.2)
        assert rec.exists(), mgr.stdout
        task_pid = json.load(open(rec))["pid"]
        os.kill(mgr.pid, signal.SIGTERM)
        assert mgr.wait(timeout=30) == 0
        # task survived the manager
        os.kill(task_pid, 0)
    finally:
        if mgr.poll() is None:
            mgr.kill()
        try:
            os.kill(json.load(open(rec))["pid"], signal.SIGKILL)
        except (OSError, ValueError, FileNotFoundError):
            pass


def test_leader_election_lock(tmp_path):
    """Leader-election analog (reference main.go:77-83): a second
    manager on the same workdir cannot become leader until the first
    releases; a different workdir is independent."""
    import pytest as _pytest
    from torch_on_k8s_amd.manager import Manager

    m1 = Manager(str(tmp_path), num_gpus=1, sync_period=0.05)
    m1.acquire_leadership(block=False)
    m2 = Manager(str(tmp_path), num_gpus=1, sync_period=0.05)
    with _pytest.raises(RuntimeError, match="another manager"):
        m2.acquire_leadership(block=False)
    other = Manager(str(tmp_path / "other"), num_gpus=1, sync_period=0.05)
    other.acquire_leadership(block=False)  # independent workdir: fine
    other.release_leadership()
    # leader exits -> takeover succeeds
    m1.release_leadership()
    m2.acquire_leadership(block=False)
    m2.release_leadership()


def test_zero_downtime_upgrade_with_overlapping_managers(tmp_path):
    """The upgrade story end to end across PROCESSES: manager B starts
    while A is still leader, BLOCKS on the manager.lock flock, and on
    A's SIGTERM takes over and adopts A's still-running task (same
    pid)."""
    import signal
    import subprocess
    import sys

    def start_mgr():
        return subprocess.Popen(
            [sys.executable, "-m", "torch_on_k8s_amd.manager",
             "--workdir", str(tmp_path), "--num-gpus", "0",
             "--metrics-addr", "0", "--sync-period", "0.05"],
            env=dict(os.environ, PYTHONPATH=ROOT),
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)

    a = start_mgr()
    b = None
    rec = tmp_path / "jobs" / "zd" / "tasks" / "zd-master-0.json"
    try:
        spec = ("kind: TorchJob\nmetadata: {name: zd}\nspec:\n  tasks:\n"
                "    master: {replicas: 1, gpusPerTask: 0, env: {"
                "TOK_BACKEND: gloo, TOK_TRAIN_STEPS: '400', "
                "TOK_STEP_DELAY: '0.2', TOK_TRAINER_CONFIG: '"
                '{"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}'
                "'}}\n")
        os.makedirs(tmp_path / "spool", exist_ok=True)
        (tmp_path / "spool" / "zd.yaml").write_text(spec)
        t0 = time.time()
        while time.time() - t0 < 60 and not rec.exists():
            time.sleep(0.2)
        assert rec.exists(), a.stdout
        task_pid = json.load(open(rec))["pid"]

        b = start_mgr()                      # overlapping daemon
        time.sleep(3.0)
        assert b.poll() is None              # B is alive, blocked on lock
        # B must NOT have acted yet: A's pid is still in the lockfile
        assert (tmp_path / "manager.lock").read_text().strip() == str(a.pid)

        os.kill(a.pid, signal.SIGTERM)       # upgrade: stop A
        assert a.wait(timeout=30) == 0
        # B becomes leader and adopts the SAME process
        t0 = time.time()
        while time.time() - t0 < 60:
            txt = (tmp_path / "manager.lock").read_text().strip()
            if txt == str(b.pid):
                break
            time.sleep(0.2)
        assert (tmp_path / "manager.lock").read_text().strip() == str(b.pid)
        os.kill(task_pid, 0)                 # task survived both managers
        # and B is reconciling it: status file keeps updating
        st = tmp_path / "status" / "zd.json"
        t0 = time.time()
        while time.time() - t0 < 60 and not st.exists():
            time.sleep(0.2)
        assert st.exists(), b.stdout
    finally:
        for p in (a, b):
            if p is not None and p.poll() is None:
                p.kill()
        try:
            os.kill(json.load(open(rec))["pid"], signal.SIGKILL)
        except (OSError, ValueError, FileNotFoundError):
            pass
