"""Task runtimes: where the reference creates pods via the k8s API
(controllers/common/pod.go:503-637), this framework starts one local
process per task on the 8xMI355X node, with GPU slots assigned through
HIP_VISIBLE_DEVICES and the torch env contract injected
(SetClusterSpec parity, torchjob_controller.go:314-449).

FakeRuntime is the envtest analog: tests flip task phases
by hand, no process ever runs.
"""
from __future__ import annotations

import os
import signal
import subprocess
import sys
import time
from dataclasses import dataclass, field

from torch_on_k8s_amd.controlplane.api import (TaskPhase, TaskType, TorchJob,
                                               DEFAULT_MASTER_PORT)


@dataclass
class TaskHandle:
    job_name: str
    task_type: TaskType
    index: int
    phase: TaskPhase = TaskPhase.PENDING
    exit_code: int | None = None
    reason: str = ""
    gpu_slots: tuple = ()
    generation: int = 1
    annotations: dict = field(default_factory=dict)
    start_time: float = field(default_factory=time.time)
    restart_count: int = 0
    spot: bool = False   # preemptible overflow replica (SpotTaskSpec)
    proc: object = None  # subprocess.Popen for the local runtime
    pid: int | None = None  # adopted orphan (manager restart): no Popen

    @property
    def key(self):
        return (self.job_name, self.task_type, self.index)

    @property
    def finished(self):
        return self.phase in (TaskPhase.SUCCEEDED, TaskPhase.FAILED)


def task_name(job: str, t: TaskType, index: int) -> str:
    """'<job>-<tasktype>-<index>' (reference utils.go:75-77)."""
    return f"{job}-{t.value}-{index}"


def _world_override(job: TorchJob, default: int) -> int:
    """The elastic WORLD_SIZE annotation (torchjob_controller.go:419-439
    downward-API analog) — but annotations are user-writable strings, so
    a non-numeric or non-positive value falls back to the computed world
    instead of crashing every task at env parse."""
    v = job.annotations.get("world-size")
    if v is None:
        return default
    try:
        n = int(v)
    except ValueError:
        return default
    return n if n >= 1 else default


def cluster_env(job: TorchJob, t: TaskType, index: int,
                master_port: int | None = None) -> dict:
    """The torch env contract (SetClusterSpec parity,
    torchjob_controller.go:332-445): master rank 0, workers index+1;
    WORLD_SIZE excludes the AIMaster; master addr is localhost on the
    single-node design (the reference's TorchLocalMasterAddr gate,
    features.go:31-63, is the natural choice here)."""
    world = job.total_replicas(include_aimaster=False)
    if t == TaskType.MASTER:
        rank = 0
    elif t == TaskType.WORKER:
        # +1 iff a master task exists (reference :339-348)
        rank = index + (1 if TaskType.MASTER in job.tasks else 0)
    else:
        rank = -1  # AIMaster is not part of the process group
    env = {
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(master_port or DEFAULT_MASTER_PORT),
        "WORLD_SIZE": str(_world_override(job, world)),
        "RANK": str(rank),
        "PYTHONUNBUFFERED": "1",  # reference torchjob_controller.go:394-445
        "TOK_JOB_NAME": job.name,
        "TOK_TASK_TYPE": t.value,
        "TOK_TASK_INDEX": str(index),
        "TOK_GENERATION": str(job.generation),
    }
    # torchelastic rendezvous args (torchjob_controller.go:385-392): the
    # in-tree entrypoint rendezvouses over the fast-rejoin TCPStore, but
    # custom-command jobs that run torchrun themselves read these.
    if job.elastic is not None and job.elastic.rdzv_backend:
        env["TOK_RDZV_BACKEND"] = job.elastic.rdzv_backend
        env["TOK_RDZV_ENDPOINT"] = (job.elastic.rdzv_endpoint or
                                    f"127.0.0.1:{master_port or DEFAULT_MASTER_PORT}")
    return env


class Runtime:
    """Abstract task runtime."""

    def start_task(self, job: TorchJob, t: TaskType, index: int,
                   gpu_slots: tuple, extra_env: dict) -> TaskHandle:
        raise NotImplementedError

    def poll(self, h: TaskHandle) -> TaskHandle:
        raise NotImplementedError

    def kill(self, h: TaskHandle, grace: bool = True) -> None:
        raise NotImplementedError


class FakeRuntime(Runtime):
    """envtest analog: the test drives phases."""

    def __init__(self):
        self.tasks: dict = {}
        self.started: list = []
        self.killed: list = []
        # default behavior: tasks move PENDING->RUNNING on first poll
        self.auto_run = True
        # adoption surface (tests seed job -> [task records])
        self.adoptable: dict = {}

    def start_task(self, job, t, index, gpu_slots, extra_env):
        h = TaskHandle(job.name, t, index, gpu_slots=tuple(gpu_slots),
                       generation=job.generation)
        h.env = dict(cluster_env(job, t, index), **extra_env)
        self.tasks[h.key] = h
        self.started.append(h.key)
        return h

    def poll(self, h):
        if self.auto_run and h.phase == TaskPhase.PENDING:
            h.phase = TaskPhase.RUNNING
        return h

    def kill(self, h, grace=True):
        self.killed.append(h.key)
        if not h.finished:
            h.phase = TaskPhase.FAILED
            h.exit_code = 137 if not grace else 143
            h.reason = "Killed"

    # --- test helpers ------------------------------------------------
    def set_phase(self, key, phase, exit_code=None, reason=""):
        h = self.tasks[key]
        h.phase = phase
        h.exit_code = exit_code
        h.reason = reason

    def adoptable_tasks(self, job_name: str) -> list:
        return list(self.adoptable.get(job_name, []))

    def adopt_task(self, job, t, index, rec) -> TaskHandle:
        h = TaskHandle(job.name, t, index,
                       gpu_slots=tuple(rec.get("gpu_slots") or ()),
                       generation=int(rec.get("generation", job.generation)))
        h.pid = int(rec.get("pid", 0))
        h.phase = TaskPhase.RUNNING
        self.tasks[h.key] = h
        return h


class LocalProcessRuntime(Runtime):
    """One OS process per task ('kubelet' analog). GPU isolation via
    HIP_VISIBLE_DEVICES over the node's GPU slots."""

    def __init__(self, workdir: str, python: str | None = None,
                 storage=None):
        self.workdir = workdir
        self.python = python or sys.executable
        self.storage = storage  # StorageProvider: provenance env for tasks
        os.makedirs(workdir, exist_ok=True)

    def start_task(self, job, t, index, gpu_slots, extra_env):
        h = TaskHandle(job.name, t, index, gpu_slots=tuple(gpu_slots),
                       generation=job.generation)
        spec = job.tasks[t]
        env = dict(os.environ)
        if self.storage is not None:
            env.update(self.storage.task_env())
        env.update(cluster_env(job, t, index))
        env.update(spec.env)
        env.update(extra_env)
        if gpu_slots:
            env["HIP_VISIBLE_DEVICES"] = ",".join(str(s) for s in gpu_slots)
            env["LOCAL_RANK"] = "0"  # each process sees exactly its GPUs
        logdir = os.path.join(self.workdir, job.name)
        env.setdefault("TOK_STATE_DIR", logdir)
        os.makedirs(logdir, exist_ok=True)
        argv = spec.command or [
            self.python, "-m", "torch_on_k8s_amd.entrypoint"]
        logf = open(os.path.join(
            logdir, f"{task_name(job.name, t, index)}.log"), "ab")
        h.proc = subprocess.Popen(argv, env=env, stdout=logf, stderr=logf,
                                  start_new_session=True)
        h.phase = TaskPhase.RUNNING
        self._write_task_record(logdir, job, t, index, h)
        return h

    # -- adoption (reference adopt/claim of orphans, pod.go:717-745):
    # a restarted manager rebuilds handles from per-task pid records so
    # running gangs are neither lost nor duplicated -------------------
    @staticmethod
    def _task_record_path(logdir: str, name: str) -> str:
        return os.path.join(logdir, "tasks", f"{name}.json")

    def _write_task_record(self, logdir, job, t, index, h):
        import json
        d = os.path.join(logdir, "tasks")
        os.makedirs(d, exist_ok=True)
        rec = {
            "pid": h.proc.pid,
            "task_type": t.value,
            "index": index,
            "generation": h.generation,
            "gpu_slots": list(h.gpu_slots),
            "spot": h.spot,
            "start_time": h.start_time,
        }
        path = self._task_record_path(logdir, task_name(job.name, t, index))
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(rec, f)
        os.replace(tmp, path)
        try:  # a fresh start invalidates any previous exit marker
            os.unlink(path.replace(".json", ".exit"))
        except OSError:
            pass

    def adoptable_tasks(self, job_name: str) -> list:
        """Live task records for a job (pid still alive). Called by the
        controller at job (re-)creation before it would start fresh
        tasks."""
        import json
        d = os.path.join(self.workdir, job_name, "tasks")
        out = []
        try:
            files = os.listdir(d)
        except OSError:
            return out
        for f in files:
            if not f.endswith(".json"):
                continue
            try:
                with open(os.path.join(d, f)) as fh:
                    rec = json.load(fh)
            except (OSError, ValueError):
                continue
            if "pid" in rec and self._pid_running(int(rec["pid"])):
                out.append(rec)
        return out

    def adopt_task(self, job, t, index, rec) -> TaskHandle:
        """Build a handle over an already-running orphan process."""
        h = TaskHandle(job.name, t, index,
                       gpu_slots=tuple(rec.get("gpu_slots") or ()),
                       generation=int(rec.get("generation", job.generation)))
        h.spot = bool(rec.get("spot"))
        h.start_time = rec.get("start_time", h.start_time)
        h.pid = int(rec["pid"])
        h.phase = TaskPhase.RUNNING
        return h

    def poll(self, h):
        if h.finished:
            return h
        if h.proc is not None:
            rc = h.proc.poll()
            if rc is not None:
                h.exit_code = rc
                h.phase = TaskPhase.SUCCEEDED if rc == 0 else TaskPhase.FAILED
            return h
        if h.pid is not None:  # adopted orphan: not our child
            if self._pid_running(h.pid):
                return h  # still alive

            # gone: the framework entrypoint leaves an exit marker; an
            # opaque command that died while orphaned can't be graded
            rc = self._read_exit_marker(h)
            if rc is None:
                h.exit_code = None
                h.phase = TaskPhase.FAILED
                h.reason = "AdoptedExit"
            else:
                h.exit_code = rc
                h.phase = TaskPhase.SUCCEEDED if rc == 0 else TaskPhase.FAILED
        return h

    @staticmethod
    def _pid_running(pid: int) -> bool:
        """Alive and NOT a zombie (os.kill(pid, 0) succeeds on zombies;
        an exited-but-unreaped task must grade as finished)."""
        try:
            os.kill(pid, 0)
        except ProcessLookupError:
            return False
        except PermissionError:
            return True
        try:
            with open(f"/proc/{pid}/stat") as f:
                # state is the field after the parenthesised comm
                return f.read().rpartition(")")[2].split()[0] != "Z"
        except (OSError, IndexError):
            return False

    def _read_exit_marker(self, h):
        path = self._task_record_path(
            os.path.join(self.workdir, h.job_name),
            task_name(h.job_name, h.task_type, h.index)).replace(
                ".json", ".exit")
        try:
            with open(path) as f:
                return int(f.read().strip())
        except (OSError, ValueError):
            return None

    def _live_pid(self, h):
        if h.proc is not None:
            return h.proc.pid if h.proc.poll() is None else None
        if h.pid is not None:
            return h.pid if self._pid_running(h.pid) else None
        return None

    def kill(self, h, grace=True):
        pid = self._live_pid(h)
        if pid is None:
            return
        try:
            pgid = os.getpgid(pid)
            os.killpg(pgid, signal.SIGTERM if grace else signal.SIGKILL)
        except ProcessLookupError:
            pass

    def wait(self, h, timeout=60):
        if h.proc is not None:
            try:
                h.proc.wait(timeout=timeout)
            except subprocess.TimeoutExpired:
                self.kill(h, grace=False)
                h.proc.wait(timeout=10)
        elif h.pid is not None:
            deadline = time.time() + timeout
            while self._live_pid(h) is not None:
                if time.time() > deadline:
                    self.kill(h, grace=False)
                time.sleep(0.1)
        return self.poll(h)
