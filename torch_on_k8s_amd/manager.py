"""Framework manager daemon — the operator process (reference main.go).

    python -m torch_on_k8s_amd.manager --workdir /var/run/tok \
        --num-gpus 8 --metrics-addr 8443 \
        --feature-gates GangScheduling=true,JobCoordinator=true \
        --quota teamA=8 --quota teamB=4

Watches <workdir>/spool/ for TorchJob YAMLs (the kubectl-apply analog),
runs the coordinator (100ms schedule loop), the reconcile loop, the
metric-driven elastic autoscaler pass, and the Prometheus metrics server.
Job status is published to <workdir>/status/<job>.json; deleting the
spool file deletes the job.
"""
from __future__ import annotations

import argparse
import json
import os
import time

from torch_on_k8s_amd.controlplane import features as feat
from torch_on_k8s_amd.controlplane.api import JobConditionType
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.coordinator import Coordinator
from torch_on_k8s_amd.controlplane.elastic import (ElasticScaler,
                                                   TorchElasticAutoscaler,
                                                   read_trainer_metrics)
from torch_on_k8s_amd.controlplane.jobspec import (job_from_yaml,
                                                   job_status_dict)
from torch_on_k8s_amd.controlplane.metrics import (JobMetrics,
                                                   start_metrics_server)
from torch_on_k8s_amd.controlplane.modelregistry import (ModelRegistry,
                                                         storage_from_spec)
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import LocalProcessRuntime
from torch_on_k8s_amd.utils.logging import get_logger

log = get_logger("manager")


class Manager:
    def __init__(self, workdir: str, num_gpus: int = 8, quotas=None,
                 gates: feat.FeatureGates | None = None,
                 sync_period: float = 0.5, storage_spec: dict | None = None,
                 priority_classes: dict | None = None,
                 kill_grace_seconds: float = 60.0):
        self.workdir = workdir
        self.spool = os.path.join(workdir, "spool")
        self.status_dir = os.path.join(workdir, "status")
        for d in (self.spool, self.status_dir):
            os.makedirs(d, exist_ok=True)
        self.gates = gates or feat.FeatureGates()
        self.sync_period = sync_period

        node = NodeState(num_gpus=num_gpus)
        storage = storage_from_spec(os.path.join(workdir, "models"),
                                    storage_spec)
        runtime = LocalProcessRuntime(os.path.join(workdir, "jobs"),
                                      storage=storage)
        self.registry = ModelRegistry(storage)
        self.metrics = JobMetrics()
        cfg = ControllerConfig(
            enable_gang_scheduling=self.gates.enabled(feat.GANG_SCHEDULING),
            enable_dag_scheduling=self.gates.enabled(feat.DAG_SCHEDULING),
            priority_classes=dict(priority_classes or {}),
            kill_grace_seconds=kill_grace_seconds)
        self.controller = JobController(
            node, runtime, cfg, metrics=self.metrics,
            model_registry=self.registry, elastic=ElasticScaler())
        if self.gates.enabled(feat.JOB_COORDINATOR):
            self.coordinator = Coordinator(
                dequeue_fn=self.controller.reconcile,
                tenant_usage_fn=self.controller.tenant_resource_usage,
                quotas=quotas, default_quota=num_gpus,
                priority_classes=priority_classes,
                event_fn=lambda job, reason, msg: self.controller.event(
                    job.name, "Warning", reason, msg))
            self.controller.coordinator = self.coordinator
        else:
            self.coordinator = None
        self.autoscaler = TorchElasticAutoscaler(
            read_trainer_metrics(lambda job: os.path.join(
                workdir, "jobs", job.name, "metrics.json")))
        self._spooled: dict = {}  # filename -> (job name, mtime)
        self._last_status: dict = {}  # job -> serialized status (skip
                                      # unchanged writes; ~2 Hz loop)

    # ------------------------------------------------------------------
    def sync_spool(self):
        """kubectl-apply/delete analog over the spool directory."""
        try:
            files = {f for f in os.listdir(self.spool)
                     if f.endswith((".yaml", ".yml", ".json"))}
        except OSError:
            return
        for f in sorted(files):
            path = os.path.join(self.spool, f)
            try:
                mtime = os.path.getmtime(path)
            except OSError:
                continue
            prev = self._spooled.get(f)
            if prev is not None and prev[1] == mtime:
                continue
            try:
                with open(path) as fh:
                    job = job_from_yaml(fh.read())
            except Exception as e:  # malformed spec: surface, skip
                self.controller.event("-", "Warning", "BadJobSpec",
                                      f"{f}: {e}")
                self._spooled[f] = ("", mtime)
                continue
            existing = self.controller.jobs.get(job.name)
            if existing is None:
                self.metrics.job_created_at(job.name)
                self.controller.create_job(job)
            elif prev is not None:
                # spec UPDATE (OnOwnerUpdate analog): replica changes go
                # through the elastic generation bump so running gangs
                # checkpoint + restart at the new world size
                self._apply_update(existing, job)
            else:
                # a DIFFERENT spool file already owns this job name
                # (k8s would reject the create: names are unique). Track
                # the file without ownership so removing it can't delete
                # the other file's job.
                self.controller.event(
                    job.name, "Warning", "DuplicateJobName",
                    f"{f} ignored: job {job.name!r} already exists from "
                    f"another spool file")
                self._spooled[f] = ("", mtime)
                continue
            self._spooled[f] = (job.name, mtime)
        for f in set(self._spooled) - files:   # spool file removed
            name, _ = self._spooled.pop(f)
            if name:
                self.controller.delete_job(name)

    def _apply_update(self, existing, desired):
        from torch_on_k8s_amd.controlplane.api import TaskType
        changed = False
        for t, spec in desired.tasks.items():
            cur = existing.tasks.get(t)
            if cur is not None and cur.replicas != spec.replicas:
                cur.replicas = spec.replicas
                changed = True
        existing.annotations.update(desired.annotations)
        if changed:
            existing.generation += 1
            self.controller.event(existing.name, "Normal", "SpecUpdated",
                                  f"generation {existing.generation}")

    def publish_status(self):
        live = set()
        for name, job in list(self.controller.jobs.items()):
            live.add(f"{name}.json")
            path = os.path.join(self.status_dir, f"{name}.json")
            evs = self.controller.events_for(name)
            doc = json.dumps(job_status_dict(job, evs), indent=2)
            if self._last_status.get(name) != doc:
                try:
                    tmp = path + ".tmp"
                    with open(tmp, "w") as f:
                        f.write(doc)
                    os.replace(tmp, path)
                    self._last_status[name] = doc
                except OSError as e:  # disk full etc.: stale status
                    # beats taking every tenant's controller down
                    log.warning("status write failed for %s: %s", name, e)
            # data-plane gauges from the trainer's structured metrics
            try:
                with open(os.path.join(self.workdir, "jobs", name,
                                       "metrics.json")) as f:
                    rec = json.load(f)
                if isinstance(rec, dict):  # metrics.json is task-written
                    self.metrics.set_training_metrics(name, rec)
            except (OSError, ValueError, TypeError):
                pass
        # drop status files of deleted jobs (daemon hygiene)
        try:
            for f in os.listdir(self.status_dir):
                if f.endswith(".json") and f not in live:
                    os.unlink(os.path.join(self.status_dir, f))
                    self._last_status.pop(f[:-5], None)
                    self.metrics.remove_job(f[:-5])
        except OSError:
            pass
        if self.coordinator is not None:
            for tenant, q in list(self.coordinator.queues.items()):
                self.metrics.set_queue_depth(tenant, len(q))

    def autoscale_pass(self):
        for job in list(self.controller.jobs.values()):
            if job.elastic is None:
                continue
            if job.status.phase != JobConditionType.RUNNING:
                continue
            try:
                self.autoscaler.observe(job)
                decision = self.autoscaler.decide(job)
            except Exception as e:  # same isolation as reconcile_all:
                # one job's bad metrics must not stop every autoscaler
                self.controller.event(job.name, "Warning", "AutoscaleError",
                                      f"{type(e).__name__}: {e}")
                continue
            if decision is not None:
                st = job.status.elastic
                self.controller.event(
                    job.name, "Normal", f"Elastic{decision.value}",
                    f"autoscaler: replicas {st.last_replicas} -> "
                    f"{st.replicas}" if st else "")

    def step(self):
        self.sync_spool()
        if self.coordinator is not None:
            self.coordinator.schedule_once()
        self.controller.reconcile_all()
        self.autoscale_pass()
        self.publish_status()

    def acquire_leadership(self, block: bool = True):
        """Leader election analog (reference main.go:77-83, election id
        torch-on-k8s-election): an exclusive flock on
        <workdir>/manager.lock. Two daemons on one workdir would both
        start tasks and oversubscribe the node's real GPUs; the lock
        makes the second one wait (block=True, takeover on leader exit,
        the k8s semantics) or fail fast (block=False). Held for the
        process lifetime; the OS releases it on ANY exit, so a crashed
        leader never wedges the next one."""
        import fcntl
        # "a" not "w": opening must NOT truncate — a second daemon
        # blocked on the flock would otherwise wipe the live leader's
        # pid record the moment it starts waiting
        self._lock_file = open(os.path.join(self.workdir, "manager.lock"),
                               "a")
        try:
            fcntl.flock(self._lock_file,
                        fcntl.LOCK_EX | (0 if block else fcntl.LOCK_NB))
        except OSError:
            self._lock_file.close()
            self._lock_file = None
            raise RuntimeError(
                f"another manager owns {self.workdir} (manager.lock held)")
        self._lock_file.seek(0)
        self._lock_file.truncate(0)
        self._lock_file.write(f"{os.getpid()}\n")
        self._lock_file.flush()

    def release_leadership(self):
        if getattr(self, "_lock_file", None) is not None:
            self._lock_file.close()  # closing the fd drops the flock
            self._lock_file = None

    def run_forever(self):
        """Main loop with graceful shutdown: on SIGTERM/SIGINT, publish
        final statuses and exit 0 WITHOUT killing task processes — a
        restarted manager adopts them (controller._adopt_orphans), so a
        manager upgrade never interrupts running gangs."""
        self.acquire_leadership(block=True)
        import signal as _signal
        stop = {"flag": False}

        def _stop(*_):
            stop["flag"] = True

        _signal.signal(_signal.SIGTERM, _stop)
        _signal.signal(_signal.SIGINT, _stop)
        while not stop["flag"]:
            self.step()
            time.sleep(self.sync_period)
        self.publish_status()
        log.info("manager shutting down; %d job(s) keep running for "
                 "adoption by the next manager", len(self.controller.jobs))


def main():
    ap = argparse.ArgumentParser(prog="torch-on-k8s-amd-manager")
    ap.add_argument("--workdir", default="/tmp/torch-on-k8s-amd")
    ap.add_argument("--num-gpus", type=int,
                    default=int(os.environ.get("TOK_NUM_GPUS", "8")))
    ap.add_argument("--metrics-addr", type=int, default=8443)
    ap.add_argument("--feature-gates", default="")
    ap.add_argument("--quota", action="append", default=[],
                    help='tenant=gpus or tenant={"gpu":8,"cpu":64,'
                         '"memory_mb":512000}; repeatable')
    ap.add_argument("--sync-period", type=float, default=0.5)
    ap.add_argument("--priority-class", action="append", default=[],
                    help="name=value PriorityClass-object analog consulted "
                         "when a job sets priorityClassName; repeatable")
    ap.add_argument("--kill-grace", type=float, default=60.0,
                    help="SIGTERM->SIGKILL window in seconds for task "
                         "cleanup (terminationGracePeriodSeconds default)")
    ap.add_argument("--storage", default="",
                    help='storage spec JSON, e.g. {"nfs": {"server": '
                         '"10.0.0.2", "path": "/exports/models"}} or '
                         '{"localStorage": {"path": "/data/models"}}')
    args = ap.parse_args()

    quotas = {}
    for q in args.quota:
        k, _, v = q.partition("=")
        try:
            quotas[k] = int(v)
        except ValueError:
            quotas[k] = json.loads(v)  # resource-map quota
    gates = feat.FeatureGates.from_flag(args.feature_gates)
    prio_classes = {}
    for pc in args.priority_class:
        k, _, v = pc.partition("=")
        prio_classes[k] = int(v)
    mgr = Manager(args.workdir, num_gpus=args.num_gpus,
                  quotas=quotas or None, gates=gates,
                  sync_period=args.sync_period,
                  storage_spec=json.loads(args.storage) if args.storage
                  else None,
                  priority_classes=prio_classes or None,
                  kill_grace_seconds=args.kill_grace)
    start_metrics_server(args.metrics_addr)
    log.info("workdir=%s gpus=%d gates=%s", args.workdir, args.num_gpus,
             gates.as_dict())
    mgr.run_forever()


if __name__ == "__main__":
    main()
