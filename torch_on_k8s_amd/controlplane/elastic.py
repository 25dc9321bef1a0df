"""Elastic scaling: the 2-stage checkpoint transaction + generation-based
scale (reference controllers/train/elastic_scale.go) and the metric-driven
torchelastic autoscaler (controllers/train/torchelastic/).

Protocol parity (elastic_scale.go:49-56,132-297 / SURVEY.md §3.3, §5.4):
  * annotations on the job are the dynamic channel:
      ckpt-requested-version = {version: generation, status: InProgress}
      ckpt-completed-version = {version, status: Succeeded}
      ready-to-start-worker  = "true"/"false"
      scale-state            = inflight | done
      world-size             = int (delivered to restarted tasks via env)
  * scale-out/in: controller requests a checkpoint at the current
    generation; the data plane (AIMaster role — here the trainer's
    checkpoint agent, engine/agent.py) writes it and patches
    completed-version; the controller then removes victims, bumps the
    generation, refreshes WORLD_SIZE and restarts stale tasks.

Metric-driven autoscaler parity (torchelastic/):
  * per-job decision loop; observations per replica-count
  * replicas double while latency-per-replica improves
    (computeNewReplicas x2, torchelastic/job.go:94-104), revert otherwise
  * conditions Start/Stop/Continue/ReachMaxMetric/ReachMaxReplicas
  * observations come from the trainer's STRUCTURED metrics file
    (engine/trainer.py), replacing the reference's worker-0 stdout
    regex scrape (torchelastic/observation.go:40-106). The reference's
    GetPodsForJob panic ("Implement me", torchelastic/pod.go:24-26) has
    no analog here — restart of stale tasks is implemented.
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass

from torch_on_k8s_amd.controlplane.api import (ANN_CKPT_COMPLETED,
                                               ANN_CKPT_REQUESTED,
                                               ANN_ENABLE_ELASTIC,
                                               ANN_READY_TO_START_WORKER,
                                               ANN_SCALE_STATE, ANN_WORLD_SIZE,
                                               CKPT_IN_PROGRESS,
                                               CKPT_SUCCEEDED, ElasticCondition,
                                               ElasticStatus, TaskType,
                                               TorchJob)


class ElasticScaler:
    """Generation-based scale with checkpoint gating. Driven from the
    controller's reconcile (controller.py calls reconcile())."""

    def enabled(self, job: TorchJob) -> bool:
        return job.annotations.get(ANN_ENABLE_ELASTIC) == "true" or \
            job.elastic is not None

    # -- the AIMaster-side hook (data plane) ---------------------------
    @staticmethod
    def complete_checkpoint(job: TorchJob):
        """Called by the checkpoint agent once the checkpoint at the
        requested version is durably written (elastic_scale.go: AIMaster
        patches completed-version)."""
        req = job.annotations.get(ANN_CKPT_REQUESTED)
        if not req:
            return
        req = json.loads(req) if isinstance(req, str) else req
        job.annotations[ANN_CKPT_COMPLETED] = json.dumps(
            {"version": req["version"], "status": CKPT_SUCCEEDED})

    # -- controller-side protocol --------------------------------------
    def reconcile(self, ctl, job: TorchJob, hs: dict) -> bool:
        """Returns True if an elastic transaction is in flight and the
        normal reconcile should hold off (job.go:221-248)."""
        desired = job.tasks.get(TaskType.WORKER)
        if desired is None:
            return False
        # A scale transaction is needed iff a generation bump left stale
        # handles behind, or scale-in victims exist. Plain creation and
        # failover recreates run at the current generation and never
        # trigger the checkpoint protocol (the reference guards this with
        # generation > 1, job.go:236-247).
        stale = [h for h in hs.values() if h.generation != job.generation]
        victims = [h for h in hs.values()
                   if h.task_type == TaskType.WORKER and
                   h.index >= desired.replicas]
        needs_scale = bool(stale) or bool(victims)
        if not needs_scale:
            # transaction epilogue: clear state
            if job.annotations.get(ANN_SCALE_STATE) == "inflight":
                job.annotations[ANN_SCALE_STATE] = "done"
                job.annotations[ANN_READY_TO_START_WORKER] = "false"
                ctl.event(job.name, "Normal", "ScaleSucceed",
                          f"generation {job.generation}")
            return False

        if not hs:
            return False  # nothing running yet: plain create path

        # stage 1: request a checkpoint at this generation
        req = job.annotations.get(ANN_CKPT_REQUESTED)
        req = json.loads(req) if isinstance(req, str) and req else req
        if not req or req.get("version") != job.generation:
            job.annotations[ANN_CKPT_REQUESTED] = json.dumps(
                {"version": job.generation, "status": CKPT_IN_PROGRESS})
            ctl.event(job.name, "Normal", "CheckpointRequested",
                      f"version {job.generation}")
            return True

        # stage 2: wait for completion by the data plane
        comp = job.annotations.get(ANN_CKPT_COMPLETED)
        comp = json.loads(comp) if isinstance(comp, str) and comp else comp
        if not comp or comp.get("version") != job.generation:
            return True  # checkpoint still in flight

        # checkpoint done: perform the scale
        job.annotations[ANN_SCALE_STATE] = "inflight"
        job.annotations[ANN_READY_TO_START_WORKER] = "true"
        new_world = desired.replicas + \
            (1 if TaskType.MASTER in job.tasks else 0)
        job.annotations[ANN_WORLD_SIZE] = str(new_world)

        # Fast-rejoin (r1 VERDICT next-#9): tasks running the framework
        # entrypoint keep their process (model/optimizer state resident)
        # across the scale — they tear down and re-init only the process
        # group (entrypoint._fast_rejoin). Survivors are ADOPTED into
        # the new generation instead of killed; only scale-in victims
        # die. Opaque command tasks can't rejoin -> full restart path.
        rejoin = self.rejoin_capable(job)

        # Kill victims (scale-in) — and, without rejoin, every
        # stale-generation task — then WAIT for them to exit before the
        # normal pass recreates anything: the new gang reuses the job's
        # master port, and racing a dying master's rendezvous store
        # hangs or mis-wires the new ranks (restartStalePod analog,
        # elastic_scale.go:303-397; the reference gets the equivalent
        # ordering from Kruise CRR completion before patching the
        # generation label).
        pending = False
        adopted = False
        for h in list(hs.values()):
            victim = (h.task_type == TaskType.WORKER and
                      h.index >= desired.replicas)
            stale = h.generation != job.generation
            if not (victim or stale):
                continue
            if rejoin and not victim:
                h.generation = job.generation  # adopt; process rejoins
                adopted = True
                continue
            if not h.finished:
                ctl.runtime.kill(h)
                ctl.runtime.poll(h)
            if h.finished:
                if h.gpu_slots:
                    ctl.node.release(h.gpu_slots)
                    h.gpu_slots = ()
                hs.pop(h.key, None)
            else:
                pending = True
        if pending:
            return True  # victims still terminating; hold recreation
        if adopted:
            # hold new-task creation until the surviving master reports
            # its old rendezvous store closed (agent.json rejoin-ready,
            # lifted into annotations by _sync_state_files) — otherwise
            # fresh ranks could join the stale store
            ready = job.annotations.get("rejoin-ready")
            if str(ready) != str(job.generation):
                return True
        ctl.event(job.name, "Normal", "ScaleExecuted",
                  f"world_size={new_world} gen={job.generation} "
                  f"rejoin={rejoin}")
        return False  # let the normal pass recreate tasks now

    @staticmethod
    def rejoin_capable(job: TorchJob) -> bool:
        """Fast-rejoin needs every gang member to run the framework
        entrypoint (opaque commands can't re-init in place)."""
        if job.annotations.get("disable-fast-rejoin") == "true":
            return False
        return all(s.command is None for t, s in job.tasks.items()
                   if t != TaskType.AIMASTER)

    # -- user-facing scale API (AIMaster analog) -----------------------
    @staticmethod
    def scale(job: TorchJob, replicas: int):
        """Change desired worker replicas; bumps the generation like a
        k8s spec update would (elastic_scale.go:519-546 generation
        self-increment)."""
        job.tasks[TaskType.WORKER].replicas = replicas
        job.generation += 1


@dataclass
class Observation:
    step: int
    latency: float   # step_time_s
    tokens_per_s: float
    loss: float


class TorchElasticAutoscaler:
    """Metric-driven autoscaler (torchelastic/elastic_scale.go:42-246).

    Decision per pass (given >= metric_window fresh observations at the
    current replica count):
      * if latency-per-replica improved vs the previous replica count:
        double replicas (cap at max) -> CONTINUE / MAX_REPLICAS
      * else revert to last replicas and stop -> STOP
      * too many total samples -> MAX_METRIC
    """

    def __init__(self, read_metrics_fn):
        # read_metrics_fn(job) -> Observation | None (structured metrics)
        self.read_metrics = read_metrics_fn

    def observe(self, job: TorchJob) -> None:
        if job.elastic is None:
            return
        st = job.status.elastic
        if st is None:
            st = job.status.elastic = ElasticStatus(
                replicas=job.tasks[TaskType.WORKER].replicas,
                last_replicas=job.tasks[TaskType.WORKER].replicas)
        obs = self.read_metrics(job)
        if obs is None:
            return
        lst = st.observations.setdefault(st.replicas, [])
        if lst and lst[-1].step == obs.step:
            return  # no new sample
        lst.append(obs)

    def decide(self, job: TorchJob) -> ElasticCondition | None:
        """Returns the action taken (None = keep observing)."""
        pol = job.elastic
        st = job.status.elastic
        if pol is None or st is None:
            return None
        samples = st.observations.get(st.replicas, [])
        total = sum(len(v) for v in st.observations.values())
        if total >= pol.max_num_metrics:
            st.condition = ElasticCondition.MAX_METRIC
            st.continue_training = False
            return st.condition
        if len(samples) < pol.metric_window:
            return None

        if not self._improved(st):
            # revert and stop (torchelastic/elastic_scale.go:186-233)
            prev = st.last_replicas
            st.condition = ElasticCondition.STOP
            st.continue_training = False
            if prev and prev != st.replicas:
                ElasticScaler.scale(job, prev)
                st.replicas = prev
            return st.condition
        if st.replicas >= pol.max_replicas:
            st.condition = ElasticCondition.MAX_REPLICAS
            return st.condition
        new = min(st.replicas * 2, pol.max_replicas)  # computeNewReplicas x2
        st.last_replicas = st.replicas
        st.replicas = new
        st.condition = ElasticCondition.CONTINUE
        ElasticScaler.scale(job, new)
        return st.condition

    @staticmethod
    def _improved(st: ElasticStatus) -> bool:
        """IsSatisfyElasticContinue parity (torchelastic/job.go:94-100):
        latency per replica must improve vs the previous replica count."""
        cur = st.observations.get(st.replicas, [])
        prev = st.observations.get(st.last_replicas, [])
        if not cur:
            return False
        if st.last_replicas == st.replicas or not prev:
            return True  # first scale decision: allow growth
        cur_lat = sum(o.latency for o in cur) / len(cur)
        prev_lat = sum(o.latency for o in prev) / len(prev)
        # lastLat/lastN > curLat/curN  <=>  cur_lat*lastN < prev_lat*curN
        # (latency per replica must drop; with DP more replicas process
        # proportionally more data per step, so flat step latency at 2x
        # replicas IS an improvement)
        return cur_lat * st.last_replicas < prev_lat * st.replicas


def read_trainer_metrics(metrics_path_fn):
    """Builds a read_metrics_fn from per-job metrics file paths."""
    def _read(job: TorchJob) -> Observation | None:
        path = metrics_path_fn(job)
        try:
            with open(path) as f:
                rec = json.load(f)
            lat = rec["step_time_s"]
            if not isinstance(lat, (int, float)):
                return None  # custom entrypoints may write null/strings;
                # a non-numeric latency must not poison the decide() math
            return Observation(step=int(rec["step"]), latency=float(lat),
                               tokens_per_s=rec.get("tokens_per_s") or 0.0,
                               loss=rec.get("loss") or 0.0)
        except (OSError, ValueError, TypeError, KeyError):
            return None
    return _read
