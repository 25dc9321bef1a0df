"""YAML/dict TorchJob specs — the kubectl-facing surface.

The reference's CRD YAML (config/crd/bases/, samples in config/samples/)
maps to this schema:

    apiVersion: train.distributed.io/v1alpha1   # accepted, informational
    kind: TorchJob
    metadata:
      name: llama-dp8
      namespace: default
      annotations: {enable-elastic-training: "true"}
    spec:
      schedulingPolicy: {queue: teamA, priority: 10, minAvailable: 8}
      runPolicy: {backoffLimit: 3, activeDeadlineSeconds: 3600,
                  ttlSecondsAfterFinished: 600, cleanTaskPolicy: Running}
      elasticPolicy: {minReplicas: 2, maxReplicas: 8, metricWindow: 5}
      modelName: llama3-8b
      tasks:
        master: {replicas: 1, gpusPerTask: 1, restartPolicy: OnExitCode,
                 env: {TOK_TRAIN_STEPS: "100"}}
        worker: {replicas: 7, gpusPerTask: 1}
"""
from __future__ import annotations

from torch_on_k8s_amd.controlplane.api import (CleanPodPolicy, DAGCondition,
                                               ElasticPolicy, RestartPolicy,
                                               RunPolicy, SchedulingPolicy,
                                               SpotTaskSpec, TaskPhase,
                                               TaskSpec, TaskType, TorchJob,
                                               set_defaults)


def _task_spec(d: dict) -> TaskSpec:
    spec = TaskSpec(
        replicas=int(d.get("replicas", 1)),
        gpus_per_task=int(d.get("gpusPerTask", 1)),
        cpus_per_task=float(d.get("cpusPerTask", 0)),
        mem_mb_per_task=int(d.get("memMbPerTask", 0)),
        command=d.get("command"),
        env={str(k): str(v) for k, v in (d.get("env") or {}).items()},
    )
    if d.get("restartPolicy"):
        spec.restart_policy = RestartPolicy(d["restartPolicy"])
    for c in d.get("dagConditions") or []:
        spec.dag_conditions.append(DAGCondition(
            upstream=TaskType(c["upstream"].lower()),
            on_phase=TaskPhase(c.get("onPhase", "Running"))))
    if d.get("spot"):
        s = d["spot"]
        spec.spot = SpotTaskSpec(
            num_spot_replicas=int(s.get("numSpotReplicas", 0)),
            priority=int(s.get("priority", -10)))
    return spec


def job_from_dict(doc: dict) -> TorchJob:
    meta = doc.get("metadata") or {}
    spec = doc.get("spec") or {}
    tasks = {TaskType(t.lower()): _task_spec(s or {})
             for t, s in (spec.get("tasks") or {}).items()}
    job = TorchJob(
        name=meta.get("name", "torchjob"),
        namespace=meta.get("namespace", "default"),
        annotations={str(k): str(v)
                     for k, v in (meta.get("annotations") or {}).items()},
        labels=dict(meta.get("labels") or {}),
        tasks=tasks,
        model_name=spec.get("modelName"),
    )
    sp = spec.get("schedulingPolicy") or {}
    job.scheduling = SchedulingPolicy(
        min_available=sp.get("minAvailable"),
        queue=sp.get("queue", ""),
        priority=sp.get("priority"))
    rp = spec.get("runPolicy") or {}
    job.run_policy = RunPolicy(
        clean_task_policy=CleanPodPolicy(rp.get("cleanTaskPolicy", "Running")),
        ttl_seconds_after_finished=rp.get("ttlSecondsAfterFinished"),
        active_deadline_seconds=rp.get("activeDeadlineSeconds"),
        backoff_limit=int(rp.get("backoffLimit", 3)))
    ep = spec.get("elasticPolicy")
    if ep:
        job.elastic = ElasticPolicy(
            min_replicas=int(ep.get("minReplicas", 1)),
            max_replicas=int(ep.get("maxReplicas", 1)),
            nproc_per_node=int(ep.get("nprocPerNode", 1)),
            metric_window=int(ep.get("metricWindow", 5)),
            max_num_metrics=int(ep.get("maxNumMetrics", 50)))
    return set_defaults(job)


def job_from_yaml(text: str) -> TorchJob:
    import yaml
    return job_from_dict(yaml.safe_load(text))


def job_status_dict(job: TorchJob, events=None) -> dict:
    return {
        "events": [
            {"type": e.type, "reason": e.reason, "message": e.message,
             "ts": e.ts}
            for e in (events or [])][-20:],
        "name": job.name,
        "phase": job.status.phase.value if job.status.phase else None,
        "generation": job.generation,
        "restartCount": job.status.restart_count,
        "conditions": [
            {"type": c.type.value, "reason": c.reason, "ts": c.ts}
            for c in job.status.conditions],
        "tasks": {t.value: {"active": s.active, "succeeded": s.succeeded,
                            "failed": s.failed}
                  for t, s in job.status.tasks.items()},
        "modelVersion": job.status.model_version,
    }
