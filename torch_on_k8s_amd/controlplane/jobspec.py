"""YAML/dict TorchJob specs — the kubectl-facing surface.

TWO dialects are accepted (auto-detected per document):

1. The native node dialect:

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
        worker: {replicas: 7, gpusPerTask: 1,
                 cpusPerTask: 8, memMbPerTask: 65536}  # quota-tracked

2. The reference CRD dialect (group train.distributed.io/v1alpha1,
   apis/train/v1alpha1/torchjob_types.go:88-206): `spec.torchTaskSpecs`
   maps TaskType -> {numTasks, restartPolicy, spotTaskSpec, template};
   the pod template's `torch` container carries command/env/resources.
   GPU resources may be named `amd.com/gpu` (this framework's node) or
   `nvidia.com/gpu` (reference manifests; accepted and mapped 1:1 so a
   job written against the reference parses unchanged). RunPolicy fields
   are inlined on spec — including the reference's `clenPodPolicy`
   spelling (torchjob_types.go:142) — plus `minMembers`,
   `enableTorchElastic`/`torchElasticPolicy` and `modelVersion`.
"""
from __future__ import annotations

from torch_on_k8s_amd.controlplane.api import (CleanPodPolicy, DAGCondition,
                                               ElasticPolicy, RestartPolicy,
                                               RunPolicy, SchedulingPolicy,
                                               SpotTaskSpec, TaskPhase,
                                               TaskSpec, TaskType, TorchJob,
                                               set_defaults)

GPU_RESOURCES = ("amd.com/gpu", "nvidia.com/gpu")


def parse_quantity(q) -> float:
    """k8s resource quantity ('4', '500m', '8Gi', '2G') -> float in base
    units (memory suffixes return bytes)."""
    if isinstance(q, (int, float)):
        return float(q)
    s = str(q).strip()
    suffixes = {"Ki": 1 << 10, "Mi": 1 << 20, "Gi": 1 << 30, "Ti": 1 << 40,
                "K": 10**3, "M": 10**6, "G": 10**9, "T": 10**12,
                "k": 10**3, "m": 1e-3}
    for suf in sorted(suffixes, key=len, reverse=True):
        if s.endswith(suf):
            return float(s[:-len(suf)]) * suffixes[suf]
    return float(s)


def _check_non_negative(**named) -> None:
    """Resource amounts must be >= 0; a negative value is a spec error
    (surfaced as a BadJobSpec event by the manager), not a default."""
    for name, v in named.items():
        if v is not None and v < 0:
            raise ValueError(f"negative {name}: {v}")


def _task_spec(d: dict) -> TaskSpec:
    spec = TaskSpec(
        replicas=int(d.get("replicas", 1)),
        gpus_per_task=int(d.get("gpusPerTask", 1)),
        cpus_per_task=float(d.get("cpusPerTask", 0)),
        mem_mb_per_task=int(d.get("memMbPerTask", 0)),
        command=d.get("command"),
        env={str(k): str(v) for k, v in (d.get("env") or {}).items()},
    )
    if d.get("terminationGracePeriodSeconds") is not None:
        spec.termination_grace_seconds = float(
            d["terminationGracePeriodSeconds"])
    _check_non_negative(gpusPerTask=spec.gpus_per_task,
                        cpusPerTask=spec.cpus_per_task,
                        memMbPerTask=spec.mem_mb_per_task)
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
            priority=int(s.get("priority", -10)),
            priority_class_name=s.get("priorityClassName", ""))
    return spec


_CRD_RESTART = {"ExitCode": RestartPolicy.ON_EXIT_CODE,
                "OnFailure": RestartPolicy.ON_FAILURE,
                "Always": RestartPolicy.ALWAYS,
                "Never": RestartPolicy.NEVER}


def _task_spec_from_crd(d: dict) -> TaskSpec:
    """TaskSpec from the reference shape: numTasks + pod template whose
    `torch` (or first) container carries command/env/resources
    (torchjob_types.go:88-104, constants.go:96-103)."""
    spec = TaskSpec(replicas=int(d.get("numTasks", 1)))
    rp = d.get("restartPolicy")
    if rp:
        spec.restart_policy = _CRD_RESTART.get(rp) or RestartPolicy(rp)
    st = d.get("spotTaskSpec")
    if st:
        spec.spot = SpotTaskSpec(
            num_spot_replicas=int(st.get("numSpotTasks", 0)),
            priority_class_name=st.get("priorityClassName", ""),
            labels=dict(st.get("labels") or {}))
    tmpl = (d.get("template") or {}).get("spec") or {}
    if tmpl.get("terminationGracePeriodSeconds") is not None:
        spec.termination_grace_seconds = float(
            tmpl["terminationGracePeriodSeconds"])
    containers = tmpl.get("containers") or []
    cont = None
    for c in containers:
        if c.get("name") == "torch":  # reference default container name
            cont = c
            break
    if cont is None and containers:
        cont = containers[0]
    if cont:
        if cont.get("command"):
            spec.command = list(cont["command"]) + list(cont.get("args") or [])
        for e in cont.get("env") or []:
            if "name" in e:
                spec.env[str(e["name"])] = str(e.get("value", ""))
        res = cont.get("resources") or {}
        req = dict(res.get("requests") or {})
        req.update(res.get("limits") or {})  # limits win (GPU convention)
        for gpu_key in GPU_RESOURCES:
            if gpu_key in req:
                spec.gpus_per_task = int(parse_quantity(req[gpu_key]))
                break
        else:
            spec.gpus_per_task = 0
        if "cpu" in req:
            spec.cpus_per_task = parse_quantity(req["cpu"])
        if "memory" in req:
            spec.mem_mb_per_task = int(parse_quantity(req["memory"]) /
                                       (1 << 20))
    _check_non_negative(gpu=spec.gpus_per_task, cpu=spec.cpus_per_task,
                        memory=spec.mem_mb_per_task)
    return spec


def _job_from_crd_dict(doc: dict) -> TorchJob:
    """Reference-CRD-dialect parser (torchjob_types.go:178-206 shape)."""
    meta = doc.get("metadata") or {}
    spec = doc.get("spec") or {}
    tasks = {TaskType(t.lower()): _task_spec_from_crd(s or {})
             for t, s in (spec.get("torchTaskSpecs") or {}).items()}
    job = TorchJob(
        name=meta.get("name", "torchjob"),
        namespace=meta.get("namespace", "default"),
        annotations={str(k): str(v)
                     for k, v in (meta.get("annotations") or {}).items()},
        labels=dict(meta.get("labels") or {}),
        tasks=tasks,
    )
    # RunPolicy is INLINED on spec (torchjob_types.go:182), incl. the
    # reference's `clenPodPolicy` field spelling
    clean = spec.get("clenPodPolicy") or spec.get("cleanPodPolicy")
    job.run_policy = RunPolicy(
        clean_task_policy=CleanPodPolicy(clean) if clean
        else CleanPodPolicy.RUNNING,
        ttl_seconds_after_finished=spec.get("TTLSecondsAfterFinished"),
        active_deadline_seconds=spec.get("activeDurations"),
        backoff_limit=int(spec.get("backoffLimit", 3)))
    sp = spec.get("schedulingPolicy") or {}
    job.scheduling = SchedulingPolicy(
        min_available=sp.get("minAvailable"),
        queue=sp.get("queue", ""),
        priority=sp.get("priority"),
        priority_class_name=sp.get("priorityClassName", ""))
    if spec.get("minMembers"):
        job.min_members = {TaskType(t.lower()): int(v)
                           for t, v in spec["minMembers"].items()}
    mv = spec.get("modelVersion")
    if mv:
        job.model_name = mv.get("modelName") or mv.get("name")
    if spec.get("enableTorchElastic"):
        ep = spec.get("torchElasticPolicy") or {}
        job.elastic = ElasticPolicy(
            min_replicas=int(ep.get("numMinReplicas", 1)),
            max_replicas=int(ep.get("numMaxReplicas", 1)),
            nproc_per_node=int(ep.get("numWorkersPerNodePolicy", 1)),
            rdzv_backend=ep.get("rendezvousBackend", ""),
            rdzv_endpoint=ep.get("rendezvousEndpoint", ""))
    return set_defaults(job)


def job_to_crd_dict(job: TorchJob) -> dict:
    """Emit the reference CRD shape (amd.com/gpu resources) — the
    round-trip counterpart of _job_from_crd_dict."""
    task_specs = {}
    for t, s in job.tasks.items():
        res: dict = {}
        if s.gpus_per_task:
            res["amd.com/gpu"] = s.gpus_per_task
        if s.cpus_per_task:
            res["cpu"] = s.cpus_per_task
        if s.mem_mb_per_task:
            res["memory"] = f"{s.mem_mb_per_task}Mi"
        cont = {"name": "torch",
                "env": [{"name": k, "value": v} for k, v in s.env.items()]}
        if s.command:
            cont["command"] = list(s.command)
        if res:
            cont["resources"] = {"limits": res}
        d = {"numTasks": s.replicas,
             "template": {"spec": {"containers": [cont]}}}
        if s.termination_grace_seconds is not None:
            d["template"]["spec"]["terminationGracePeriodSeconds"] = \
                s.termination_grace_seconds
        if s.restart_policy is not None:
            inv = {v: k for k, v in _CRD_RESTART.items()}
            d["restartPolicy"] = inv.get(s.restart_policy,
                                         s.restart_policy.value)
        if s.spot is not None:
            d["spotTaskSpec"] = {"numSpotTasks": s.spot.num_spot_replicas,
                                 "labels": dict(s.spot.labels)}
            if s.spot.priority_class_name:
                d["spotTaskSpec"]["priorityClassName"] = \
                    s.spot.priority_class_name
        task_specs[t.value.capitalize() if t != TaskType.AIMASTER
                   else "AIMaster"] = d
    spec: dict = {"torchTaskSpecs": task_specs,
                  "backoffLimit": job.run_policy.backoff_limit,
                  "clenPodPolicy": job.run_policy.clean_task_policy.value}
    if job.run_policy.ttl_seconds_after_finished is not None:
        spec["TTLSecondsAfterFinished"] = \
            job.run_policy.ttl_seconds_after_finished
    if job.run_policy.active_deadline_seconds is not None:
        spec["activeDurations"] = job.run_policy.active_deadline_seconds
    if job.scheduling.min_available or job.scheduling.queue or \
            job.scheduling.priority or job.scheduling.priority_class_name:
        spec["schedulingPolicy"] = {
            k: v for k, v in [("minAvailable", job.scheduling.min_available),
                              ("queue", job.scheduling.queue or None),
                              ("priority", job.scheduling.priority),
                              ("priorityClassName",
                               job.scheduling.priority_class_name or None)]
            if v is not None}
    if job.min_members:
        spec["minMembers"] = {
            (t.value.capitalize() if t != TaskType.AIMASTER else "AIMaster"):
            v for t, v in job.min_members.items()}
    if job.model_name:
        spec["modelVersion"] = {"modelName": job.model_name}
    if job.elastic is not None:
        spec["enableTorchElastic"] = True
        spec["torchElasticPolicy"] = {
            "numMinReplicas": job.elastic.min_replicas,
            "numMaxReplicas": job.elastic.max_replicas,
            "numWorkersPerNodePolicy": job.elastic.nproc_per_node,
        }
        if job.elastic.rdzv_backend:
            spec["torchElasticPolicy"]["rendezvousBackend"] = \
                job.elastic.rdzv_backend
        if job.elastic.rdzv_endpoint:
            spec["torchElasticPolicy"]["rendezvousEndpoint"] = \
                job.elastic.rdzv_endpoint
    return {
        "apiVersion": "train.distributed.io/v1alpha1",
        "kind": "TorchJob",
        "metadata": {"name": job.name, "namespace": job.namespace,
                     "annotations": dict(job.annotations),
                     "labels": dict(job.labels)},
        "spec": spec,
    }


def job_from_dict(doc: dict) -> TorchJob:
    meta = doc.get("metadata") or {}
    spec = doc.get("spec") or {}
    if "torchTaskSpecs" in spec:  # reference CRD dialect
        return _job_from_crd_dict(doc)
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
        priority=sp.get("priority"),
        priority_class_name=sp.get("priorityClassName", ""))
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
            max_num_metrics=int(ep.get("maxNumMetrics", 50)),
            rdzv_backend=ep.get("rdzvBackend", ""),
            rdzv_endpoint=ep.get("rdzvEndpoint", ""))
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
        # TorchElasticStatus analog (torchjob_types.go:259-289)
        "elastic": None if job.status.elastic is None else {
            "currentReplicas": job.status.elastic.replicas,
            "lastReplicas": job.status.elastic.last_replicas,
            "continue": job.status.elastic.continue_training,
            "elasticCondition": job.status.elastic.condition.value,
        },
    }
