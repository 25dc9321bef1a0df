"""Reference-CRD-dialect acceptance (r1 VERDICT missing #2): a TorchJob
manifest written against hliangzhao/torch-on-k8s
(apis/train/v1alpha1/torchjob_types.go:88-206 shape — spec.torchTaskSpecs
with pod templates, nvidia.com/gpu resources, inlined RunPolicy incl.
the `clenPodPolicy` spelling) must parse and run unchanged."""
from __future__ import annotations

import os

import yaml

from torch_on_k8s_amd.controlplane.api import (CleanPodPolicy, RestartPolicy,
                                               TaskType)
from torch_on_k8s_amd.controlplane.jobspec import (job_from_dict,
                                                   job_from_yaml,
                                                   job_to_crd_dict,
                                                   parse_quantity)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REFERENCE_STYLE = """
apiVersion: train.distributed.io/v1alpha1
kind: TorchJob
metadata:
  name: ref-job
  namespace: team-a
  annotations: {enable-elastic-training: "true"}
spec:
  backoffLimit: 5
  clenPodPolicy: Running
  TTLSecondsAfterFinished: 600
  activeDurations: 3600
  schedulingPolicy: {minAvailable: 3, queue: prod, priority: 7,
                     priorityClassName: high-priority}
  torchTaskSpecs:
    Master:
      numTasks: 1
      restartPolicy: ExitCode
      template:
        spec:
          containers:
            - name: torch
              command: ["python", "train.py"]
              env:
                - {name: EPOCHS, value: "3"}
              resources:
                limits: {nvidia.com/gpu: 1, cpu: "4", memory: 8Gi}
    Worker:
      numTasks: 2
      restartPolicy: OnFailure
      spotTaskSpec: {numSpotTasks: 1, labels: {tier: spot},
                     priorityClassName: spot-low}
      template:
        spec:
          containers:
            - name: torch
              resources:
                limits: {nvidia.com/gpu: 2, cpu: 500m, memory: 512Mi}
  minMembers: {Master: 1, Worker: 2}
  enableTorchElastic: true
  torchElasticPolicy:
    numMinReplicas: 2
    numMaxReplicas: 8
    numWorkersPerNodePolicy: 1
    rendezvousBackend: etcd
    rendezvousEndpoint: etcd-host:2379
  modelVersion: {modelName: my-model}
"""


def test_reference_manifest_parses():
    job = job_from_yaml(REFERENCE_STYLE)
    assert job.name == "ref-job"
    assert job.namespace == "team-a"
    assert job.annotations["enable-elastic-training"] == "true"
    m = job.tasks[TaskType.MASTER]
    assert m.replicas == 1
    assert m.restart_policy == RestartPolicy.ON_EXIT_CODE
    assert m.command == ["python", "train.py"]
    assert m.env["EPOCHS"] == "3"
    # nvidia.com/gpu maps 1:1 onto the node's MI355X GPUs
    assert m.gpus_per_task == 1
    assert m.cpus_per_task == 4.0
    assert m.mem_mb_per_task == 8192
    w = job.tasks[TaskType.WORKER]
    assert w.replicas == 2
    assert w.restart_policy == RestartPolicy.ON_FAILURE
    assert w.gpus_per_task == 2
    assert w.cpus_per_task == 0.5
    assert w.mem_mb_per_task == 512
    assert w.spot is not None and w.spot.num_spot_replicas == 1
    # inlined RunPolicy fields
    assert job.run_policy.backoff_limit == 5
    assert job.run_policy.clean_task_policy == CleanPodPolicy.RUNNING
    assert job.run_policy.ttl_seconds_after_finished == 600
    assert job.run_policy.active_deadline_seconds == 3600
    assert job.scheduling.min_available == 3
    assert job.scheduling.queue == "prod"
    assert job.scheduling.priority == 7
    assert job.scheduling.priority_class_name == "high-priority"
    assert w.spot.priority_class_name == "spot-low"
    assert job.elastic.rdzv_backend == "etcd"
    assert job.elastic.rdzv_endpoint == "etcd-host:2379"
    assert job.min_members[TaskType.MASTER] == 1
    assert job.elastic is not None
    assert job.elastic.min_replicas == 2 and job.elastic.max_replicas == 8
    assert job.model_name == "my-model"
    # defaults applied on top (DAG edges)
    assert w.dag_conditions and \
        w.dag_conditions[0].upstream == TaskType.MASTER


def test_crd_sample_file_parses():
    with open(os.path.join(REPO, "configs", "samples",
                           "llama8b-dp8-crd.yaml")) as f:
        job = job_from_yaml(f.read())
    assert job.tasks[TaskType.WORKER].replicas == 7
    assert job.tasks[TaskType.MASTER].gpus_per_task == 1
    assert job.total_gpus() == 8
    assert job.scheduling.min_available == 8


def test_round_trip_crd_dialect():
    """native -> CRD dict -> native must preserve semantics."""
    job1 = job_from_yaml(REFERENCE_STYLE)
    doc = job_to_crd_dict(job1)
    job2 = job_from_dict(doc)
    assert job2.name == job1.name
    for t in job1.tasks:
        a, b = job1.tasks[t], job2.tasks[t]
        assert a.replicas == b.replicas
        assert a.gpus_per_task == b.gpus_per_task
        assert a.cpus_per_task == b.cpus_per_task
        assert a.mem_mb_per_task == b.mem_mb_per_task
        assert a.restart_policy == b.restart_policy
        assert a.env == b.env
    assert job2.run_policy.backoff_limit == job1.run_policy.backoff_limit
    assert job2.scheduling.queue == job1.scheduling.queue
    assert (job2.elastic.min_replicas, job2.elastic.max_replicas) == \
        (job1.elastic.min_replicas, job1.elastic.max_replicas)
    assert (job2.elastic.rdzv_backend, job2.elastic.rdzv_endpoint) == \
        ("etcd", "etcd-host:2379")
    assert job2.scheduling.priority_class_name == "high-priority"
    assert job2.tasks[TaskType.WORKER].spot.priority_class_name == "spot-low"
    assert job2.model_name == job1.model_name
    # emitted dialect uses the node's GPU resource name
    res = doc["spec"]["torchTaskSpecs"]["Master"]["template"]["spec"][
        "containers"][0]["resources"]["limits"]
    assert "amd.com/gpu" in res


def test_native_dialect_round_trip_via_crd():
    """A native-dialect job survives export to CRD shape and re-import."""
    with open(os.path.join(REPO, "configs", "samples",
                           "llama8b-dp8.yaml")) as f:
        job1 = job_from_yaml(f.read())
    job2 = job_from_dict(job_to_crd_dict(job1))
    assert {t: s.replicas for t, s in job2.tasks.items()} == \
        {t: s.replicas for t, s in job1.tasks.items()}
    assert job2.total_gpus() == job1.total_gpus()


def test_parse_quantity():
    assert parse_quantity("4") == 4.0
    assert parse_quantity("500m") == 0.5
    assert parse_quantity("8Gi") == 8 * (1 << 30)
    assert parse_quantity("512Mi") == 512 * (1 << 20)
    assert parse_quantity("2G") == 2e9
    assert parse_quantity(3) == 3.0


def test_crd_job_runs_through_manager(tmp_path):
    """End-to-end: a reference-dialect manifest submitted to the manager
    spool runs to completion (the r1 gap: 'a job written against the
    reference CRD does not parse')."""
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane.api import JobConditionType

    mgr = Manager(str(tmp_path), num_gpus=2, sync_period=0.05)
    doc = {
        "apiVersion": "train.distributed.io/v1alpha1",
        "kind": "TorchJob",
        "metadata": {"name": "crd-e2e"},
        "spec": {
            "torchTaskSpecs": {
                "Master": {
                    "numTasks": 1,
                    "restartPolicy": "ExitCode",
                    "template": {"spec": {"containers": [{
                        "name": "torch",
                        "env": [
                            {"name": "TOK_BACKEND", "value": "gloo"},
                            {"name": "TOK_TRAIN_STEPS", "value": "2"},
                            {"name": "TOK_TRAINER_CONFIG",
                             "value": '{"model": "llama-tiny", '
                                      '"micro_batch": 1, "seq_len": 32}'},
                        ],
                        "resources": {"limits": {"nvidia.com/gpu": 1}},
                    }]}},
                },
            },
        },
    }
    with open(os.path.join(mgr.spool, "crd-e2e.yaml"), "w") as f:
        yaml.safe_dump(doc, f)
    import time
    deadline = time.time() + 240
    job = None
    while time.time() < deadline:
        mgr.step()
        job = mgr.controller.jobs.get("crd-e2e")
        if job is not None and job.status.phase in (
                JobConditionType.SUCCEEDED, JobConditionType.FAILED):
            break
        time.sleep(0.05)
    assert job is not None
    assert job.status.phase == JobConditionType.SUCCEEDED, \
        (job.status.phase, mgr.controller.events_for("crd-e2e"))


def test_rdzv_env_exported_for_custom_elastic_jobs():
    """TorchElasticPolicy rendezvous fields surface in the task env
    (reference emits torchrun --rdzv_* args, torchjob_controller.go:385-392;
    here custom-command jobs read TOK_RDZV_BACKEND/ENDPOINT)."""
    from torch_on_k8s_amd.controlplane.runtime import cluster_env

    job = job_from_yaml(REFERENCE_STYLE)
    env = cluster_env(job, TaskType.WORKER, 0)
    assert env["TOK_RDZV_BACKEND"] == "etcd"
    assert env["TOK_RDZV_ENDPOINT"] == "etcd-host:2379"
    # endpoint defaults to the master address when unset
    job.elastic.rdzv_endpoint = ""
    env = cluster_env(job, TaskType.WORKER, 0, master_port=29400)
    assert env["TOK_RDZV_ENDPOINT"] == "127.0.0.1:29400"
    # non-elastic jobs carry no rendezvous keys
    job.elastic = None
    assert "TOK_RDZV_BACKEND" not in cluster_env(job, TaskType.WORKER, 0)


def test_emitted_crd_fields_declared_in_schema():
    """Schema/parser drift check: every key job_to_crd_dict emits for a
    maximal job must appear somewhere in the CRD schema document
    (configs/crd/torchjobs...yaml), so a manifest round-tripped through
    the exporter always validates against the published schema."""
    from torch_on_k8s_amd.controlplane.api import (ElasticPolicy,
                                                   SchedulingPolicy,
                                                   SpotTaskSpec, TaskSpec,
                                                   TorchJob, set_defaults)

    job = set_defaults(TorchJob(
        name="maximal",
        tasks={TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=1,
                                         cpus_per_task=2.0,
                                         mem_mb_per_task=1024,
                                         command=["python", "x.py"]),
               TaskType.WORKER: TaskSpec(
                   replicas=2, gpus_per_task=1,
                   spot=SpotTaskSpec(num_spot_replicas=1,
                                     priority_class_name="spot"))},
        scheduling=SchedulingPolicy(min_available=3, queue="q", priority=5,
                                    priority_class_name="gold"),
        elastic=ElasticPolicy(min_replicas=1, max_replicas=2,
                              rdzv_backend="etcd",
                              rdzv_endpoint="e:2379"),
        model_name="m"))
    job.run_policy.ttl_seconds_after_finished = 60
    job.run_policy.active_deadline_seconds = 600
    doc = job_to_crd_dict(job)

    def all_keys(d):
        out = set()
        if isinstance(d, dict):
            for k, v in d.items():
                out.add(k)
                out |= all_keys(v)
        elif isinstance(d, list):
            for v in d:
                out |= all_keys(v)
        return out

    with open(os.path.join(REPO, "configs", "crd",
                           "torchjobs.train.distributed.io.yaml")) as f:
        schema_text = f.read()
    emitted = all_keys(doc["spec"]) | all_keys(doc["metadata"])
    # not schema keys: task-type map keys, resource names, standard
    # ObjectMeta, and the pod-template subtree (declared open via
    # x-kubernetes-preserve-unknown-fields — the hand-maintained
    # equivalent of the reference's 7k-line expanded PodTemplateSpec)
    assert "x-kubernetes-preserve-unknown-fields: true" in schema_text
    skip = {"Master", "Worker", "AIMaster", "amd.com/gpu", "cpu", "memory",
            "name", "value", "annotations", "labels", "namespace",
            "containers", "limits", "spec"}
    missing = sorted(k for k in emitted - skip
                     if f"{k}:" not in schema_text and
                     f"{k} " not in schema_text)
    assert not missing, f"emitted keys absent from CRD schema: {missing}"


def test_production_knobs_sample_parses():
    """The kitchen-sink sample exercises every late-round policy field
    in one manifest and must round-trip cleanly."""
    with open(os.path.join(REPO, "configs", "samples",
                           "production-knobs.yaml")) as f:
        job = job_from_yaml(f.read())
    assert job.scheduling.priority_class_name == "gold"
    assert job.elastic.rdzv_backend == "tcpstore"
    assert job.tasks[TaskType.MASTER].termination_grace_seconds == 90
    w = job.tasks[TaskType.WORKER]
    assert w.spot.priority_class_name == "spot-low"
    assert w.mem_mb_per_task == 65536
    job2 = job_from_dict(job_to_crd_dict(job))
    assert job2.tasks[TaskType.WORKER].termination_grace_seconds == 90
    assert job2.scheduling.priority_class_name == "gold"


def test_all_sample_manifests_parse():
    """Every file under configs/samples/ must parse in whichever dialect
    it is written (docs stay executable)."""
    d = os.path.join(REPO, "configs", "samples")
    parsed = 0
    for f in sorted(os.listdir(d)):
        if not f.endswith((".yaml", ".yml")):
            continue
        with open(os.path.join(d, f)) as fh:
            job = job_from_yaml(fh.read())
        assert job.tasks, f
        parsed += 1
    assert parsed >= 6  # grows with new samples
