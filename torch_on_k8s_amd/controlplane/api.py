"""TorchJob API types — node-native redesign of the reference CRDs.

The reference defines a k8s CRD (`apis/train/v1alpha1/torchjob_types.go`);
this framework targets one 8xMI355X node, so a "task" wraps one local GPU
process instead of a pod, and the API object is a plain dataclass tree
with the same semantics:

  * task types AIMaster/Master/Worker (torchjob_types.go:36-42)
  * RestartPolicy incl. OnExitCode (torchjob_types.go:66-74)
  * DAGConditions gating task start order (torchjob_types.go:79-84,
    defaults AIMaster -> Master -> Worker, torchjob_defaults.go:95-124)
  * SchedulingPolicy: MinAvailable/Queue/Priority (torchjob_types.go:120-135)
  * RunPolicy: CleanPodPolicy/TTL/ActiveDeadline/BackoffLimit
    (torchjob_types.go:139-154)
  * TorchElasticPolicy: min/max replicas, nproc, metric window
    (torchjob_types.go:160-173)
  * Job conditions Created/Queuing/Running/Restarting/Succeeded/Failed
    (torchjob_types.go:214-239)
  * annotations as the dynamic channel (constants.go:62-78) — used by the
    elastic checkpoint protocol (elastic_scale.go:49-56)

Reference bugs deliberately NOT replicated: MinMembers
defaulting reads the task specs (not the nil MinMembers map).
"""
from __future__ import annotations

import enum
import itertools
import re
import time
from dataclasses import dataclass, field


class TaskType(str, enum.Enum):
    AIMASTER = "aimaster"
    MASTER = "master"
    WORKER = "worker"


# Reconcile/start order: AIMaster -> Master -> Worker
# (reference torchjob_controller.go:464-471)
TASK_ORDER = [TaskType.AIMASTER, TaskType.MASTER, TaskType.WORKER]


class RestartPolicy(str, enum.Enum):
    NEVER = "Never"
    ON_FAILURE = "OnFailure"
    ALWAYS = "Always"
    ON_EXIT_CODE = "OnExitCode"   # controller-owned restarts by exit code


class TaskPhase(str, enum.Enum):
    PENDING = "Pending"
    RUNNING = "Running"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"


# phase ordering for DAG gating (reference dag.go:111-116)
_PHASE_ORDER = {
    TaskPhase.PENDING: 1,
    TaskPhase.RUNNING: 2,
    TaskPhase.SUCCEEDED: 3,
    TaskPhase.FAILED: 3,
}


def phase_reached(actual: TaskPhase, wanted: TaskPhase) -> bool:
    return _PHASE_ORDER[actual] >= _PHASE_ORDER[wanted]


class JobConditionType(str, enum.Enum):
    CREATED = "Created"
    QUEUING = "Queuing"
    RUNNING = "Running"
    RESTARTING = "Restarting"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"


class CleanPodPolicy(str, enum.Enum):
    RUNNING = "Running"   # stop still-running tasks on job end (default)
    ALL = "All"
    NONE = "None"


@dataclass
class DAGCondition:
    upstream: TaskType
    on_phase: TaskPhase = TaskPhase.RUNNING


@dataclass
class SpotTaskSpec:
    """Lower-priority overflow replicas (torchjob_types.go:50-61)."""
    num_spot_replicas: int = 0
    priority: int = -10
    # reference SpotTaskSpec.PriorityClassName (torchjob_types.go:56-58):
    # resolved against the manager's priority-class table when set
    priority_class_name: str = ""
    labels: dict = field(default_factory=dict)


@dataclass
class TaskSpec:
    replicas: int = 1
    restart_policy: RestartPolicy | None = None
    gpus_per_task: int = 1
    # full resource requests (reference quota filtering covers every
    # resource in the namespace ResourceQuotas, plugins/quota.go:97-131)
    cpus_per_task: float = 0.0
    mem_mb_per_task: int = 0
    dag_conditions: list[DAGCondition] = field(default_factory=list)
    spot: SpotTaskSpec | None = None
    # entrypoint override: argv list; None -> framework training entrypoint
    command: list[str] | None = None
    env: dict = field(default_factory=dict)
    # pod terminationGracePeriodSeconds analog: per-task override of the
    # manager's kill_grace_seconds (SIGTERM -> SIGKILL window)
    termination_grace_seconds: float | None = None


@dataclass
class SchedulingPolicy:
    min_available: int | None = None
    queue: str = ""
    priority: int | None = None
    # torchjob_types.go:127-130: used when `priority` is unset; the
    # coordinator's PriorityPlugin resolves it against the manager's
    # priority-class table (the PriorityClass-object analog)
    priority_class_name: str = ""


@dataclass
class RunPolicy:
    # DELIBERATE divergence from the reference default of None
    # (torchjob_defaults.go:30-33): on k8s, pods left behind only hold
    # kubelet bookkeeping, but here a straggler process holds REAL GPU
    # slots on the one node until TTL/delete. Running (kill stragglers
    # on completion) is the safe single-node default; set
    # cleanTaskPolicy/clenPodPolicy: None explicitly for reference
    # semantics (both honored, tests cover both).
    clean_task_policy: CleanPodPolicy = CleanPodPolicy.RUNNING
    ttl_seconds_after_finished: float | None = None
    active_deadline_seconds: float | None = None
    backoff_limit: int = 3


@dataclass
class ElasticPolicy:
    min_replicas: int = 1
    max_replicas: int = 1
    nproc_per_node: int = 1
    metric_window: int = 5     # observations before a scale decision
    max_num_metrics: int = 50  # stop autoscaling after this many samples
    # reference TorchElasticPolicy rendezvous fields
    # (torchjob_types.go:168-171, emitted as torchrun --rdzv_* args at
    # torchjob_controller.go:385-392). The in-tree entrypoint rendezvous
    # is the fast-rejoin TCPStore; these are exported to the task env as
    # TOK_RDZV_BACKEND/TOK_RDZV_ENDPOINT for custom-command jobs that
    # run torchrun themselves.
    rdzv_backend: str = ""
    rdzv_endpoint: str = ""


@dataclass
class JobCondition:
    type: JobConditionType
    reason: str = ""
    message: str = ""
    ts: float = field(default_factory=time.time)


@dataclass
class TaskStatus:
    active: int = 0
    succeeded: int = 0
    failed: int = 0


class ElasticCondition(str, enum.Enum):
    START = "Start"
    STOP = "Stop"
    CONTINUE = "ContinueTraining"
    MAX_METRIC = "ReachMaxMetric"
    MAX_REPLICAS = "ReachMaxReplicas"


@dataclass
class ElasticStatus:
    """Per-replica-count observations (reference TorchElasticStatus,
    torchjob_types.go:259-289)."""
    replicas: int = 0
    last_replicas: int = 0
    continue_training: bool = True
    condition: ElasticCondition = ElasticCondition.START
    observations: dict = field(default_factory=dict)  # replicas -> [metric]


@dataclass
class JobStatus:
    conditions: list[JobCondition] = field(default_factory=list)
    tasks: dict = field(default_factory=dict)  # TaskType -> TaskStatus
    start_time: float | None = None
    completion_time: float | None = None
    restart_count: int = 0
    model_version: str | None = None
    elastic: ElasticStatus | None = None

    @property
    def phase(self) -> JobConditionType | None:
        return self.conditions[-1].type if self.conditions else None

    def set_condition(self, t: JobConditionType, reason: str = "",
                      message: str = ""):
        if self.conditions and self.conditions[-1].type == t:
            return
        self.conditions.append(JobCondition(t, reason, message))

    def has_condition(self, t: JobConditionType) -> bool:
        return any(c.type == t for c in self.conditions)


_uid = itertools.count(1)


@dataclass
class TorchJob:
    name: str
    tasks: dict = field(default_factory=dict)  # TaskType -> TaskSpec
    run_policy: RunPolicy = field(default_factory=RunPolicy)
    scheduling: SchedulingPolicy = field(default_factory=SchedulingPolicy)
    elastic: ElasticPolicy | None = None
    min_members: dict = field(default_factory=dict)  # TaskType -> int
    namespace: str = "default"
    annotations: dict = field(default_factory=dict)
    labels: dict = field(default_factory=dict)
    # model packaging request: output dir of the master becomes a
    # ModelVersion on success (reference job.go:462-508)
    model_name: str | None = None
    status: JobStatus = field(default_factory=JobStatus)
    generation: int = 1
    uid: int = field(default_factory=lambda: next(_uid))
    deleted: bool = False

    def total_replicas(self, include_aimaster: bool = True) -> int:
        return sum(s.replicas for t, s in self.tasks.items()
                   if include_aimaster or t != TaskType.AIMASTER)

    def total_gpus(self, include_aimaster: bool = True) -> int:
        return sum(s.replicas * s.gpus_per_task for t, s in self.tasks.items()
                   if include_aimaster or t != TaskType.AIMASTER)

    def total_resources(self) -> dict:
        """Aggregate resource request (the quota plugin's unit; reference
        computes this over pod template resources, resources.go:28-109)."""
        return {
            "gpu": sum(s.replicas * s.gpus_per_task
                       for s in self.tasks.values()),
            "cpu": sum(s.replicas * s.cpus_per_task
                       for s in self.tasks.values()),
            "memory_mb": sum(s.replicas * s.mem_mb_per_task
                             for s in self.tasks.values()),
        }


# Annotation keys (reference constants.go:62-78, elastic_scale.go:49-56)
ANN_CKPT_REQUESTED = "ckpt-requested-version"
ANN_CKPT_COMPLETED = "ckpt-completed-version"
ANN_READY_TO_START_WORKER = "ready-to-start-worker"
ANN_SCALE_STATE = "scale-state"          # inflight | done
ANN_WORLD_SIZE = "world-size"
ANN_ENABLE_ELASTIC = "enable-elastic-training"

CKPT_IN_PROGRESS = "InProgress"
CKPT_SUCCEEDED = "Succeeded"

DEFAULT_MASTER_PORT = 23456  # reference constants.go:96-103


def set_defaults(job: TorchJob) -> TorchJob:
    """SetDefaults_TorchJob parity (torchjob_defaults.go:29-74)."""
    job.name = job.name.lower().replace("_", "-")
    # k8s enforces DNS-1123 object names before any controller runs;
    # here job.name becomes filesystem paths (spool/status/jobs dirs),
    # so an unvalidated name like '../x' would escape the workdir
    if not re.fullmatch(r"[a-z0-9]([-a-z0-9.]{0,251}[a-z0-9])?", job.name):
        raise ValueError(f"invalid job name {job.name!r}: must match "
                         f"DNS-1123 (lowercase alphanumerics, '-', '.')")
    if not job.tasks:
        job.tasks = {TaskType.MASTER: TaskSpec()}
    for t, spec in job.tasks.items():
        if spec.replicas is None or spec.replicas < 0:
            spec.replicas = 1
        if spec.restart_policy is None:
            # master OnExitCode, workers OnFailure (constants.go:106-109)
            spec.restart_policy = (RestartPolicy.ON_EXIT_CODE
                                   if t == TaskType.MASTER
                                   else RestartPolicy.ON_FAILURE)
    if TaskType.MASTER in job.tasks and job.tasks[TaskType.MASTER].replicas != 1:
        job.tasks[TaskType.MASTER].replicas = 1
    # default DAG edges AIMaster -> Master -> Worker
    # (torchjob_defaults.go:95-124)
    has_aim = TaskType.AIMASTER in job.tasks
    has_master = TaskType.MASTER in job.tasks
    for t, spec in job.tasks.items():
        if spec.dag_conditions:
            continue
        if t == TaskType.MASTER and has_aim:
            spec.dag_conditions = [DAGCondition(TaskType.AIMASTER)]
        elif t == TaskType.WORKER:
            if has_master:
                spec.dag_conditions = [DAGCondition(TaskType.MASTER)]
            elif has_aim:
                spec.dag_conditions = [DAGCondition(TaskType.AIMASTER)]
    # MinMembers default := replicas per task (fixes reference bug where
    # the defaulting loop iterated the nil MinMembers map,
    # torchjob_defaults.go:192-197 / SURVEY.md §2.3)
    if not job.min_members:
        job.min_members = {t: s.replicas for t, s in job.tasks.items()}
    # spec sanity (surfaced as BadJobSpec by the manager): a negative
    # MinAvailable would make the gang trivially satisfied and the
    # Running-at-MinMember rule fire with zero tasks running; inverted
    # elastic bounds would make the autoscaler oscillate
    if job.scheduling.min_available is not None and \
            job.scheduling.min_available < 0:
        raise ValueError(
            f"negative minAvailable: {job.scheduling.min_available}")
    if job.run_policy.backoff_limit < 0:
        raise ValueError(
            f"negative backoffLimit: {job.run_policy.backoff_limit}")
    if job.elastic is not None and \
            job.elastic.min_replicas > job.elastic.max_replicas:
        raise ValueError(
            f"elastic minReplicas {job.elastic.min_replicas} > "
            f"maxReplicas {job.elastic.max_replicas}")
    return job
