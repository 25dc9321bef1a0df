"""Gang scheduling on the node's GPU slots.

Reference: Volcano PodGroups (pkg/gangscheduler/volcano/volcano.go) with
per-role groups when DAG is on (:108-172) or one per-job group (:174-230),
MinMember semantics, and the README-fixed rule that a gang job becomes
Running at MinMember running tasks (README.md:28-30). Here a "PodGroup"
is a GPU-slot reservation on the node: the whole gang's minimum GPU
demand must be admissible before ANY task starts (no partial starts, no
deadlock between two half-admitted jobs).

The reference's bug of returning (nil, err) even on successful creation
(volcano.go:96-103) is not replicated.
"""
from __future__ import annotations

from dataclasses import dataclass, field

from torch_on_k8s_amd.controlplane.api import TaskType, TorchJob
from torch_on_k8s_amd.controlplane.node import NodeState


@dataclass
class PodGroup:
    name: str
    min_member: int          # tasks that must be schedulable together
    min_gpus: int            # GPU slots those tasks need
    per_role: dict = field(default_factory=dict)  # role -> (min_member, gpus)


class GangScheduler:
    """Per-job (or per-role when DAG scheduling is on) gang admission."""

    def __init__(self, node: NodeState, dag_scheduling: bool = True):
        self.node = node
        self.dag = dag_scheduling
        self.groups: dict[str, PodGroup] = {}

    def create_pod_group(self, job: TorchJob) -> PodGroup:
        if self.dag:
            # per-role groups, AIMaster skipped (volcano.go:108-172)
            per_role = {}
            for t, spec in job.tasks.items():
                if t == TaskType.AIMASTER:
                    continue
                mm = job.min_members.get(t, spec.replicas)
                mm = min(mm, spec.replicas)  # validated <= NumTasks
                per_role[t] = (mm, mm * spec.gpus_per_task)
            min_member = sum(m for m, _ in per_role.values())
            min_gpus = sum(g for _, g in per_role.values())
            pg = PodGroup(job.name, min_member, min_gpus, per_role)
        else:
            # one per-job group; MinAvailable override (volcano.go:174-230);
            # scale MinResources with the override (fixing the acknowledged
            # TODO at volcano.go:223-227)
            total = job.total_replicas(include_aimaster=False)
            gpus = job.total_gpus(include_aimaster=False)
            mm = job.scheduling.min_available or total
            mm = min(mm, total)
            min_gpus = (gpus * mm + total - 1) // total if total else 0
            pg = PodGroup(job.name, mm, min_gpus)
        self.groups[job.name] = pg
        return pg

    def can_admit(self, job: TorchJob) -> bool:
        pg = self.groups.get(job.name) or self.create_pod_group(job)
        return len(self.node.free_slots) >= pg.min_gpus

    def min_member_running(self, job: TorchJob, running_tasks: int) -> bool:
        """Gang jobs count as Running at MinMember running tasks, not all
        (reference README-fixed behavior)."""
        pg = self.groups.get(job.name)
        if pg is None:
            return False
        return running_tasks >= pg.min_member

    def delete_pod_group(self, job_name: str):
        self.groups.pop(job_name, None)
