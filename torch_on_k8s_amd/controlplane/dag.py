"""DAG gating of task start order (reference controllers/common/dag.go):
a task's processes are not created until every upstream task has its full
replica count at (or past) the required phase."""
from __future__ import annotations

from torch_on_k8s_amd.controlplane.api import TorchJob, phase_reached


def dag_condition_ready(job: TorchJob, task_type, task_handles: dict) -> bool:
    """task_handles: {(job, type, index) -> TaskHandle} for this job.
    Parity with CheckDAGConditionReady (dag.go:30-54): count upstream
    handles and require phase >= wanted for all of them."""
    spec = job.tasks[task_type]
    for cond in spec.dag_conditions:
        upstream_spec = job.tasks.get(cond.upstream)
        if upstream_spec is None:
            continue  # upstream task type absent -> edge vacuously true
        ups = [h for h in task_handles.values()
               if h.task_type == cond.upstream]
        if len(ups) < upstream_spec.replicas:
            return False
        if not all(phase_reached(h.phase, cond.on_phase) for h in ups):
            return False
    return True
