"""Client SDK + spot preemption tests."""
import pytest

from torch_on_k8s_amd.client import TorchJobClient
from torch_on_k8s_amd.controlplane.api import (JobConditionType,
                                               SchedulingPolicy, SpotTaskSpec,
                                               TaskSpec, TaskType, TorchJob,
                                               TaskPhase)
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import FakeRuntime


def test_client_apply_get_delete(tmp_path):
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane import features as feat
    mgr = Manager(str(tmp_path), num_gpus=4,
                  gates=feat.FeatureGates({"JobCoordinator": False}))
    cli = TorchJobClient(str(tmp_path))
    name = cli.apply({
        "kind": "TorchJob",
        "metadata": {"name": "cli-job"},
        "spec": {"tasks": {"worker": {"replicas": 2, "gpusPerTask": 1}}},
    })
    assert name == "cli-job"
    mgr.step()
    mgr.step()
    st = cli.get("cli-job")
    assert st is not None and st["phase"] in ("Created", "Running")
    assert "cli-job" in cli.list()
    cli.delete("cli-job")
    mgr.step()
    assert mgr.controller.jobs.get("cli-job") is None


def test_spot_preemption_frees_gpus():
    node = NodeState(num_gpus=8)
    rt = FakeRuntime()
    ctl = JobController(node, rt, ControllerConfig())
    # low-priority job with 4 workers, 2 of them spot
    lo = TorchJob(name="lo", tasks={
        TaskType.WORKER: TaskSpec(replicas=4, gpus_per_task=2,
                                  spot=SpotTaskSpec(num_spot_replicas=2)),
    }, scheduling=SchedulingPolicy(priority=1))
    ctl.create_job(lo)
    ctl.reconcile(lo)
    assert len(node.free_slots) == 0
    spot_handles = [h for h in ctl.handles["lo"].values() if h.spot]
    assert len(spot_handles) == 2

    # high-priority job needs 4 GPUs -> spot tasks get preempted
    hipri = TorchJob(name="hi", tasks={
        TaskType.WORKER: TaskSpec(replicas=4, gpus_per_task=1),
    }, scheduling=SchedulingPolicy(priority=9))
    ctl.create_job(hipri)
    # FakeRuntime.kill marks the victims failed (exit 143); lo reconcile
    # releases their GPUs, then hi admits
    ctl.reconcile(lo)
    ctl.reconcile(hipri)
    assert any(e.reason == "SpotPreempted" for e in ctl.events)
    assert len(ctl.handles["hi"]) == 4
    assert hipri.status.phase == JobConditionType.RUNNING
