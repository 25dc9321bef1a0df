"""Small-unit coverage: metrics recorder, failover signal codes, client
timeout, jobspec round-trip."""
import pytest

from torch_on_k8s_amd.client import TorchJobClient
from torch_on_k8s_amd.controlplane import failover as fo
from torch_on_k8s_amd.controlplane.jobspec import job_from_dict
from torch_on_k8s_amd.controlplane.metrics import JobMetrics
from torch_on_k8s_amd.controlplane.api import TaskType


def test_metrics_recorder_smoke():
    m = JobMetrics()
    m.job_created_at("j")
    m.created()
    m.succeeded()
    m.failed()
    m.restarted()
    m.deleted()
    m.first_task_delay("j")
    m.all_tasks_delay("j")
    m.set_queue_depth("q", 3)  # no raise = pass (prometheus wiring)


def test_failover_negative_signal_codes():
    # python subprocess reports signal deaths as negative returncodes
    assert fo.exit_code_retryable(-15)    # SIGTERM
    assert fo.exit_code_retryable(-9)     # SIGKILL
    assert not fo.exit_code_retryable(-11)  # SIGSEGV -> 139 permanent
    # 128+signal convention
    assert fo.exit_code_retryable(152)    # 128+24 unclassified signal


def test_client_wait_timeout(tmp_path):
    cli = TorchJobClient(str(tmp_path))
    with pytest.raises(TimeoutError):
        cli.wait("nope", timeout=0.3, poll=0.1)


def test_jobspec_minimal_defaults():
    job = job_from_dict({"metadata": {"name": "min"}, "spec": {}})
    # empty spec -> defaulted single master (set_defaults parity)
    assert TaskType.MASTER in job.tasks
    assert job.tasks[TaskType.MASTER].replicas == 1
    assert job.run_policy.backoff_limit == 3
