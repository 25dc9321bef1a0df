"""Manager daemon tests: spool-dir job submission, status publication,
feature gates, YAML spec parsing."""
import json
import os
import time

import pytest

from torch_on_k8s_amd.controlplane import features as feat
from torch_on_k8s_amd.controlplane.api import (JobConditionType, TaskType,
                                               RestartPolicy)
from torch_on_k8s_amd.controlplane.jobspec import job_from_yaml
from torch_on_k8s_amd.manager import Manager

JOB_YAML = """
apiVersion: train.distributed.io/v1alpha1
kind: TorchJob
metadata:
  name: spool-job
  namespace: default
spec:
  runPolicy: {backoffLimit: 2}
  tasks:
    master:
      replicas: 1
      gpusPerTask: 0
      env:
        TOK_BACKEND: gloo
        TOK_TRAIN_STEPS: "2"
        TOK_TRAINER_CONFIG: '{"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}'
"""


def test_jobspec_yaml_parse():
    job = job_from_yaml(JOB_YAML)
    assert job.name == "spool-job"
    assert job.tasks[TaskType.MASTER].replicas == 1
    assert job.tasks[TaskType.MASTER].restart_policy == RestartPolicy.ON_EXIT_CODE
    assert job.run_policy.backoff_limit == 2


def test_feature_gates_flag():
    g = feat.FeatureGates.from_flag("GangScheduling=false,JobCoordinator=true")
    assert not g.enabled(feat.GANG_SCHEDULING)
    assert g.enabled(feat.JOB_COORDINATOR)
    with pytest.raises(ValueError):
        feat.FeatureGates({"NoSuchGate": True})


@pytest.mark.timeout(300)
def test_manager_spool_to_success(tmp_path):
    mgr = Manager(str(tmp_path), num_gpus=0,
                  gates=feat.FeatureGates({"GangScheduling": False,
                                           "JobCoordinator": False}))
    # add repo root to child env via spec env (manager test runs in-repo)
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    yaml_with_path = JOB_YAML.replace(
        "TOK_BACKEND: gloo", f"TOK_BACKEND: gloo\n        PYTHONPATH: {root}")
    with open(os.path.join(mgr.spool, "job.yaml"), "w") as f:
        f.write(yaml_with_path)

    t0 = time.time()
    job = None
    while time.time() - t0 < 120:
        mgr.step()
        job = mgr.controller.jobs.get("spool-job")
        if job and job.status.phase in (JobConditionType.SUCCEEDED,
                                        JobConditionType.FAILED):
            break
        time.sleep(0.2)
    assert job is not None
    if job.status.phase != JobConditionType.SUCCEEDED:
        logdir = tmp_path / "jobs" / "spool-job"
        logs = "\n".join(f"{p.name}: {p.read_text()[-800:]}"
                         for p in logdir.glob("*.log"))
        raise AssertionError(f"phase={job.status.phase}: {logs}")
    # status published
    st = json.load(open(tmp_path / "status" / "spool-job.json"))
    assert st["phase"] == "Succeeded"
    # spool removal deletes the job
    os.unlink(os.path.join(mgr.spool, "job.yaml"))
    mgr.step()
    assert "spool-job" not in mgr.controller.jobs


def test_bad_spec_surfaces_event_and_skips(tmp_path):
    """Malformed spool YAML must not crash the manager loop: it raises
    at parse, is surfaced as a BadJobSpec event, and later fixes to the
    file are picked up."""
    from torch_on_k8s_amd.manager import Manager
    mgr = Manager(str(tmp_path), num_gpus=0, sync_period=0.05)
    bad = tmp_path / "spool" / "j.yaml"
    bad.write_text("kind: TorchJob\nmetadata: {name: j}\n"
                   "spec:\n  tasks:\n    wizard: {replicas: 1}\n")
    mgr.step()
    assert "j" not in mgr.controller.jobs
    assert any(e.reason == "BadJobSpec" for e in mgr.controller.events)
    # fix the file (mtime changes) -> job admitted
    import time as _t
    _t.sleep(0.02)
    bad.write_text("kind: TorchJob\nmetadata: {name: j}\n"
                   "spec:\n  tasks:\n    master: {replicas: 1, "
                   "gpusPerTask: 0, command: ['true']}\n")
    mgr.step()
    assert "j" in mgr.controller.jobs


def test_duplicate_spool_file_cannot_hijack_or_delete(tmp_path):
    """Two spool files declaring the same job name: the second is
    ignored with a DuplicateJobName warning, and REMOVING it must not
    delete the first file's running job (k8s name-uniqueness analog)."""
    import os
    import yaml
    from torch_on_k8s_amd.manager import Manager

    mgr = Manager(str(tmp_path), num_gpus=1, sync_period=0.05)
    doc = {"kind": "TorchJob", "metadata": {"name": "uniq"},
           "spec": {"tasks": {"master": {
               "replicas": 1, "gpusPerTask": 0, "command": ["sleep", "30"]}}}}
    with open(os.path.join(mgr.spool, "a.yaml"), "w") as f:
        yaml.safe_dump(doc, f)
    mgr.step()
    assert "uniq" in mgr.controller.jobs
    with open(os.path.join(mgr.spool, "b.yaml"), "w") as f:
        yaml.safe_dump(doc, f)
    mgr.step()
    assert any(e.reason == "DuplicateJobName"
               for e in mgr.controller.events_for("uniq"))
    os.unlink(os.path.join(mgr.spool, "b.yaml"))
    mgr.step()
    assert "uniq" in mgr.controller.jobs, \
        "removing the duplicate file deleted the original job"
    # removing the OWNING file still deletes
    os.unlink(os.path.join(mgr.spool, "a.yaml"))
    for _ in range(3):
        mgr.step()
    assert "uniq" not in mgr.controller.jobs


def test_resubmitted_job_name_starts_fresh(tmp_path):
    """Deleting a job and resubmitting the SAME name must train from
    scratch, not silently resume the old incarnation's checkpoint
    (TOK_TRAIN_STEPS already 'reached' meant instant Succeeded with
    zero training)."""
    import json
    import os
    import time
    import yaml
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane.api import JobConditionType

    mgr = Manager(str(tmp_path), num_gpus=0, sync_period=0.05)
    env = {"TOK_BACKEND": "gloo", "TOK_TRAIN_STEPS": "2",
           "TOK_TRAINER_CONFIG": json.dumps(
               {"model": "llama-tiny", "micro_batch": 1, "seq_len": 32})}
    doc = {"kind": "TorchJob", "metadata": {"name": "fresh"},
           "spec": {"tasks": {"master": {"replicas": 1, "gpusPerTask": 0,
                                         "env": env}}}}

    def run_once():
        with open(os.path.join(mgr.spool, "fresh.yaml"), "w") as f:
            yaml.safe_dump(doc, f)
        deadline = time.time() + 180
        while time.time() < deadline:
            mgr.step()
            job = mgr.controller.jobs.get("fresh")
            if job is not None and job.status.phase in (
                    JobConditionType.SUCCEEDED, JobConditionType.FAILED):
                return job
            time.sleep(0.05)
        raise AssertionError("job never finished")

    job1 = run_once()
    assert job1.status.phase == JobConditionType.SUCCEEDED
    # plant a fake old checkpoint claiming step 2 (== TOK_TRAIN_STEPS)
    ck = tmp_path / "jobs" / "fresh" / "ckpt"
    ck.mkdir(parents=True, exist_ok=True)
    (ck / "meta.json").write_text('{"step": 2}')
    os.unlink(os.path.join(mgr.spool, "fresh.yaml"))
    for _ in range(5):
        mgr.step()
        time.sleep(0.05)
    assert "fresh" not in mgr.controller.jobs
    job2 = run_once()
    assert job2.status.phase == JobConditionType.SUCCEEDED
    # the stale ckpt was cleared at create: the new life really trained
    # (its metrics.json exists and reports step 2 of THIS run)
    m = json.load(open(tmp_path / "jobs" / "fresh" / "metrics.json"))
    assert m["step"] == 2
    # and no trace of the planted stale checkpoint survived the create
    log = (tmp_path / "jobs" / "fresh" /
           "fresh-master-0.log").read_text()
    assert "resumed at step" not in log
