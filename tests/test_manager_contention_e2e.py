"""Manager-level two-tenant contention e2e (BASELINE config 5 shape, CPU
scale): two TorchJobs in separate WRR queues with GPU quotas contend for
one node; the coordinator admits within quota, the gang holds the other
job until capacity frees, both train to completion with real processes,
and the winner's checkpoint is packaged as a ModelVersion."""
import json
import os
import time

import pytest

from torch_on_k8s_amd.client import TorchJobClient
from torch_on_k8s_amd.controlplane import features as feat
from torch_on_k8s_amd.manager import Manager

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

JOB_TMPL = """
kind: TorchJob
metadata: {{name: {name}}}
spec:
  schedulingPolicy: {{queue: {queue}, minAvailable: {workers}}}
  modelName: {name}-model
  tasks:
    master:
      replicas: 1
      gpusPerTask: 1
      env: &env
        TOK_BACKEND: gloo
        TOK_TRAIN_STEPS: "4"
        TOK_TRAINER_CONFIG: '{{"model": "llama-tiny", "micro_batch": 1, "seq_len": 32}}'
        PYTHONPATH: {root}
    worker:
      replicas: {workers}
      gpusPerTask: 1
      env: *env
"""


@pytest.mark.timeout(420)
def test_two_tenant_contention(tmp_path):
    # 4-GPU node; each job wants 3 GPUs (1 master + 2 workers) and each
    # tenant has quota for exactly one such job -> strictly serialized
    mgr = Manager(str(tmp_path), num_gpus=4,
                  quotas={"qa": 3, "qb": 3},
                  gates=feat.FeatureGates())
    cli = TorchJobClient(str(tmp_path))
    cli.apply(JOB_TMPL.format(name="job-a", queue="qa", workers=2, root=ROOT))
    cli.apply(JOB_TMPL.format(name="job-b", queue="qb", workers=2, root=ROOT))

    t0 = time.time()
    seen_concurrent_gpus = 0
    while time.time() - t0 < 360:
        mgr.step()
        used = 4 - len(mgr.controller.node.free_slots)
        seen_concurrent_gpus = max(seen_concurrent_gpus, used)
        sa = cli.get("job-a") or {}
        sb = cli.get("job-b") or {}
        if sa.get("phase") == "Succeeded" and sb.get("phase") == "Succeeded":
            break
        time.sleep(0.2)

    sa, sb = cli.get("job-a"), cli.get("job-b")
    assert sa and sa["phase"] == "Succeeded", sa
    assert sb and sb["phase"] == "Succeeded", sb
    # node capacity (4) < combined demand (6): never oversubscribed
    assert seen_concurrent_gpus <= 4
    # both models packaged
    assert sa["modelVersion"] and sb["modelVersion"]
    assert mgr.registry.models["job-a-model"].latest_version == sa["modelVersion"]
    assert mgr.registry.models["job-b-model"].latest_version == sb["modelVersion"]
    # all GPU slots returned
    mgr.step()
    assert len(mgr.controller.node.free_slots) == 4
