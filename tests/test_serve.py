"""Serving endpoint: the packaged ModelVersion (OCI artifact) a job
produces is directly loadable and servable over HTTP — the
train -> package -> serve loop the reference leaves at "image pushed".
All CPU (tiny model); the decode path is the same code that runs the
flash-decode kernel + hipGraph on MI355X."""
from __future__ import annotations

import json
import os

import pytest
import torch

from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
from torch_on_k8s_amd.parallel.env import DistContext
from torch_on_k8s_amd.serve import InferenceServer, build_app


def _client(app):
    from starlette.testclient import TestClient
    return TestClient(app)


@pytest.fixture(scope="module")
def trained_ckpt(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("serve")
    torch.manual_seed(0)
    tr = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                               seq_len=32), DistContext())
    tr.train_step()
    ck = str(tmp / "final")
    tr.save_checkpoint(ck, sharded=False)
    return ck


def test_serve_from_checkpoint(trained_ckpt):
    srv = InferenceServer.from_checkpoint(trained_ckpt, "cpu")
    app = build_app(srv)
    c = _client(app)
    r = c.get("/healthz")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    r = c.get("/v1/models")
    assert r.json()["model"] == "llama-tiny"
    assert r.json()["trained_steps"] == 1

    r = c.post("/v1/generate", json={
        "prompt_ids": [[1, 2, 3, 4], [5, 6, 7, 8]],
        "max_new_tokens": 6, "temperature": 0.0})
    assert r.status_code == 200, r.text
    out = r.json()
    assert len(out["output_ids"]) == 2
    assert all(len(o) == 4 + 6 for o in out["output_ids"])
    assert out["new_tokens"] == 12
    assert out["tokens_per_s"] > 0
    # greedy decode is deterministic
    r2 = c.post("/v1/generate", json={
        "prompt_ids": [[1, 2, 3, 4], [5, 6, 7, 8]],
        "max_new_tokens": 6, "temperature": 0.0})
    assert r2.json()["output_ids"] == out["output_ids"]


def test_serve_request_validation(trained_ckpt):
    srv = InferenceServer.from_checkpoint(trained_ckpt, "cpu")
    c = _client(build_app(srv))
    assert c.post("/v1/generate", json={"prompt_ids": []}).status_code == 400
    assert c.post("/v1/generate", json={
        "prompt_ids": [[1, 2], [3]]}).status_code == 400  # ragged
    assert c.post("/v1/generate", json={
        "prompt_ids": [[10**6]]}).status_code == 400      # out of vocab
    assert c.post("/v1/generate", json={
        "prompt_ids": [[1]], "max_new_tokens": 0}).status_code == 422
    # past the trained context: RoPE extrapolation degrades silently -> 400
    max_seq = srv.meta["model_config"]["max_seq_len"]
    r = c.post("/v1/generate", json={"prompt_ids": [[1] * (max_seq - 4)],
                                     "max_new_tokens": 16})
    assert r.status_code == 400 and "context" in r.json()["detail"]
    # exactly at the limit is allowed
    assert c.post("/v1/generate", json={"prompt_ids": [[1] * 4],
                                        "max_new_tokens": 2}).status_code == 200


def test_serve_from_oci_version(tmp_path, trained_ckpt):
    """End-to-end: pack the checkpoint into an OCI ModelVersion, then
    serve it by model:version reference (reindex from disk)."""
    import shutil
    from torch_on_k8s_amd.controlplane.modelregistry import (
        ModelRegistry, StorageProvider)
    workdir = tmp_path / "wd"
    sp = StorageProvider(str(workdir / "models"))
    reg = ModelRegistry(sp)
    # artifact source mirrors a job's output/: <src>/final/<ckpt files>
    src = tmp_path / "out"
    shutil.copytree(trained_ckpt, src / "final")
    mv = reg.build_version("served-model", "v7", str(src))
    assert mv.build_phase == "Succeeded"

    srv = InferenceServer.from_version(str(workdir), "served-model:v7",
                                       "cpu")
    assert srv.meta["model_version"] == "served-model:v7"
    c = _client(build_app(srv))
    r = c.post("/v1/generate", json={"prompt_ids": [[1, 2, 3]],
                                     "max_new_tokens": 4})
    assert r.status_code == 200
    assert len(r.json()["output_ids"][0]) == 7


def test_registry_reindex(tmp_path):
    from torch_on_k8s_amd.controlplane.modelregistry import (
        ModelRegistry, StorageProvider)
    sp = StorageProvider(str(tmp_path))
    reg = ModelRegistry(sp)
    src = tmp_path / "s"
    src.mkdir()
    (src / "f").write_text("x")
    reg.build_version("m", "v1", str(src))
    reg.build_version("m", "v2", str(src))
    # a FRESH registry over the same disk sees both versions
    reg2 = ModelRegistry(StorageProvider(str(tmp_path)))
    assert reg2.reindex() == 2
    assert reg2.models["m"].latest_version == "v2"
    assert reg2.get_version("m", "v1").build_phase == "Succeeded"


def test_serve_metrics_endpoint(trained_ckpt):
    srv = InferenceServer.from_checkpoint(trained_ckpt, "cpu")
    c = _client(build_app(srv))
    c.post("/v1/generate", json={"prompt_ids": [[1, 2]],
                                 "max_new_tokens": 3})
    r = c.get("/metrics")
    assert r.status_code == 200
    assert "tok_serve_requests_total 1" in r.text
    assert "tok_serve_tokens_out_total 3" in r.text


@pytest.mark.gpu
def test_serve_on_gpu():
    """Serving path on metal: bf16 model, flash-decode + hipGraph."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    srv = InferenceServer.from_preset("llama-tiny", "cuda", torch.bfloat16)
    c = _client(build_app(srv))
    r = c.post("/v1/generate", json={"prompt_ids": [[1, 2, 3, 4]] * 4,
                                     "max_new_tokens": 16})
    assert r.status_code == 200, r.text
    out = r.json()
    assert len(out["output_ids"]) == 4
    assert all(len(o) == 20 for o in out["output_ids"])
    # deterministic greedy decode on the graph path
    r2 = c.post("/v1/generate", json={"prompt_ids": [[1, 2, 3, 4]] * 4,
                                      "max_new_tokens": 16})
    assert r2.json()["output_ids"] == out["output_ids"]
