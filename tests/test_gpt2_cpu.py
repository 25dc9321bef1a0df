"""GPT-2 family on CPU: forward/backward, trainer integration, LayerNorm
reference math."""
import torch

from torch_on_k8s_amd import ops
from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
from torch_on_k8s_amd.models.gpt2 import GPT2Model, GPT2_PRESETS
from torch_on_k8s_amd.parallel.env import DistContext


def test_layernorm_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(8, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    b = torch.randn(64, requires_grad=True)
    dy = torch.randn(8, 64)

    y = ops.layernorm(x, w, b, 1e-5)
    y.backward(dy)

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.layer_norm(x2, (64,), w2, b2, 1e-5)
    y2.backward(dy)

    assert torch.allclose(y, y2, atol=1e-5)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
    assert torch.allclose(b.grad, b2.grad, atol=1e-4)


def test_gpt2_forward_backward():
    torch.manual_seed(0)
    model = GPT2Model(GPT2_PRESETS["gpt2-tiny"])
    ids = torch.randint(0, 512, (2, 32))
    loss = model(ids, ids)
    assert torch.isfinite(loss)
    loss.backward()
    assert model.blocks[0].qkv.weight.grad is not None
    # tied embeddings: one shared parameter
    assert model.lm_head.weight is model.wte.weight


def test_gpt2_trainer_loss_decreases():
    torch.manual_seed(0)
    tr = Trainer(TrainerConfig(model="gpt2-tiny", micro_batch=2, seq_len=64,
                               lr=1e-3), DistContext())
    losses = []
    for _ in range(8):
        tr.step_count = 0
        losses.append(tr.train_step())
        tr.step_count = 1
    assert losses[-1] < losses[0], losses


def test_gpt2_generate_and_serve():
    """GPT-2 serves through the same HTTP endpoint as Llama: greedy
    generate is deterministic, respects the learned-position cap, and
    the FastAPI route round-trips it."""
    import pytest
    import torch
    from torch_on_k8s_amd.serve import InferenceServer, build_app
    from starlette.testclient import TestClient

    torch.manual_seed(0)
    srv = InferenceServer.from_preset("gpt2-tiny", "cpu")
    out = srv.generate([[1, 2, 3]], max_new_tokens=4)
    assert len(out["output_ids"][0]) == 7
    out2 = srv.generate([[1, 2, 3]], max_new_tokens=4)
    assert out["output_ids"] == out2["output_ids"]  # greedy = deterministic
    with pytest.raises(ValueError, match="learned positions"):
        srv.model.generate(torch.zeros(1, 250, dtype=torch.long),
                           max_new_tokens=16)
    c = TestClient(build_app(srv))
    r = c.post("/v1/generate", json={"prompt_ids": [[5, 6]],
                                     "max_new_tokens": 3})
    assert r.status_code == 200 and len(r.json()["new_ids"][0]) == 3
    # past max_seq_len -> 400 from the endpoint guard
    r = c.post("/v1/generate", json={"prompt_ids": [[1] * 250],
                                     "max_new_tokens": 16})
    assert r.status_code == 400
