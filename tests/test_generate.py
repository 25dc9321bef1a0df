"""Serving path: KV-cache greedy generation must match re-running the
full forward at every step (the naive O(n^2) reference)."""
import pytest
import torch

from torch_on_k8s_amd.models.llama import LlamaModel, get_config


def naive_greedy(model, ids, n):
    out = ids
    for _ in range(n):
        logits = model(out)  # full forward, no cache
        out = torch.cat([out, logits[:, -1].argmax(-1, keepdim=True)], dim=1)
    return out


def test_generate_matches_naive_cpu():
    torch.manual_seed(0)
    cfg = get_config("llama-tiny")
    model = LlamaModel(cfg).eval()
    ids = torch.randint(0, cfg.vocab_size, (2, 12))
    want = naive_greedy(model, ids, 8)
    got = model.generate(ids, max_new_tokens=8)
    assert torch.equal(got, want), (got, want)


@pytest.mark.gpu
def test_generate_matches_naive_gpu():
    torch.manual_seed(0)
    cfg = get_config("llama-tiny")
    dev = torch.device("cuda", 0)
    model = LlamaModel(cfg).bfloat16().to(dev).eval()
    ids = torch.randint(0, cfg.vocab_size, (2, 12), device=dev)
    want = naive_greedy(model, ids, 8)
    got = model.generate(ids, max_new_tokens=8)
    # bf16 decode vs prefill kernels may diverge after many steps on
    # near-ties; require the first several tokens to agree exactly
    assert torch.equal(got[:, :12 + 4], want[:, :12 + 4]), (got, want)
