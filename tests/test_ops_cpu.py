"""CPU sanity for op reference implementations (the numerics oracle that
the gfx950 HIP kernels are compared against in tests/test_ops_gpu.py)."""
import math

import torch

from torch_on_k8s_amd import ops


def test_rmsnorm_ref_formula():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y = ops.rmsnorm(x, w, 1e-5)
    expect = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(y, expect, atol=1e-5)


def test_rmsnorm_backward_matches_autograd():
    x = torch.randn(8, 64, requires_grad=True, dtype=torch.float64)
    w = torch.randn(64, requires_grad=True, dtype=torch.float64)

    def f(x, w):
        return (x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w)

    y_ref = f(x, w)
    dy = torch.randn_like(y_ref)
    gx_ref, gw_ref = torch.autograd.grad(y_ref, (x, w), dy)

    x32 = x.detach().float().requires_grad_(True)
    w32 = w.detach().float().requires_grad_(True)
    y = ops.rmsnorm(x32, w32, 1e-5)
    y.backward(dy.float())
    assert torch.allclose(x32.grad, gx_ref.float(), atol=1e-4)
    assert torch.allclose(w32.grad, gw_ref.float(), atol=1e-4)


def test_rope_ref_inverse():
    from torch_on_k8s_amd.models.llama import build_rope_table, get_config
    cfg = get_config("llama-tiny")
    cos, sin = build_rope_table(cfg, 32, torch.device("cpu"))
    x = torch.randn(2, 32, 4, cfg.head_dim)
    y = ops.rope_ref(x, cos, sin, 1.0)
    back = ops.rope_ref(y, cos, sin, -1.0)
    assert torch.allclose(back, x, atol=1e-5)


def test_rope_backward_is_inverse_rotation():
    from torch_on_k8s_amd.models.llama import build_rope_table, get_config
    cfg = get_config("llama-tiny")
    cos, sin = build_rope_table(cfg, 16, torch.device("cpu"))
    x = torch.randn(1, 16, 2, cfg.head_dim, requires_grad=True)
    y = ops.apply_rope(x, cos, sin)
    dy = torch.randn_like(y)
    y.backward(dy)
    assert torch.allclose(x.grad, ops.rope_ref(dy, cos, sin, -1.0), atol=1e-5)


def test_fused_adamw_matches_torch_adamw():
    torch.manual_seed(0)
    n = 128
    p = torch.randn(n)
    g = torch.randn(n)
    m = torch.zeros(n)
    v = torch.zeros(n)
    p2 = p.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p2], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    for step in range(1, 4):
        ops.fused_adamw_(p, g, m, v, lr=1e-2, beta1=0.9, beta2=0.95,
                         eps=1e-8, weight_decay=0.1, step=step)
        p2.grad = g.clone()
        opt.step()
    assert torch.allclose(p, p2.detach(), atol=1e-5)


def test_attention_ref_vs_sdpa():
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 2, 32, 4, 2, 32
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    o = ops.attention(q, k, v, causal=True)
    rep = Hq // Hkv
    o_ref = torch.nn.functional.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3),
        k.permute(0, 2, 1, 3).repeat_interleave(rep, dim=1),
        v.permute(0, 2, 1, 3).repeat_interleave(rep, dim=1),
        is_causal=True).permute(0, 2, 1, 3)
    assert torch.allclose(o, o_ref, atol=1e-5)


def test_swiglu_cpu_matches_eager():
    torch.manual_seed(0)
    gu = torch.randn(64, 128, requires_grad=True)
    out = ops.swiglu(gu)
    g, u = gu.detach().split(64, dim=-1)
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(out, ref, atol=1e-5)
    out.sum().backward()
    gu2 = gu.detach().clone().requires_grad_(True)
    g2, u2 = gu2.split(64, dim=-1)
    (torch.nn.functional.silu(g2) * u2).sum().backward()
    assert torch.allclose(gu.grad, gu2.grad, atol=1e-5)


def test_qkv_rope_cpu_matches_unpacked():
    torch.manual_seed(1)
    from torch_on_k8s_amd.models.llama import build_rope_table, get_config
    B, S, Hq, Hkv, D = 2, 16, 4, 2, 64
    cfg = get_config("llama-tiny", head_dim=D)
    cos, sin = build_rope_table(cfg, S, torch.device("cpu"))
    qkv = torch.randn(B, S, (Hq + 2 * Hkv) * D, requires_grad=True)
    q, k, v = ops.qkv_rope(qkv, cos, sin, Hq, Hkv, D)
    parts = qkv.detach().view(B, S, Hq + 2 * Hkv, D)
    q_ref = ops.rope_ref(parts[:, :, :Hq], cos, sin)
    k_ref = ops.rope_ref(parts[:, :, Hq:Hq + Hkv], cos, sin)
    v_ref = parts[:, :, Hq + Hkv:]
    assert torch.allclose(q, q_ref, atol=1e-5)
    assert torch.allclose(k, k_ref, atol=1e-5)
    assert torch.allclose(v, v_ref, atol=1e-5)
    # backward: rope grad is the inverse rotation (orthogonal), so
    # grad-of-sum through q+k+v must equal inverse-roped ones
    (q.sum() + k.sum() + v.sum()).backward()
    assert qkv.grad is not None and qkv.grad.shape == qkv.shape
