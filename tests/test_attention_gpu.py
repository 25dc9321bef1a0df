"""GPU numerics: gfx950 flash-attention kernel vs fp32 reference.

Reference = ops.attention_ref (plain PyTorch fp32 on CPU) with
bf16-quantized inputs, per the repo test policy.
"""
import pytest
import torch

from torch_on_k8s_amd import ops

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda", 0)


def mk(B, S, Hq, Hkv, D, seed=0):
    torch.manual_seed(seed)
    q = (torch.randn(B, S, Hq, D) * 0.5).bfloat16().float()
    k = (torch.randn(B, S, Hkv, D) * 0.5).bfloat16().float()
    v = (torch.randn(B, S, Hkv, D) * 0.5).bfloat16().float()
    return q, k, v


CASES = [
    (2, 128, 4, 2, 128, True),
    (1, 256, 4, 4, 128, True),   # MHA (no GQA)
    (1, 200, 2, 1, 64, True),    # ragged S, D=64
    (1, 128, 2, 2, 64, False),   # non-causal
    (2, 512, 8, 2, 128, True),   # multiple kv tiles, GQA 4:1
]


@pytest.mark.parametrize("B,S,Hq,Hkv,D,causal", CASES)
def test_attn_forward(B, S, Hq, Hkv, D, causal):
    q, k, v = mk(B, S, Hq, Hkv, D)
    o_ref = ops.attention_ref(q, k, v, causal).float()
    og = ops.attention(q.bfloat16().to(dev()), k.bfloat16().to(dev()),
                       v.bfloat16().to(dev()), causal)
    err = (og.float().cpu() - o_ref).abs().max().item()
    assert err < 0.03, f"forward max err {err}"


@pytest.mark.parametrize("B,S,Hq,Hkv,D,causal", CASES)
def test_attn_backward(B, S, Hq, Hkv, D, causal):
    q, k, v = mk(B, S, Hq, Hkv, D, seed=1)
    do = (torch.randn(B, S, Hq, D) * 0.5).bfloat16().float()

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ops.attention_ref(qr, kr, vr, causal).float().backward(do)

    qg = q.bfloat16().to(dev()).requires_grad_(True)
    kg = k.bfloat16().to(dev()).requires_grad_(True)
    vg = v.bfloat16().to(dev()).requires_grad_(True)
    og = ops.attention(qg, kg, vg, causal)
    og.backward(do.bfloat16().to(dev()))

    for name, g_hip, g_ref in [("dq", qg.grad, qr.grad),
                               ("dk", kg.grad, kr.grad),
                               ("dv", vg.grad, vr.grad)]:
        err = (g_hip.float().cpu() - g_ref).abs().max().item()
        scale = max(1.0, g_ref.abs().max().item())
        assert err < 0.04 * scale, f"{name} max err {err} (scale {scale})"


def test_attn_spiked_softmax_branch():
    """Forces large max jumps between KV tiles (guide §5.4 rule 26: online
    softmax rescale paths need an input that exercises them)."""
    B, S, Hq, Hkv, D = 1, 256, 2, 2, 128
    q, k, v = mk(B, S, Hq, Hkv, D, seed=2)
    # spike one late K row so the running max jumps at tile 3
    k[0, 230] *= 30.0
    q[0, 240] *= 8.0
    o_ref = ops.attention_ref(q, k, v, True).float()
    og = ops.attention(q.bfloat16().to(dev()), k.bfloat16().to(dev()),
                       v.bfloat16().to(dev()), True)
    err = (og.float().cpu() - o_ref).abs().max().item()
    assert err < 0.05, f"spiked forward max err {err}"
