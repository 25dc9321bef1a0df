"""MI355X-native fused ops.

Dispatch policy:
  * GPU tensors ALWAYS run the handwritten gfx950 HIP kernels from the
    in-tree extension ``torch_on_k8s_amd.ops._C``. If the extension is
    missing on a GPU machine, ops raise instead of silently falling back
    to eager PyTorch.
  * CPU tensors use plain fp32 PyTorch reference implementations (used by
    the CPU test suite and as the numerics oracle for the HIP kernels).

Capability parity: the reference (hliangzhao/torch-on-k8s) ships no
kernels at all -- it schedules opaque containers. These ops
are the data plane its env-var contract assumes exists inside the
container, built MI355X-first per SURVEY.md §7 step 2.
"""
from __future__ import annotations

import torch

try:  # the in-tree gfx950 extension; built by `setup.py build_ext --inplace`
    from . import _C  # type: ignore
except ImportError:  # pragma: no cover - exercised only on unbuilt checkouts
    _C = None


def hip_ext_available() -> bool:
    return _C is not None


def _require_ext(op: str):
    if _C is None:
        raise RuntimeError(
            f"torch_on_k8s_amd HIP extension is required for {op} on GPU but "
            "is not built. Run: PYTORCH_ROCM_ARCH=gfx950 python setup.py "
            "build_ext --inplace"
        )
    return _C


# --------------------------------------------------------------------------
# RMSNorm
# --------------------------------------------------------------------------
def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    """fp32 reference: y = x * w / sqrt(mean(x^2) + eps)."""
    xf = x.float()
    r = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * r * w.float()).to(x.dtype)


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        if x.is_cuda:
            y, invr = _require_ext("rmsnorm").rmsnorm_fwd(x, w, eps)
        else:
            xf = x.float()
            invr = torch.rsqrt(xf.pow(2).mean(-1) + eps).reshape(-1)
            y = (xf * invr.view(*x.shape[:-1], 1) * w.float()).to(x.dtype)
        ctx.save_for_backward(x, w, invr)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, invr = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            dx, dw = _C.rmsnorm_bwd(x, w, dy, invr)
        else:
            xf, wf, dyf = x.float(), w.float(), dy.float()
            H = x.shape[-1]
            r = invr.view(*x.shape[:-1], 1)
            c = (dyf * wf * xf).sum(-1, keepdim=True)
            dx = (r * (wf * dyf - xf * (r * r / H) * c)).to(x.dtype)
            dw = (dyf * xf * r).reshape(-1, H).sum(0)
        return dx, dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    return _RMSNormFn.apply(x.contiguous(), w, eps)


class _RMSNormResFn(torch.autograd.Function):
    """Fused residual-add + RMSNorm: (x, res, w) -> (y, xr) with
    xr = x + res (bf16) and y = rmsnorm(xr) * w. Backward fuses the
    downstream residual grad (dxr) into the norm's dx pass, and dres
    equals dx (the add routes gradients unchanged)."""

    @staticmethod
    def forward(ctx, x, res, w, eps):
        if x.is_cuda:
            y, xr, invr = _require_ext("rmsnorm_res").rmsnorm_res_fwd(
                x, res, w, eps)
        else:
            xr = x + res if res is not None else x
            xf = xr.float()
            invr = torch.rsqrt(xf.pow(2).mean(-1) + eps).reshape(-1)
            y = (xf * invr.view(*xr.shape[:-1], 1) * w.float()).to(x.dtype)
        ctx.save_for_backward(xr, w, invr)
        ctx.has_res = res is not None
        return y, xr

    @staticmethod
    def backward(ctx, dy, dxr):
        xr, w, invr = ctx.saved_tensors
        dy = dy.contiguous()
        if dxr is not None:
            dxr = dxr.contiguous()
        if xr.is_cuda:
            dx, dw = _C.rmsnorm_res_bwd(xr, w, dy, dxr, invr)
        else:
            xf, wf, dyf = xr.float(), w.float(), dy.float()
            H = xr.shape[-1]
            r = invr.view(*xr.shape[:-1], 1)
            c = (dyf * wf * xf).sum(-1, keepdim=True)
            dx = (r * (wf * dyf - xf * (r * r / H) * c)).to(xr.dtype)
            if dxr is not None:
                dx = dx + dxr
            dw = (dyf * xf * r).reshape(-1, H).sum(0)
        dres = dx if ctx.has_res else None
        return dx, dres, dw.to(w.dtype), None


def rmsnorm_residual(x: torch.Tensor, res: torch.Tensor | None,
                     w: torch.Tensor, eps: float = 1e-5):
    """Returns (normed, x+res). One pass instead of add + norm; the
    residual stream stays fused through backward too."""
    return _RMSNormResFn.apply(
        x.contiguous(), res.contiguous() if res is not None else None, w, eps)


# --------------------------------------------------------------------------
# RoPE (Llama rotate-half convention)
# --------------------------------------------------------------------------
def rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             sign: float = 1.0) -> torch.Tensor:
    """x: [B, S, H, D]; cos/sin: [S, D/2] fp32."""
    B, S, H, D = x.shape
    xf = x.float()
    x1, x2 = xf[..., : D // 2], xf[..., D // 2:]
    cs = cos.view(1, S, 1, D // 2)
    sn = sin.view(1, S, 1, D // 2) * sign
    return torch.cat([x1 * cs - x2 * sn, x2 * cs + x1 * sn], dim=-1).to(x.dtype)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        ctx.save_for_backward(cos, sin)
        ctx.heads = x.shape[2]
        if x.is_cuda:
            return _require_ext("rope").rope(x.contiguous(), cos, sin, x.shape[2], 1.0)
        return rope_ref(x, cos, sin, 1.0)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        if dy.is_cuda:
            dx = _C.rope(dy.contiguous(), cos, sin, ctx.heads, -1.0)
        else:
            dx = rope_ref(dy, cos, sin, -1.0)
        return dx, None, None


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """Apply rotary embedding to [B, S, H, D] activations."""
    return _RopeFn.apply(x, cos, sin)


# --------------------------------------------------------------------------
# Packed-QKV split + RoPE (one pass over the fused qkv projection output)
# --------------------------------------------------------------------------
def qkv_rope_ref(qkv: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                 Hq: int, Hkv: int, D: int, sign: float = 1.0):
    """fp32 reference. qkv: [B, S, (Hq+2Hkv)*D]; returns roped q, k and
    copied v with contiguous [B,S,H,D] layouts."""
    B, S, _ = qkv.shape
    parts = qkv.view(B, S, Hq + 2 * Hkv, D)
    q = parts[:, :, :Hq].contiguous()
    k = parts[:, :, Hq:Hq + Hkv].contiguous()
    v = parts[:, :, Hq + Hkv:].contiguous()
    return (rope_ref(q, cos, sin, sign), rope_ref(k, cos, sin, sign), v)


class _QKVRopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, cos, sin, Hq, Hkv, D):
        ctx.save_for_backward(cos, sin)
        ctx.dims = (Hq, Hkv, D)
        if qkv.is_cuda:
            q, k, v = _require_ext("qkv_rope").qkv_rope_fwd(
                qkv.contiguous(), cos, sin, Hq, Hkv, D)
            return q, k, v
        return qkv_rope_ref(qkv, cos, sin, Hq, Hkv, D)

    @staticmethod
    def backward(ctx, dq, dk, dv):
        cos, sin = ctx.saved_tensors
        Hq, Hkv, D = ctx.dims
        if dq.is_cuda:
            dqkv = _C.qkv_rope_bwd(dq.contiguous(), dk.contiguous(),
                                   dv.contiguous(), cos, sin)
        else:
            B, S = dq.shape[:2]
            dqf = rope_ref(dq, cos, sin, -1.0).view(B, S, Hq * D)
            dkf = rope_ref(dk, cos, sin, -1.0).view(B, S, Hkv * D)
            dqkv = torch.cat([dqf, dkf, dv.reshape(B, S, Hkv * D)], dim=-1)
        return dqkv, None, None, None, None, None


def qkv_rope(qkv: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             Hq: int, Hkv: int, D: int):
    """Split the packed qkv projection [B,S,(Hq+2Hkv)*D] into contiguous
    q/k/v with RoPE applied to q and k — one kernel instead of three
    splits + two rope launches (GEMM-packing lever, ROADMAP §1a)."""
    return _QKVRopeFn.apply(qkv, cos, sin, Hq, Hkv, D)


# --------------------------------------------------------------------------
# Fused SwiGLU over the packed gate_up projection
# --------------------------------------------------------------------------
def swiglu_ref(gu: torch.Tensor) -> torch.Tensor:
    """fp32 reference: silu(gate) * up over packed [..., 2I]."""
    I = gu.shape[-1] // 2
    g, u = gu.float().split(I, dim=-1)
    return (torch.nn.functional.silu(g) * u).to(gu.dtype)


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        ctx.save_for_backward(gu)
        if gu.is_cuda:
            return _require_ext("swiglu").swiglu_fwd(gu.contiguous())
        return swiglu_ref(gu)

    @staticmethod
    def backward(ctx, dout):
        (gu,) = ctx.saved_tensors
        if gu.is_cuda:
            return _C.swiglu_bwd(dout.contiguous(), gu)
        I = gu.shape[-1] // 2
        g, u = gu.float().split(I, dim=-1)
        sg = torch.sigmoid(g)
        silu = g * sg
        dsilu = sg * (1 + g * (1 - sg))
        df = dout.float()
        return torch.cat([df * u * dsilu, df * silu], dim=-1).to(gu.dtype)


def swiglu(gu: torch.Tensor) -> torch.Tensor:
    """silu(gate)*up over the PACKED gate_up output [..., 2I]; fwd+bwd
    each one fused pass (no materialized silu intermediate)."""
    return _SwiGLUFn.apply(gu)


# --------------------------------------------------------------------------
# Fused AdamW on a flat bucket
# --------------------------------------------------------------------------
def fused_adamw_(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
                 v: torch.Tensor, *, lr: float, beta1: float, beta2: float,
                 eps: float, weight_decay: float, step: int,
                 grad_scale: float = 1.0,
                 step_dev: torch.Tensor | None = None) -> None:
    """In-place AdamW over one flat bucket (p/g bf16, m/v fp32).

    step_dev: optional int32 device scalar overriding `step` — used under
    hipGraph replay, where host-side scalars are frozen into the graph.
    """
    if p.is_cuda:
        _require_ext("fused_adamw").adamw_(p, g, m, v, lr, beta1, beta2, eps,
                                           weight_decay, step, grad_scale,
                                           step_dev)
        return
    gf = g.float() * grad_scale
    pf = p.float()
    pf -= lr * weight_decay * pf
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    bc1 = 1.0 / (1.0 - beta1 ** step)
    bc2 = 1.0 / (1.0 - beta2 ** step)
    pf -= lr * (m * bc1) / ((v * bc2).sqrt() + eps)
    p.copy_(pf.to(p.dtype))


# --------------------------------------------------------------------------
# Flash attention (causal, GQA)
# --------------------------------------------------------------------------
def attention_ref(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  causal: bool = True) -> torch.Tensor:
    """fp32 reference attention. q: [B,S,Hq,D], k/v: [B,S,Hkv,D]."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qf = q.float().permute(0, 2, 1, 3)  # B,Hq,S,D
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    vf = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) / (D ** 0.5)
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    o = torch.matmul(p, vf)
    return o.permute(0, 2, 1, 3).to(q.dtype)


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        if q.is_cuda:
            ext = _require_ext("flash_attention")
            if not hasattr(ext, "attn_fwd"):
                raise RuntimeError(
                    "HIP flash attention kernel missing from extension; "
                    "rebuild the extension")
            o, lse = ext.attn_fwd(q, k, v, causal)
            ctx.save_for_backward(q, k, v, o, lse)
            ctx.causal = causal
            return o
        # CPU path: differentiable reference (no custom backward needed).
        raise RuntimeError("use attention() entry point for CPU tensors")

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = _C.attn_bwd(q, k, v, o, lse, do.contiguous(), ctx.causal)
        return dq, dk, dv, None


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              causal: bool = True) -> torch.Tensor:
    """Causal GQA attention: HIP flash kernel on GPU, reference on CPU."""
    if q.is_cuda:
        return _FlashAttnFn.apply(q.contiguous(), k.contiguous(),
                                  v.contiguous(), causal)
    return attention_ref(q, k, v, causal)


# --------------------------------------------------------------------------
# KV-cache attention decode (serving path; no autograd)
# --------------------------------------------------------------------------
@torch.no_grad()
def attention_decode(q: torch.Tensor, kcache: torch.Tensor,
                     vcache: torch.Tensor, T: int,
                     T_dev: torch.Tensor | None = None) -> torch.Tensor:
    """q: [B, Hq, D] new-token queries (post-RoPE); kcache/vcache:
    [B, Tmax, Hkv, D] with the first T rows valid. Returns [B, Hq, D].
    T_dev (int32 device scalar) overrides T under hipGraph replay."""
    if q.is_cuda:
        return _require_ext("attention_decode").attn_decode(
            q.contiguous(), kcache, vcache, T, T_dev)
    B, Hq, D = q.shape
    Hkv = kcache.shape[2]
    rep = Hq // Hkv
    kf = kcache[:, :T].float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    vf = vcache[:, :T].float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    s = torch.einsum("bhd,bhtd->bht", q.float(), kf) / (D ** 0.5)
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bht,bhtd->bhd", p, vf).to(q.dtype)


# --------------------------------------------------------------------------
# LayerNorm (GPT-family normalization)
# --------------------------------------------------------------------------
class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, eps):
        if x.is_cuda:
            y, mu, rstd = _require_ext("layernorm").layernorm_fwd(x, w, b, eps)
        else:
            xf = x.float()
            mu = xf.mean(-1)
            var = xf.var(-1, unbiased=False)
            rstd = torch.rsqrt(var + eps)
            y = (((xf - mu.unsqueeze(-1)) * rstd.unsqueeze(-1)) * w.float() +
                 b.float()).to(x.dtype)
            mu, rstd = mu.reshape(-1), rstd.reshape(-1)
        ctx.save_for_backward(x, w, mu, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, mu, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            dx, dw, db = _C.layernorm_bwd(x, w, dy, mu, rstd)
        else:
            H = x.shape[-1]
            xf, wf, dyf = x.float(), w.float(), dy.float()
            muv = mu.view(*x.shape[:-1], 1)
            rs = rstd.view(*x.shape[:-1], 1)
            xhat = (xf - muv) * rs
            dyw = dyf * wf
            a1 = dyw.mean(-1, keepdim=True)
            a2 = (dyw * xhat).mean(-1, keepdim=True)
            dx = (rs * (dyw - a1 - xhat * a2)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, H).sum(0)
            db = dyf.reshape(-1, H).sum(0)
        return dx, dw.to(w.dtype), db.to(w.dtype), None


def layernorm(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
              eps: float = 1e-5) -> torch.Tensor:
    return _LayerNormFn.apply(x.contiguous(), w, b, eps)


# --------------------------------------------------------------------------
# Fused cross-entropy (mean reduction over all tokens)
# --------------------------------------------------------------------------
class _FusedCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        loss, lse = _require_ext("cross_entropy").ce_fwd(logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return loss.mean()

    @staticmethod
    def backward(ctx, grad_out):
        logits, labels, lse = ctx.saved_tensors
        # device-scalar grad scale: stays correct under hipGraph replay
        gscale = (grad_out.float() / logits.numel() * logits.size(-1)) \
            .reshape(())
        dlogits = _C.ce_bwd(logits, labels, lse, gscale.contiguous())
        return dlogits, None


def cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean next-token CE. GPU bf16: fused single-pass kernels (one logits
    read fwd, one read + bf16 grad write bwd). CPU: eager fp32."""
    if logits.is_cuda and logits.dtype == torch.bfloat16:
        return _FusedCEFn.apply(logits.contiguous().view(-1, logits.shape[-1]),
                                labels.reshape(-1).int())
    return torch.nn.functional.cross_entropy(
        logits.float().view(-1, logits.shape[-1]), labels.reshape(-1))


__all__ = [
    "rmsnorm", "rmsnorm_ref", "apply_rope", "rope_ref", "fused_adamw_",
    "attention", "attention_ref", "cross_entropy", "layernorm",
    "qkv_rope", "qkv_rope_ref", "swiglu", "swiglu_ref",
    "hip_ext_available",
]
