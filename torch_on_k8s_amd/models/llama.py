"""Llama-family transformer, MI355X-first.

This is the flagship model of the data plane (SURVEY.md §7 step 2 /
BASELINE.json: gang-scheduled Llama-3-8B tokens/sec is the headline
metric). Design choices for MI355X:

  * bf16 weights/activations end-to-end; fp32 accumulation inside the
    fused HIP ops (RMSNorm, RoPE, flash attention, AdamW).
  * Plain projection GEMMs go through hipBLASLt/rocBLAS via torch.matmul;
    everything fusable is a handwritten gfx950 kernel in ops/.
  * RoPE cos/sin tables precomputed on host (fp32) -- no device trig.
  * 288 GB HBM3E per GPU: default training config holds the whole model,
    grads and fp32 Adam state resident without sharding; activation
    checkpointing is optional and off by default.
"""
from __future__ import annotations

from dataclasses import dataclass, asdict

import torch
import torch.nn as nn
import torch.nn.functional as F

from torch_on_k8s_amd import ops


@dataclass
class LlamaConfig:
    name: str = "llama3-8b"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    tie_embeddings: bool = False
    # "hip": fused gfx950 flash-attention kernel (CPU: fp32 reference);
    # "sdpa": torch scaled_dot_product_attention (used for A/B comparison).
    attn_impl: str = "hip"
    init_std: float = 0.02

    def to_dict(self):
        return asdict(self)


PRESETS = {
    # The headline config named by BASELINE.json.
    "llama3-8b": LlamaConfig(),
    # Single-layer-scale configs for tests / smoke.
    "llama-tiny": LlamaConfig(
        name="llama-tiny", vocab_size=512, hidden_size=256,
        intermediate_size=512, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=64, max_seq_len=256, rope_theta=10000.0),
    # Mid-size config for single-GPU kernel iteration.
    "llama-1b": LlamaConfig(
        name="llama-1b", vocab_size=32000, hidden_size=2048,
        intermediate_size=5504, num_layers=16, num_heads=16, num_kv_heads=8,
        head_dim=128, max_seq_len=4096, rope_theta=500000.0),
}


def get_config(name: str, **overrides) -> LlamaConfig:
    cfg = PRESETS[name]
    if overrides:
        d = cfg.to_dict()
        d.update(overrides)
        cfg = LlamaConfig(**d)
    return cfg


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.eps = eps

    def forward(self, x):
        return ops.rmsnorm(x, self.weight, self.eps)


def build_rope_table(cfg: LlamaConfig, seq_len: int, device) -> tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed fp32 [S, D/2] cos/sin tables."""
    half = cfg.head_dim // 2
    inv_freq = 1.0 / (cfg.rope_theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    t = torch.arange(seq_len, dtype=torch.float64)
    ang = torch.outer(t, inv_freq)
    return (ang.cos().float().contiguous().to(device),
            ang.sin().float().contiguous().to(device))


class Attention(nn.Module):
    """q/k/v projections PACKED into one GEMM (fewer, larger hipBLASLt
    calls — ROADMAP §1a); the packed output is split + roped by the
    single-pass qkv_rope kernel. Head order in qkv_proj's output:
    [q heads | k heads | v heads]."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        H, D = cfg.num_heads, cfg.head_dim
        Hkv = cfg.num_kv_heads
        self.qkv_proj = nn.Linear(cfg.hidden_size, (H + 2 * Hkv) * D,
                                  bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)

    def project_qkv(self, x, cos, sin):
        """Packed projection -> roped contiguous q, k, v."""
        cfg = self.cfg
        qkv = self.qkv_proj(x)
        return ops.qkv_rope(qkv, cos, sin, cfg.num_heads, cfg.num_kv_heads,
                            cfg.head_dim)

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        cfg = self.cfg
        q, k, v = self.project_qkv(x, cos, sin)
        if cfg.attn_impl == "sdpa" and x.is_cuda:
            rep = cfg.num_heads // cfg.num_kv_heads
            qt = q.transpose(1, 2)
            kt = k.transpose(1, 2).repeat_interleave(rep, dim=1)
            vt = v.transpose(1, 2).repeat_interleave(rep, dim=1)
            o = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True)
            o = o.transpose(1, 2).contiguous()
        else:
            o = ops.attention(q, k, v, causal=True)
        return self.o_proj(o.reshape(B, S, cfg.num_heads * cfg.head_dim))


class MLP(nn.Module):
    """gate/up packed into one GEMM; fused SwiGLU kernel reads the
    packed output directly (one pass fwd, one pass bwd, no silu
    intermediate in HBM)."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_up_proj = nn.Linear(cfg.hidden_size,
                                      2 * cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(ops.swiglu(self.gate_up_proj(x)))


class Block(nn.Module):
    """Residual stream threaded as (branch, residual) pairs so every
    residual-add fuses into the NEXT RMSNorm's read (ops.rmsnorm_residual:
    one kernel does add + norm fwd, and norm-dx + residual-grad add bwd).
    forward(x, res) returns (mlp_out, x_after_attn_add): the caller (next
    block or final norm) performs the pending add inside its fused norm."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.input_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.attn = Attention(cfg)
        self.post_attn_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.mlp = MLP(cfg)

    def forward(self, x, res, cos, sin):
        h, xr = ops.rmsnorm_residual(x, res, self.input_norm.weight,
                                     self.input_norm.eps)
        a = self.attn(h, cos, sin)
        h2, xr2 = ops.rmsnorm_residual(a, xr, self.post_attn_norm.weight,
                                       self.post_attn_norm.eps)
        return self.mlp(h2), xr2


class LlamaModel(nn.Module):
    """Decoder-only Llama. forward() returns mean next-token CE loss."""

    def __init__(self, cfg: LlamaConfig, activation_checkpointing: bool = False):
        super().__init__()
        self.cfg = cfg
        self.activation_checkpointing = activation_checkpointing
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(Block(cfg) for _ in range(cfg.num_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed_tokens.weight
        self._rope_cache: tuple[int, torch.Tensor, torch.Tensor] | None = None
        self.reset_parameters()

    def reset_parameters(self):
        std = self.cfg.init_std
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, mean=0.0, std=std)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, mean=0.0, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)
        # scale down output projections per residual-branch count
        scale = (2 * self.cfg.num_layers) ** -0.5
        for blk in self.layers:
            with torch.no_grad():
                blk.attn.o_proj.weight.mul_(scale)
                blk.mlp.down_proj.weight.mul_(scale)

    def _rope(self, S: int, device):
        if self._rope_cache is None or self._rope_cache[0] != S or \
                self._rope_cache[1].device != device:
            cos, sin = build_rope_table(self.cfg, S, device)
            self._rope_cache = (S, cos, sin)
        return self._rope_cache[1], self._rope_cache[2]

    def forward_hidden(self, input_ids: torch.Tensor) -> torch.Tensor:
        B, S = input_ids.shape
        cos, sin = self._rope(S, input_ids.device)
        x = self.embed_tokens(input_ids)
        res = None  # pending residual, consumed by each fused norm
        for blk in self.layers:
            if self.activation_checkpointing and self.training:
                x, res = torch.utils.checkpoint.checkpoint(
                    blk, x, res, cos, sin, use_reentrant=False)
            else:
                x, res = blk(x, res, cos, sin)
        y, _ = ops.rmsnorm_residual(x, res, self.norm.weight, self.norm.eps)
        return y

    def forward(self, input_ids: torch.Tensor,
                labels: torch.Tensor | None = None):
        x = self.forward_hidden(input_ids)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        return ops.cross_entropy(logits, labels)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0,
                 use_graph: bool | None = None) -> torch.Tensor:
        """Greedy (temperature=0) or sampled autoregressive generation —
        the serving path. Prefill runs through the flash-attention
        kernel; each decode step uses the KV-cache single-token decode
        kernel (ops.attention_decode)."""
        cfg = self.cfg
        B, S0 = input_ids.shape
        dev = input_ids.device
        dtype = self.embed_tokens.weight.dtype
        Tmax = S0 + max_new_tokens
        Hkv, Hq, D = cfg.num_kv_heads, cfg.num_heads, cfg.head_dim
        kc = [torch.zeros(B, Tmax, Hkv, D, dtype=dtype, device=dev)
              for _ in self.layers]
        vc = [torch.zeros(B, Tmax, Hkv, D, dtype=dtype, device=dev)
              for _ in self.layers]
        cos, sin = build_rope_table(cfg, Tmax, dev)

        def run_block(blk, i, x, pos0, T):
            """One block over x ([B,s,H]); writes this slice's K/V into
            the caches and attends over cache[:T]."""
            B_, s, _ = x.shape
            h = blk.input_norm(x)
            q, k, v = blk.attn.project_qkv(
                h, cos[pos0:pos0 + s].contiguous(),
                sin[pos0:pos0 + s].contiguous())
            kc[i][:, pos0:pos0 + s] = k
            vc[i][:, pos0:pos0 + s] = v
            if s > 1:  # prefill
                o = ops.attention(q, k, v, causal=True)
            else:      # single-token decode over the cache
                o = ops.attention_decode(q[:, 0], kc[i], vc[i], T)
                o = o.view(B_, 1, Hq, D)
            x = x + blk.attn.o_proj(o.reshape(B_, s, Hq * D))
            x = x + blk.mlp(blk.post_attn_norm(x))
            return x

        def next_token(logits):
            if temperature and temperature > 0:
                probs = torch.softmax(logits.float() / temperature, dim=-1)
                return torch.multinomial(probs, 1)
            return logits.argmax(-1, keepdim=True)

        # prefill
        x = self.embed_tokens(input_ids)
        for i, blk in enumerate(self.layers):
            x = run_block(blk, i, x, 0, S0)
        logits = self.lm_head(self.norm(x[:, -1:]))[:, -1]
        tokens = [next_token(logits)]

        # Greedy GPU decode is launch-bound (hundreds of tiny kernels per
        # token); capture one whole decode step in a hipGraph and replay
        # it per token. Position/length live on-device (index_copy_/
        # index_select + the decode kernel's T_dev) so replay stays
        # correct; the graph is self-feeding (argmax writes the token
        # buffer the next replay embeds).
        if use_graph is None:
            use_graph = (dev.type == "cuda" and not temperature and
                         max_new_tokens >= 8)

        if not use_graph:
            for step in range(1, max_new_tokens):
                pos = S0 + step - 1
                x = self.embed_tokens(tokens[-1])
                for i, blk in enumerate(self.layers):
                    x = run_block(blk, i, x, pos, pos + 1)
                logits = self.lm_head(self.norm(x))[:, -1]
                tokens.append(next_token(logits))
            return torch.cat([input_ids] + tokens, dim=1)

        token_buf = tokens[0].clone()                       # [B, 1]
        pos_long = torch.tensor([S0], dtype=torch.long, device=dev)
        t32 = torch.zeros((), dtype=torch.int32, device=dev)

        def decode_step():
            t32.copy_((pos_long[0] + 1).to(torch.int32))
            x = self.embed_tokens(token_buf)
            cs = torch.index_select(cos, 0, pos_long).contiguous()
            sn = torch.index_select(sin, 0, pos_long).contiguous()
            for i, blk in enumerate(self.layers):
                h = blk.input_norm(x)
                q, k, v = blk.attn.project_qkv(h, cs, sn)
                kc[i].index_copy_(1, pos_long, k)
                vc[i].index_copy_(1, pos_long, v)
                o = ops.attention_decode(q[:, 0], kc[i], vc[i], 0, T_dev=t32)
                x = x + blk.attn.o_proj(o.view(B, 1, Hq * D))
                x = x + blk.mlp(blk.post_attn_norm(x))
            logits = self.lm_head(self.norm(x))[:, -1]
            nxt = logits.argmax(-1, keepdim=True)
            token_buf.copy_(nxt)       # self-feed the next replay
            pos_long.add_(1)

        graph = None
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        n_warm = min(2, max_new_tokens - 1)
        with torch.cuda.stream(side):
            for _ in range(n_warm):    # real decode steps (allocator/rng warm)
                decode_step()
                tokens.append(token_buf.clone())
        torch.cuda.current_stream().wait_stream(side)
        if max_new_tokens - 1 > n_warm:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                decode_step()          # capture only; not executed
            for _ in range(max_new_tokens - 1 - n_warm):
                graph.replay()
                tokens.append(token_buf.clone())
        return torch.cat([input_ids] + tokens, dim=1)

    def num_params(self) -> int:
        seen, total = set(), 0
        for p in self.parameters():
            if id(p) not in seen:
                seen.add(id(p))
                total += p.numel()
        return total
