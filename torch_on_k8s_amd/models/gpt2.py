"""GPT-2-family transformer (second model family of the data plane).

Demonstrates that the MI355X op layer generalizes beyond Llama:
LayerNorm (fused gfx950 kernel), learned positional embeddings, MHA
(num_kv_heads == num_heads through the same flash-attention kernel),
GELU MLP, tied embeddings, pre-LN blocks. Same flat-bucket DP / fused
AdamW / fused CE path as Llama.
"""
from __future__ import annotations

from dataclasses import dataclass, asdict

import torch
import torch.nn as nn
import torch.nn.functional as F

from torch_on_k8s_amd import ops


@dataclass
class GPT2Config:
    name: str = "gpt2-small"
    vocab_size: int = 50304          # 50257 padded to a multiple of 64
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    head_dim: int = 64
    intermediate_size: int = 3072
    max_seq_len: int = 1024
    ln_eps: float = 1e-5
    tie_embeddings: bool = True
    attn_impl: str = "hip"
    init_std: float = 0.02

    # shared-config surface used by the trainer/registry
    @property
    def num_kv_heads(self) -> int:
        return self.num_heads

    def to_dict(self):
        return asdict(self)


GPT2_PRESETS = {
    "gpt2-small": GPT2Config(),
    "gpt2-medium": GPT2Config(name="gpt2-medium", hidden_size=1024,
                              num_layers=24, num_heads=16),
    "gpt2-tiny": GPT2Config(name="gpt2-tiny", vocab_size=512, hidden_size=128,
                            num_layers=2, num_heads=2, intermediate_size=256,
                            max_seq_len=256),
}


class LayerNorm(nn.Module):
    def __init__(self, hidden: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))
        self.eps = eps

    def forward(self, x):
        return ops.layernorm(x, self.weight, self.bias, self.eps)


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        H = cfg.hidden_size
        self.ln_1 = LayerNorm(H, cfg.ln_eps)
        self.qkv = nn.Linear(H, 3 * H, bias=True)
        self.proj = nn.Linear(H, H, bias=True)
        self.ln_2 = LayerNorm(H, cfg.ln_eps)
        self.fc = nn.Linear(H, cfg.intermediate_size, bias=True)
        self.fc_out = nn.Linear(cfg.intermediate_size, H, bias=True)
        self.cfg = cfg

    def forward(self, x):
        cfg = self.cfg
        B, S, H = x.shape
        h = self.ln_1(x)
        q, k, v = self.qkv(h).split(H, dim=-1)
        q = q.view(B, S, cfg.num_heads, cfg.head_dim)
        k = k.view(B, S, cfg.num_heads, cfg.head_dim)
        v = v.view(B, S, cfg.num_heads, cfg.head_dim)
        if cfg.attn_impl == "sdpa" and x.is_cuda:
            o = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                is_causal=True).transpose(1, 2).contiguous()
        else:
            o = ops.attention(q, k, v, causal=True)
        x = x + self.proj(o.reshape(B, S, H))
        x = x + self.fc_out(F.gelu(self.fc(self.ln_2(x)), approximate="tanh"))
        return x


class GPT2Model(nn.Module):
    """Decoder-only GPT-2. forward() returns mean next-token CE loss."""

    def __init__(self, cfg: GPT2Config, activation_checkpointing: bool = False):
        super().__init__()
        self.cfg = cfg
        self.activation_checkpointing = activation_checkpointing
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.max_seq_len, cfg.hidden_size)
        self.blocks = nn.ModuleList(GPT2Block(cfg)
                                    for _ in range(cfg.num_layers))
        self.ln_f = LayerNorm(cfg.hidden_size, cfg.ln_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.wte.weight
        self.reset_parameters()

    def reset_parameters(self):
        std = self.cfg.init_std
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, mean=0.0, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, mean=0.0, std=std)
            elif isinstance(m, LayerNorm):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        scale = (2 * self.cfg.num_layers) ** -0.5
        for blk in self.blocks:
            with torch.no_grad():
                blk.proj.weight.mul_(scale)
                blk.fc_out.weight.mul_(scale)

    def forward(self, input_ids: torch.Tensor,
                labels: torch.Tensor | None = None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)
        for blk in self.blocks:
            if self.activation_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    blk, x, use_reentrant=False)
            else:
                x = blk(x)
        x = self.ln_f(x)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        return ops.cross_entropy(logits, labels)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, **_) -> torch.Tensor:
        """Greedy/sampled decoding so GPT-2 checkpoints serve through
        the same endpoint as Llama (serve.py). Full-context recompute
        per token — honest but unoptimized: the KV-cache/flash-decode/
        hipGraph serving fast path is Llama's (models/llama.py
        generate()); GPT-2 is the op-layer-generality family. Learned
        positions cap the total length at max_seq_len."""
        total = input_ids.shape[1] + max_new_tokens
        if total > self.cfg.max_seq_len:
            raise ValueError(
                f"prompt+max_new_tokens {total} exceeds learned positions "
                f"({self.cfg.max_seq_len})")
        out = input_ids
        for _ in range(max_new_tokens):
            logits = self.forward(out)[:, -1]
            if temperature and temperature > 0:
                probs = torch.softmax(logits.float() / temperature, dim=-1)
                nxt = torch.multinomial(probs, 1)
            else:
                nxt = logits.argmax(-1, keepdim=True)
            out = torch.cat([out, nxt], dim=1)
        return out
