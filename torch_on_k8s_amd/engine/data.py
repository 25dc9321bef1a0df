"""Synthetic token stream for training benchmarks.

There is no dataset/network access in the benchmark environment; the
headline metric (BASELINE.json) is defined on synthetic data with
random-init weights. Tokens are generated directly on the target device
(deterministic per rank+step), so the input pipeline costs ~0 and never
hides or inflates step time.
"""
from __future__ import annotations

import torch


class SyntheticTokens:
    def __init__(self, vocab_size: int, micro_batch: int, seq_len: int,
                 device: torch.device, rank: int = 0, seed: int = 1234):
        self.vocab_size = vocab_size
        self.micro_batch = micro_batch
        self.seq_len = seq_len
        self.device = device
        self.rank = rank
        self.seed = seed
        self._gen = torch.Generator(device=device)

    def batch(self, step: int) -> tuple[torch.Tensor, torch.Tensor]:
        """Returns (input_ids, labels), each [B, S]."""
        self._gen.manual_seed(self.seed + 1000003 * self.rank + step)
        toks = torch.randint(
            0, self.vocab_size, (self.micro_batch, self.seq_len + 1),
            device=self.device, generator=self._gen)
        return toks[:, :-1].contiguous(), toks[:, 1:].contiguous()
