#!/usr/bin/env python3
"""Model-scale numerics check: train the flagship config for K steps
with the fully-fused HIP path vs torch SDPA attention (same seed, same
synthetic batches) and compare loss trajectories. Catches fused-kernel
numerics drift that elementwise unit tolerances can miss.

Run on a GPU box:  python tools/loss_ab.py --steps 30
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
from torch_on_k8s_amd.parallel.env import DistContext


def run(attn: str, steps: int, model: str, mbs: int, seq: int):
    torch.manual_seed(1234)
    ctx = DistContext(device=torch.device("cuda", 0))
    tr = Trainer(TrainerConfig(model=model, micro_batch=mbs, seq_len=seq,
                               model_overrides={"attn_impl": attn},
                               lr=3e-4), ctx)
    losses = []
    for _ in range(steps):
        losses.append(float(tr.train_step()))
    del tr
    torch.cuda.empty_cache()
    return losses


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--micro-batch", type=int, default=4)
    ap.add_argument("--seq-len", type=int, default=4096)
    args = ap.parse_args()

    hip = run("hip", args.steps, args.model, args.micro_batch, args.seq_len)
    sdpa = run("sdpa", args.steps, args.model, args.micro_batch,
               args.seq_len)
    diffs = [abs(a - b) for a, b in zip(hip, sdpa)]
    rel = [d / max(1e-6, abs(b)) for d, b in zip(diffs, sdpa)]
    out = {
        "steps": args.steps,
        "hip_first3": hip[:3], "hip_last3": hip[-3:],
        "sdpa_first3": sdpa[:3], "sdpa_last3": sdpa[-3:],
        "max_abs_diff": max(diffs), "max_rel_diff": max(rel),
        "hip_decreasing": hip[-1] < hip[0],
        "sdpa_decreasing": sdpa[-1] < sdpa[0],
    }
    print(json.dumps(out))
    # bf16 training: trajectories drift slowly; early steps must agree
    # tightly and the end state must stay in the same regime
    assert out["hip_decreasing"] and out["sdpa_decreasing"]
    assert max(rel[:5]) < 0.01, f"early-step divergence: {rel[:5]}"
    assert rel[-1] < 0.10, f"end-state divergence: {rel[-1]}"
    print("LOSS_AB_OK")


if __name__ == "__main__":
    main()
