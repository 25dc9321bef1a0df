#!/usr/bin/env python3
"""Flagship benchmark: gang-scheduled Llama-3-8B DP training step on
N MI355X GPUs (BASELINE.json metric: tokens/sec at 1/2/4/8 workers).

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one
rank per GPU (RCCL over xGMI). W untimed warmup steps, then EXACTLY K
timed steps bracketed by barrier + torch.cuda.synchronize on both sides;
elapsed = MAX over ranks; rank 0 prints one JSON line.

Synthetic data (no network for datasets), random-init weights, bf16.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from torch_on_k8s_amd.tunable import setup_tunableop  # noqa: E402

setup_tunableop()  # load committed hipBLASLt tuning results (if any)

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", type=str, default="llama3-8b")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--bucket-mb", type=int, default=256)
    ap.add_argument("--activation-checkpointing", action="store_true")
    ap.add_argument("--attn", type=str, default="hip", choices=["hip", "sdpa"])
    ap.add_argument("--no-overlap", action="store_true",
                    help="disable grad-sync/backward overlap (ablation)")
    ap.add_argument("--hip-graph", action="store_true",
                    help="capture the training step in a hipGraph")
    return ap.parse_args()


def main():
    args = parse_args()
    from torch_on_k8s_amd.parallel.env import init_distributed, destroy
    from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig

    ctx = init_distributed()
    n_gpus = ctx.world_size if ctx.is_distributed else args.gpus
    if not torch.cuda.is_available():
        print("bench.py requires a GPU", file=sys.stderr)
        sys.exit(1)

    cfg = TrainerConfig(
        model=args.model,
        model_overrides={"attn_impl": args.attn},
        micro_batch=args.micro_batch,
        seq_len=args.seq_len,
        bucket_mb=args.bucket_mb,
        activation_checkpointing=args.activation_checkpointing,
        overlap_grad_sync=not args.no_overlap,
        hip_graph=args.hip_graph,
    )
    trainer = Trainer(cfg, ctx)

    def barrier_sync():
        if ctx.is_distributed:
            dist.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        trainer.train_step(sync=False)

    barrier_sync()
    t0 = time.perf_counter()
    loss = None
    for _ in range(args.steps):
        loss = trainer.train_step(sync=False)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    if ctx.is_distributed:
        t = torch.tensor([elapsed], dtype=torch.float64, device=ctx.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    tokens_total = args.micro_batch * args.seq_len * n_gpus * args.steps
    value = tokens_total / elapsed
    if ctx.is_main:
        out = {
            "metric": "tokens_per_s",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.micro_batch * n_gpus,
                "seq_len": args.seq_len,
                "parallelism": f"dp{n_gpus}",
                "attn_impl": args.attn,
                "loss": float(loss.float().item()) if loss is not None else None,
            },
        }
        print(json.dumps(out), flush=True)
    destroy()


if __name__ == "__main__":
    main()
