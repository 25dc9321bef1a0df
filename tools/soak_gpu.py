#!/usr/bin/env python3
"""Stability soak: run the flagship training config for N steps on one
GPU and assert production invariants — loss decreases, no NaN/Inf, and
device memory is STABLE after warmup (the allocator reaches steady
state; growth means a leak in the fused-op/autograd wiring).

  python tools/soak_gpu.py --steps 150
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torch_on_k8s_amd.tunable import setup_tunableop  # noqa: E402
setup_tunableop()

import torch  # noqa: E402

from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig  # noqa
from torch_on_k8s_amd.parallel.env import DistContext  # noqa


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=150)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--hip-graph", action="store_true", default=True)
    ap.add_argument("--ckpt-every", type=int, default=0,
                    help="async periodic snapshot every N steps")
    ap.add_argument("--ckpt-dir", default="/tmp/soak-ckpt")
    args = ap.parse_args()

    torch.manual_seed(7)
    ctx = DistContext(device=torch.device("cuda", 0))
    tr = Trainer(TrainerConfig(
        model=args.model, micro_batch=args.micro_batch,
        seq_len=args.seq_len, lr=1e-4, hip_graph=args.hip_graph), ctx)

    losses = []
    mem_marks = {}
    writer = None
    snapshots = 0
    for i in range(args.steps):
        loss = float(tr.train_step())
        assert loss == loss and abs(loss) < 1e4, f"loss blew up: {loss}"
        losses.append(loss)
        if args.ckpt_every and (i + 1) % args.ckpt_every == 0 and \
                (writer is None or not writer.is_alive()):
            writer = tr.snapshot_checkpoint_async(args.ckpt_dir)
            snapshots += 1
        if i in (20, args.steps - 1):
            torch.cuda.synchronize()
            mem_marks[i] = torch.cuda.memory_allocated()
    if writer is not None:
        writer.join(timeout=600)
        # the published checkpoint must resume
        tr.load_checkpoint(args.ckpt_dir)
    first = sum(losses[:10]) / 10
    last = sum(losses[-10:]) / 10
    growth = mem_marks[args.steps - 1] - mem_marks[20]
    out = {
        "steps": args.steps,
        "loss_first10": round(first, 4),
        "loss_last10": round(last, 4),
        "mem_at_20_gb": round(mem_marks[20] / 2**30, 2),
        "mem_at_end_gb": round(mem_marks[args.steps - 1] / 2**30, 2),
        "mem_growth_mb": round(growth / 2**20, 2),
        "max_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 2),
        "async_snapshots": snapshots,
    }
    print(json.dumps(out))
    assert last < first, "loss did not decrease over the soak"
    assert growth <= 64 * 2**20, f"memory grew {growth/2**20:.0f} MB"
    print("SOAK_OK")


if __name__ == "__main__":
    main()
