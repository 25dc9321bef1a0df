#!/usr/bin/env python3
"""Flagship benchmark: gang-scheduled Llama-3-8B DP training step on
N MI355X GPUs (BASELINE.json metric: tokens/sec at 1/2/4/8 workers;
p50 job-to-Running).

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one
rank per GPU (RCCL over xGMI). W untimed warmup steps, then EXACTLY K
timed steps bracketed by barrier + torch.cuda.synchronize on both sides;
elapsed = MAX over ranks; rank 0 prints one JSON line.

Self-defending N-GPU semantics:
  * n_gpus in the output is ALWAYS the measured torch.distributed world
    size — never the --gpus flag.
  * If --gpus > 1 and the process is not already one rank of an N-rank
    launch (WORLD_SIZE unset), bench.py re-execs itself under
    torch.distributed.run --standalone so all N ranks really exist.
  * If WORLD_SIZE is set but != --gpus, it hard-fails: a mis-launched
    benchmark must not report an inflated aggregate.

Modes:
  default        bare training step loop (what the driver runs)
  --via-manager  the BASELINE.json headline path: submit a gang-scheduled
                 TorchJob through the control-plane manager; tokens/s is
                 measured INSIDE the job (entrypoint TOK_BENCH_* timed
                 region, same bracketing) and job-to-Running latency
                 comes from the job's condition timestamps.

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


def parse_args(argv=None):
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
    ap.add_argument("--hip-graph", dest="hip_graph", action="store_true",
                    default=None,
                    help="capture the training step in a hipGraph "
                         "(default: ON for single-GPU runs — measured "
                         "+1.1%% at mbs8; OFF at N>1 where RCCL-in-graph "
                         "is unvalidated)")
    ap.add_argument("--no-hip-graph", dest="hip_graph", action="store_false")
    ap.add_argument("--via-manager", action="store_true",
                    help="run the step through a gang-scheduled TorchJob "
                         "(control-plane path; reports job-to-Running too)")
    ap.add_argument("--device", type=str, default="cuda",
                    choices=["cuda", "cpu"],
                    help="cpu is for harness self-tests only (gloo)")
    ap.add_argument("--timeout", type=float, default=1800.0,
                    help="via-manager: max seconds to wait for the job")
    return ap.parse_args(argv)


def _resolve_hip_graph(args, world: int) -> bool:
    if args.hip_graph is not None:
        return args.hip_graph
    return world == 1 and args.device == "cuda"


def _maybe_self_spawn(args):
    """Direct `python bench.py --gpus N` (N>1) without a torchrun wrapper:
    re-exec under torch.distributed.run so N real ranks exist. The
    re-execed children see WORLD_SIZE and fall through."""
    if args.gpus <= 1 or "WORLD_SIZE" in os.environ or args.via_manager:
        return
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--standalone", "--nnodes=1", f"--nproc-per-node={args.gpus}",
           "--local-addr", "127.0.0.1",
           os.path.abspath(__file__)] + sys.argv[1:]
    print(f"[bench] --gpus {args.gpus} without torchrun: re-exec "
          f"{' '.join(cmd[:6])} ...", file=sys.stderr, flush=True)
    os.execv(sys.executable, cmd)


def emit(args, *, value, elapsed, n_gpus, loss, extra_config=None):
    out = {
        "metric": "tokens_per_s",
        "value": value,
        "unit": "tokens/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",  # per-GPU work fixed (micro-batch per rank)
        "vs_baseline": None,  # reference publishes no numbers
        "dtype": "bf16" if args.device == "cuda" else "fp32",
        "data": "synthetic",
        "config": {
            "model": args.model,
            "global_batch": args.micro_batch * n_gpus,
            "seq_len": args.seq_len,
            "parallelism": f"dp{n_gpus}",
            "attn_impl": args.attn,
            "loss": loss,
            **(extra_config or {}),
        },
    }
    print(json.dumps(out), flush=True)


def run_direct(args):
    import torch
    import torch.distributed as dist
    from torch_on_k8s_amd.parallel.env import init_distributed, destroy
    from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig

    if args.device == "cuda" and not torch.cuda.is_available():
        print("bench.py requires a GPU (use --device cpu only for "
              "harness self-tests)", file=sys.stderr)
        sys.exit(1)
    # guard BEFORE init_process_group (which would block waiting for
    # ranks that don't exist): the env world must match --gpus
    env_world = int(os.environ.get("WORLD_SIZE", "1"))
    if env_world != args.gpus:
        print(f"bench.py: WORLD_SIZE={env_world} but --gpus {args.gpus}; "
              f"refusing to report a mis-scaled aggregate", file=sys.stderr)
        sys.exit(2)
    ctx = init_distributed(backend="gloo" if args.device == "cpu" else None)
    if args.device == "cpu":
        ctx.device = torch.device("cpu")
    # measured world size is the ONLY source of n_gpus
    n_gpus = ctx.world_size
    if n_gpus != args.gpus:
        print(f"bench.py: launched with world_size={n_gpus} but "
              f"--gpus {args.gpus}; refusing to report a mis-scaled "
              f"aggregate", file=sys.stderr)
        destroy()
        sys.exit(2)

    cfg = TrainerConfig(
        model=args.model,
        model_overrides={"attn_impl": args.attn},
        micro_batch=args.micro_batch,
        seq_len=args.seq_len,
        bucket_mb=args.bucket_mb,
        activation_checkpointing=args.activation_checkpointing,
        overlap_grad_sync=not args.no_overlap,
        hip_graph=_resolve_hip_graph(args, n_gpus),
        dtype="bf16" if args.device == "cuda" else "fp32",
    )
    trainer = Trainer(cfg, ctx)

    def barrier_sync():
        if ctx.is_distributed:
            dist.barrier()
        if args.device == "cuda":
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
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=ctx.device if args.device == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    tokens_total = args.micro_batch * args.seq_len * n_gpus * args.steps
    if ctx.is_main:
        emit(args, value=tokens_total / elapsed, elapsed=elapsed,
             n_gpus=n_gpus,
             loss=float(loss.float().item()) if loss is not None else None)
    destroy()


def run_via_manager(args):
    """The BASELINE.json headline path: tokens/s of a gang-scheduled
    TorchJob (measured inside the job, same bracketing) + job-to-Running
    latency from the manager's condition timestamps (the launch-delay
    histogram's observation, reference pkg/metrics/metrics.go:58-66)."""
    import tempfile
    from torch_on_k8s_amd.manager import Manager
    from torch_on_k8s_amd.controlplane.api import JobConditionType

    workdir = tempfile.mkdtemp(prefix="tok-bench-")
    mgr = Manager(workdir, num_gpus=max(1, args.gpus), sync_period=0.05)

    n = args.gpus
    trainer_cfg = {
        "model": args.model,
        "model_overrides": {"attn_impl": args.attn},
        "micro_batch": args.micro_batch,
        "seq_len": args.seq_len,
        "bucket_mb": args.bucket_mb,
        "activation_checkpointing": args.activation_checkpointing,
        "overlap_grad_sync": not args.no_overlap,
        "hip_graph": _resolve_hip_graph(args, n),
        "dtype": "bf16" if args.device == "cuda" else "fp32",
    }
    env = {
        "TOK_TRAIN_STEPS": str(args.warmup + args.steps),
        "TOK_BENCH_STEPS": str(args.steps),
        "TOK_BENCH_WARMUP": str(args.warmup),
        "TOK_TRAINER_CONFIG": json.dumps(trainer_cfg),
    }
    if args.device == "cpu":
        env["TOK_BACKEND"] = "gloo"
    tasks = {"master": {"replicas": 1, "gpusPerTask": 1, "env": env}}
    if n > 1:
        tasks["worker"] = {"replicas": n - 1, "gpusPerTask": 1, "env": env}
    spec = {
        "apiVersion": "train.distributed.io/v1alpha1",
        "kind": "TorchJob",
        "metadata": {"name": "bench"},
        "spec": {"schedulingPolicy": {"minAvailable": n}, "tasks": tasks},
    }
    import yaml
    with open(os.path.join(mgr.spool, "bench.yaml"), "w") as f:
        yaml.safe_dump(spec, f)

    t_submit = time.time()
    deadline = t_submit + args.timeout
    job = None
    while time.time() < deadline:
        mgr.step()
        job = mgr.controller.jobs.get("bench")
        if job is not None and job.status.phase in (
                JobConditionType.SUCCEEDED, JobConditionType.FAILED):
            break
        time.sleep(0.05)
    if job is None or job.status.phase != JobConditionType.SUCCEEDED:
        phase = job.status.phase if job is not None else None
        print(f"bench.py --via-manager: job did not succeed "
              f"(phase={phase})", file=sys.stderr)
        sys.exit(3)

    # p50 job-to-Running over this run's sample (created -> Running
    # condition ts; what the launch-delay histogram observes)
    created_ts = running_ts = None
    for c in job.status.conditions:
        if c.type == JobConditionType.CREATED and created_ts is None:
            created_ts = c.ts
        if c.type == JobConditionType.RUNNING and running_ts is None:
            running_ts = c.ts
    job_to_running = (running_ts - created_ts) \
        if created_ts and running_ts else None

    with open(os.path.join(workdir, "jobs", "bench", "bench.json")) as f:
        rec = json.load(f)
    measured_world = rec["world_size"]
    if measured_world != n:
        print(f"bench.py --via-manager: job measured world_size="
              f"{measured_world} != --gpus {n}", file=sys.stderr)
        sys.exit(2)
    elapsed = rec["elapsed_s"]
    tokens_total = args.micro_batch * args.seq_len * measured_world * args.steps
    emit(args, value=tokens_total / elapsed, elapsed=elapsed,
         n_gpus=measured_world, loss=rec.get("loss"),
         extra_config={"via_manager": True,
                       "p50_job_to_running_s": job_to_running})


def main():
    args = parse_args()
    _maybe_self_spawn(args)
    if args.via_manager:
        run_via_manager(args)
    else:
        run_direct(args)


if __name__ == "__main__":
    main()
