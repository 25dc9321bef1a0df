"""Distributed environment contract.

The control plane injects exactly the env-var contract the reference
operator wires into pods (reference: controllers/train/torchjob_controller.go:394-445):
MASTER_ADDR / MASTER_PORT / RANK / WORLD_SIZE (+ LOCAL_RANK for one
process per GPU on the 8xMI355X node). ``torch.distributed`` with the
"nccl" backend IS RCCL on ROCm; CPU test jobs use gloo.
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    device: torch.device = torch.device("cpu")
    backend: str | None = None

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    @property
    def is_main(self) -> bool:
        return self.rank == 0


def init_distributed(backend: str | None = None,
                     timeout_s: int = 300) -> DistContext:
    """Initialise torch.distributed from the env contract.

    WORLD_SIZE may be refreshed between restarts by the elastic control
    plane (reference behavior: WORLD_SIZE via downward-API annotation,
    torchjob_controller.go:419-445); we always re-read the env here.
    """
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        device = torch.device("cuda", torch.cuda.current_device())
    else:
        device = torch.device("cpu")

    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if use_gpu else "gloo"
        if backend == "nccl":
            # xGMI-aware RCCL tuning (flagged; parallel/rccl.py) must be
            # in the env before communicator init
            from torch_on_k8s_amd.parallel.rccl import apply_rccl_env
            apply_rccl_env(world)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "23456")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    return DistContext(rank=rank, world_size=world, local_rank=local_rank,
                       device=device, backend=backend)


def barrier(ctx: DistContext):
    if ctx.is_distributed and dist.is_initialized():
        dist.barrier()


def destroy():
    if dist.is_initialized():
        dist.destroy_process_group()
