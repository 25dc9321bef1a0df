"""RCCL environment tuning for the 8xMI355X xGMI topology.

Rationale: each MI355X has 7 point-to-point xGMI links
at ~153 GB/s — there is no switch, so a ring algorithm's per-step
traffic is bound by ONE link regardless of the other six. RCCL spreads
a collective across links by running multiple channels; with few
channels (RCCL picks a small default for small worlds) the all-reduce
of large DP gradient buckets under-uses the fabric. These settings are
therefore about channel count and buffering, not algorithm overrides —
RCCL's topology detection already prefers the right algorithm for the
fully-connected node.

Applied only when TOK_RCCL_TUNE == "1" (OPT-IN; flag per r1 VERDICT
next-#2), and only via setdefault so an operator's explicit env always
wins. Opt-in because these values are unvalidated on real multi-GPU
hardware (an 8-GPU node was not available to this builder): forcing
more channels can also COST compute throughput during overlap — each
RCCL channel occupies CUs that the backward pass wants. Validate with
`bench.py --gpus N` A/B before enabling in production.
"""
from __future__ import annotations

import os

# xGMI-aware starting points:
#   MIN_NCHANNELS 28 = 4 channels per peer link (7 links) so large
#     bucket all-reduces can saturate more than one link;
#   NCCL_BUFFSIZE 8 MiB keeps per-channel staging buffers large enough
#     that 256 MiB gradient buckets don't fragment into latency-bound
#     chunks.
XGMI_TUNING = {
    "NCCL_MIN_NCHANNELS": "28",
    "NCCL_BUFFSIZE": str(8 << 20),
}

# The host driver in this deployment only supports dmabuf IPC; legacy
# IPC mode makes cross-process CUDA-tensor sharing fail with
# hipIpcGetMemHandle errors (environment contract).
REQUIRED = {
    "HSA_ENABLE_IPC_MODE_LEGACY": "0",
}


def apply_rccl_env(world_size: int = 0) -> dict:
    """Set RCCL tuning env (setdefault semantics). Returns what was
    applied. Call BEFORE init_process_group — RCCL reads env at
    communicator init."""
    applied = {}
    for k, v in REQUIRED.items():
        os.environ.setdefault(k, v)
        applied[k] = os.environ[k]
    if os.environ.get("TOK_RCCL_TUNE", "0") != "1":
        return applied
    if world_size and world_size < 2:
        return applied  # single rank: no collectives to tune
    for k, v in XGMI_TUNING.items():
        os.environ.setdefault(k, v)
        applied[k] = os.environ[k]
    return applied
