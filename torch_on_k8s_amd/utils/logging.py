"""Unified logging (the reference mixes zap/klog/logrus/glog —
SURVEY.md §5.5 calls for one): a single stdlib logger hierarchy rooted
at "tok", rank-aware formatting for data-plane processes."""
from __future__ import annotations

import logging
import os
import sys

_CONFIGURED = False


def get_logger(name: str = "tok") -> logging.Logger:
    global _CONFIGURED
    if not _CONFIGURED:
        rank = os.environ.get("RANK")
        prefix = f"[rank {rank}] " if rank is not None else ""
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(logging.Formatter(
            f"%(asctime)s {prefix}%(name)s %(levelname)s: %(message)s",
            datefmt="%H:%M:%S"))
        root = logging.getLogger("tok")
        root.addHandler(handler)
        root.setLevel(os.environ.get("TOK_LOG_LEVEL", "INFO").upper())
        root.propagate = False
        _CONFIGURED = True
    return logging.getLogger(name if name.startswith("tok") else f"tok.{name}")
