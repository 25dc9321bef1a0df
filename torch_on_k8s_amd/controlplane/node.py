"""Node GPU-slot accounting: the scheduling substrate for one 8xMI355X
box. Replaces the reference's cluster-level resource model (pods +
ResourceQuota, nvidia.com/gpu) with amd.com/gpu slots on a single node
(resource name parity change from
constants.go:28)."""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class NodeState:
    num_gpus: int = 8
    # slot -> (job_name, task_key) or None
    alloc: dict = field(default_factory=dict)

    def __post_init__(self):
        for i in range(self.num_gpus):
            self.alloc.setdefault(i, None)

    @property
    def free_slots(self) -> list:
        return [s for s, owner in self.alloc.items() if owner is None]

    def allocate(self, n: int, owner) -> tuple:
        if n < 0:
            # a negative slice would silently grant len(free)+n slots
            raise ValueError(f"negative GPU request: {n}")
        free = self.free_slots
        if len(free) < n:
            raise RuntimeError(
                f"insufficient GPUs: want {n}, free {len(free)}")
        got = tuple(free[:n])
        for s in got:
            self.alloc[s] = owner
        return got

    def allocate_specific(self, slots, owner) -> tuple:
        """Claim EXACT slots (adoption of an already-running task after
        a manager restart — the process still holds those GPUs)."""
        for s in slots:
            if self.alloc.get(s) is not None and self.alloc[s] != owner:
                raise RuntimeError(f"slot {s} already owned by "
                                   f"{self.alloc[s]}")
        for s in slots:
            self.alloc[s] = owner
        return tuple(slots)

    def release(self, slots) -> None:
        for s in slots:
            self.alloc[s] = None

    def release_owner(self, owner) -> None:
        for s, o in list(self.alloc.items()):
            if o == owner:
                self.alloc[s] = None

    def used_by(self, owner) -> int:
        return sum(1 for o in self.alloc.values() if o == owner)
