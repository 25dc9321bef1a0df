"""Feature gates (reference pkg/features/features.go:31-63 +
component-base featuregate): same gate names, same defaults, CLI
integration via --feature-gates k=v[,k=v...]."""
from __future__ import annotations

GANG_SCHEDULING = "GangScheduling"
DAG_SCHEDULING = "DAGScheduling"
JOB_COORDINATOR = "JobCoordinator"
TORCH_LOCAL_MASTER_ADDR = "TorchLocalMasterAddr"
HOST_NET_WITH_HEADLESS_SVC = "HostNetWithHeadlessSvc"

_DEFAULTS = {
    GANG_SCHEDULING: True,
    DAG_SCHEDULING: True,
    JOB_COORDINATOR: True,
    TORCH_LOCAL_MASTER_ADDR: True,   # single node: master addr = localhost
    HOST_NET_WITH_HEADLESS_SVC: False,
}


class FeatureGates:
    def __init__(self, overrides: dict | None = None):
        self._gates = dict(_DEFAULTS)
        if overrides:
            for k, v in overrides.items():
                if k not in self._gates:
                    raise ValueError(f"unknown feature gate: {k}")
                self._gates[k] = bool(v)

    @classmethod
    def from_flag(cls, flag: str | None) -> "FeatureGates":
        """Parse 'Gate=true,Other=false' (main.go:66 style)."""
        overrides = {}
        if flag:
            for part in flag.split(","):
                if not part.strip():
                    continue
                k, _, v = part.partition("=")
                overrides[k.strip()] = v.strip().lower() in ("1", "true", "t")
        return cls(overrides)

    def enabled(self, gate: str) -> bool:
        return self._gates[gate]

    def as_dict(self) -> dict:
        return dict(self._gates)
