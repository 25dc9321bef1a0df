from torch_on_k8s_amd.controlplane.api import TorchJob, TaskSpec, TaskType, set_defaults  # noqa: F401
