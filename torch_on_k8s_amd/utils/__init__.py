from torch_on_k8s_amd.utils.logging import get_logger  # noqa: F401
