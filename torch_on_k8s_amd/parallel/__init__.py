from torch_on_k8s_amd.parallel.env import DistContext, init_distributed  # noqa: F401
from torch_on_k8s_amd.parallel.ddp import FlatBucketModel, FlatAdamW  # noqa: F401
