from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig  # noqa: F401
