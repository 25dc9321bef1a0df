from torch_on_k8s_amd.models.llama import LlamaConfig, LlamaModel, get_config, PRESETS  # noqa: F401
