"""torch-on-k8s-amd: an MI355X-native distributed training framework.

Brand-new construction with the capability set of hliangzhao/torch-on-k8s
(see SURVEY.md): a node-level control plane (TorchJob semantics: gang/DAG
scheduling, WRR coordinator queues, failover, elastic scaling, checkpoint
coordination, model packaging) plus an MI355X-native data plane
(PyTorch-ROCm + handwritten gfx950 HIP kernels + RCCL over xGMI).
"""
__version__ = "0.1.0"
