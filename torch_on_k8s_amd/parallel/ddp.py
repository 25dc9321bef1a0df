"""Flat-bucket data parallelism over RCCL/xGMI.

MI355X-first design: instead of per-parameter
all-reduce or torch DDP's bucket views, every parameter lives inside a
large flat buffer ("bucket"):

  * parameters  -> views into flat bf16 param buffers,
  * gradients   -> views into flat bf16 grad buffers (autograd
                   accumulates in place), so the DP all-reduce is a
                   single RCCL call per bucket on the already-contiguous
                   buffer - zero copy, few large collectives, which is
                   what the per-link-bound xGMI topology wants
                   (7 p2p links x ~153 GB/s; small collectives are
                   latency-bound),
  * the fused AdamW kernel updates one whole bucket per launch, and the
    1/world_size gradient average is folded into its grad read.

Overlap: buckets are formed in reverse parameter order; a bucket's
all-reduce is launched (async, on RCCL's comm stream) as soon as the
last gradient of that bucket is accumulated, overlapping the rest of
backward. The reference delegates all of this to "a PyTorch container"
(SURVEY.md §0); this module is the MI355X-native replacement.
"""
from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.distributed as dist

ALIGN = 64  # elements; keeps every param view 128-byte aligned


@dataclass
class _Seg:
    param: torch.nn.Parameter
    offset: int
    numel: int


@dataclass
class Bucket:
    index: int
    decay: bool
    segs: list = field(default_factory=list)
    numel: int = 0
    flat_param: torch.Tensor | None = None
    flat_grad: torch.Tensor | None = None
    pending: int = 0
    work: object | None = None


def _pad(n: int) -> int:
    return (n + ALIGN - 1) // ALIGN * ALIGN


class FlatBucketModel:
    """Wraps a module: flattens params/grads into buckets and (optionally)
    overlaps gradient all-reduce with backward."""

    def __init__(self, module: torch.nn.Module, *, bucket_mb: int = 256,
                 process_group=None, overlap: bool = True,
                 no_decay_keywords=("norm", "bias")):
        self.module = module
        self.group = process_group
        self.overlap = overlap
        self.world_size = (dist.get_world_size(process_group)
                           if dist.is_initialized() else 1)

        # Partition params (deduped, reverse order ~= backward completion
        # order) into homogeneous (dtype, decay) buckets.
        seen = set()
        params: list[tuple[str, torch.nn.Parameter]] = []
        for name, p in module.named_parameters():
            if p.requires_grad and id(p) not in seen:
                seen.add(id(p))
                params.append((name, p))
        params.reverse()

        bucket_bytes = bucket_mb * (1 << 20)
        self.buckets: list[Bucket] = []
        open_buckets: dict[tuple, Bucket] = {}
        self._param_bucket: dict[int, Bucket] = {}
        for name, p in params:
            decay = not any(k in name.lower() for k in no_decay_keywords) \
                and p.dim() >= 2
            key = (p.dtype, decay)
            b = open_buckets.get(key)
            esize = p.element_size()
            if b is None or (b.numel + _pad(p.numel())) * esize > bucket_bytes:
                b = Bucket(index=len(self.buckets), decay=decay)
                b.dtype = p.dtype  # type: ignore[attr-defined]
                self.buckets.append(b)
                open_buckets[key] = b
            b.segs.append(_Seg(p, b.numel, p.numel()))
            b.numel += _pad(p.numel())
            self._param_bucket[id(p)] = b

        # Materialise flat buffers and re-point params/grads.
        dev = params[0][1].device if params else torch.device("cpu")
        for b in self.buckets:
            dtype = b.segs[0].param.dtype
            b.flat_param = torch.zeros(b.numel, dtype=dtype, device=dev)
            b.flat_grad = torch.zeros(b.numel, dtype=dtype, device=dev)
            for s in b.segs:
                with torch.no_grad():
                    b.flat_param[s.offset:s.offset + s.numel].copy_(
                        s.param.data.reshape(-1))
                s.param.data = b.flat_param[s.offset:s.offset + s.numel] \
                    .view(s.param.shape)
                s.param.grad = b.flat_grad[s.offset:s.offset + s.numel] \
                    .view(s.param.shape)
            b.pending = len(b.segs)

        # gradient accumulation: while True, backward only accumulates
        # into the flat grads; the all-reduce fires on the final
        # micro-step (set_accumulate(False) before it)
        self.accumulate_only = False
        self._hooks = []
        if self.overlap and self.world_size > 1:
            self._register_hooks()

    def _register_hooks(self):
        for b in self.buckets:
            for s in b.segs:
                h = s.param.register_post_accumulate_grad_hook(
                    self._make_hook(b))
                self._hooks.append(h)

    def set_world(self, world_size: int):
        """Elastic fast-rejoin: the process group was re-initialized at a
        new world size with model/optimizer state kept resident. Hooks
        self-guard on world_size/is_initialized, so growing from a
        world-1 start just needs them registered."""
        self.world_size = world_size
        if self.overlap and world_size > 1 and not self._hooks:
            self._register_hooks()

    def _make_hook(self, bucket: Bucket):
        def hook(_param):
            bucket.pending -= 1
            if bucket.pending == 0:
                bucket.pending = len(bucket.segs)
                if not self.accumulate_only and self.world_size > 1 and \
                        dist.is_initialized():
                    bucket.work = dist.all_reduce(
                        bucket.flat_grad, group=self.group, async_op=True)
        return hook

    def set_accumulate(self, accumulate: bool):
        self.accumulate_only = accumulate

    # -- training-step API --------------------------------------------
    def zero_grads(self):
        for b in self.buckets:
            b.flat_grad.zero_()
            b.pending = len(b.segs)
            b.work = None

    def finish_grad_sync(self):
        """Wait for overlapped all-reduces (or run them now if overlap
        is off). After this, flat_grad holds the SUM over ranks."""
        if self.world_size <= 1:
            return
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            else:
                # The hook never launched this bucket's all-reduce this
                # step: overlap off, frozen params leaving pending > 0,
                # or grad accumulation where the final micro-step touched
                # none of its params (grads from earlier micro-steps
                # still need syncing). Reducing every un-reduced bucket
                # keeps the collective count identical on all ranks
                # (conditional compute can make 'did params get grads'
                # rank-dependent).
                dist.all_reduce(b.flat_grad, group=self.group)
            b.work = None

    def grad_norm(self) -> torch.Tensor:
        """Global grad L2 norm of the AVERAGED grads (flat_grad holds the
        rank-sum after finish_grad_sync)."""
        dev = self.buckets[0].flat_grad.device
        total = torch.zeros((), dtype=torch.float32, device=dev)
        for b in self.buckets:
            total += b.flat_grad.float().pow(2).sum()
        return total.sqrt() / self.world_size

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    __call__ = forward


class FlatAdamW:
    """AdamW over flat buckets via the fused gfx950 kernel.

    The DP gradient average (1/world) and any grad-clip coefficient are
    folded into the kernel's grad read (one less pass over HBM).
    """

    def __init__(self, fb: FlatBucketModel, *, lr: float = 3e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.1):
        self.fb = fb
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.exp_avg = []
        self.exp_avg_sq = []
        dev = fb.buckets[0].flat_param.device if fb.buckets else "cpu"
        for b in fb.buckets:
            self.exp_avg.append(torch.zeros(
                b.numel, dtype=torch.float32, device=b.flat_param.device))
            self.exp_avg_sq.append(torch.zeros(
                b.numel, dtype=torch.float32, device=b.flat_param.device))
        # device-side step counter: under hipGraph replay the host step
        # is frozen into the capture, so the kernel reads this instead
        self.step_dev = (torch.zeros((), dtype=torch.int32, device=dev)
                         if str(dev).startswith("cuda") else None)

    @torch.no_grad()
    def step(self, grad_scale: float | None = None, lr: float | None = None):
        from torch_on_k8s_amd import ops
        self.step_count += 1
        if self.step_dev is not None:
            self.step_dev += 1  # capturable; source of truth under replay
        if grad_scale is None:
            grad_scale = 1.0 / self.fb.world_size
        for i, b in enumerate(self.fb.buckets):
            ops.fused_adamw_(
                b.flat_param, b.flat_grad, self.exp_avg[i],
                self.exp_avg_sq[i],
                lr=(lr if lr is not None else self.lr),
                beta1=self.betas[0], beta2=self.betas[1], eps=self.eps,
                weight_decay=self.weight_decay if b.decay else 0.0,
                step=self.step_count, grad_scale=grad_scale,
                step_dev=self.step_dev)

    def state_dict(self):
        if self.step_dev is not None:
            # under hipGraph replay the device counter is authoritative
            self.step_count = int(self.step_dev.item())
        return {
            "step": self.step_count,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        if self.step_dev is not None:
            self.step_dev.fill_(self.step_count)
        for dst, src in zip(self.exp_avg, sd["exp_avg"]):
            dst.copy_(src)
        for dst, src in zip(self.exp_avg_sq, sd["exp_avg_sq"]):
            dst.copy_(src)
