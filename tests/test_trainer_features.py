"""Trainer features: gradient accumulation + LR schedule."""
import torch

from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
from torch_on_k8s_amd.parallel.env import DistContext


def test_grad_accum_sums_micro_grads():
    torch.manual_seed(0)
    ctx = DistContext()
    tr = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                               seq_len=32, grad_accum_steps=2), ctx)
    # manual: two micro backwards accumulated
    tr.fb.zero_grads()
    for micro in range(2):
        inp, lab = tr.data.batch(micro)
        tr.fb(inp, lab).backward()
    ref = [b.flat_grad.clone() for b in tr.fb.buckets]

    # trainer step does the same micro sequence (step_count=0 -> batches 0,1)
    torch.manual_seed(0)
    tr2 = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                                seq_len=32, grad_accum_steps=2), ctx)
    tr2.train_step()
    # grads were consumed by the optimizer but flat_grad still holds the
    # accumulated sum (scale folded into the kernel)
    for b, r in zip(tr2.fb.buckets, ref):
        assert torch.allclose(b.flat_grad, r, atol=1e-6)
    assert tr2.tokens_per_step() == 2 * 32  # accum counted in throughput


def test_lr_schedule_warmup_cosine():
    ctx = DistContext()
    tr = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1, seq_len=32,
                               lr=1.0, lr_warmup_steps=10,
                               lr_decay_steps=110, lr_min_ratio=0.1), ctx)
    tr.step_count = 0
    assert abs(tr.current_lr() - 0.1) < 1e-6          # step 1 of warmup
    tr.step_count = 4
    assert abs(tr.current_lr() - 0.5) < 1e-6
    tr.step_count = 9
    assert abs(tr.current_lr() - 1.0) < 1e-6          # warmup done
    tr.step_count = 200
    assert abs(tr.current_lr() - 0.1) < 1e-6          # decayed to floor
    # midpoint of cosine: (floor + lr)/2
    tr.step_count = 59                                 # t = 50/100
    assert abs(tr.current_lr() - 0.55) < 1e-2


def test_sharded_checkpoint_roundtrip(tmp_path):
    """Sharded save (bucket-indexed) must reload into any world size.
    Simulated single-process: save world-1 style shards manually via the
    API with sharded semantics."""
    import os
    from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
    from torch_on_k8s_amd.parallel.env import DistContext
    torch.manual_seed(0)
    cfg = TrainerConfig(model="llama-tiny", micro_batch=1, seq_len=32)
    tr = Trainer(cfg, DistContext())
    tr.train_step()
    tr.train_step()
    ck = str(tmp_path / "ck")
    tr.save_checkpoint(ck, sharded=True)  # world 1: one shard
    assert os.path.exists(os.path.join(ck, "optim-shard-0.pt"))

    tr2 = Trainer(cfg, DistContext())
    tr2.load_checkpoint(ck)
    assert tr2.step_count == 2
    assert tr2.opt.step_count == 2
    for a, b in zip(tr.opt.exp_avg, tr2.opt.exp_avg):
        assert torch.equal(a, b)
    # resumed training continues identically
    l1 = tr.train_step()
    l2 = tr2.train_step()
    assert abs(l1 - l2) < 1e-6


def test_async_snapshot_resumable(tmp_path):
    """snapshot_checkpoint_async: training continues while the writer
    serializes; the published checkpoint is complete, atomic, and
    resumable at the snapshot step."""
    import torch
    from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
    from torch_on_k8s_amd.parallel.env import DistContext
    torch.manual_seed(0)
    tr = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                               seq_len=32), DistContext())
    tr.train_step()
    tr.train_step()
    ck = str(tmp_path / "ckpt")
    t = tr.snapshot_checkpoint_async(ck)
    # training continues immediately (mutates weights AFTER the snapshot)
    tr.train_step()
    t.join(timeout=120)
    assert not t.is_alive()
    import os as _os
    assert _os.path.exists(_os.path.join(ck, "meta.json"))
    assert not any(".tmp-" in d for d in _os.listdir(str(tmp_path)))

    torch.manual_seed(0)
    tr2 = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                                seq_len=32), DistContext())
    tr2.load_checkpoint(ck)
    assert tr2.step_count == 2  # the snapshot step, not the later one
    # weights equal the state at snapshot time: one more step from the
    # restore must match a fresh run's third step bit-for-bit
    torch.manual_seed(0)
    ref = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                                seq_len=32), DistContext())
    ref.train_step()
    ref.train_step()
    ref.train_step()
    tr2.train_step()
    for b2, br in zip(tr2.fb.buckets, ref.fb.buckets):
        assert torch.equal(b2.flat_param, br.flat_param)


def test_lr_schedule_warmup_and_cosine_floor():
    """current_lr: linear warmup then cosine to the floor."""
    import torch
    from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
    from torch_on_k8s_amd.parallel.env import DistContext
    torch.manual_seed(0)
    tr = Trainer(TrainerConfig(model="llama-tiny", micro_batch=1,
                               seq_len=32, lr=1e-2, lr_warmup_steps=4,
                               lr_decay_steps=10, lr_min_ratio=0.1),
                 DistContext())
    lrs = []
    for step in range(12):
        tr.step_count = step
        lrs.append(tr.current_lr())
    # warmup strictly increasing to peak
    assert lrs[0] < lrs[1] < lrs[2] < lrs[3]
    assert abs(lrs[3] - 1e-2) < 1e-9
    # decay monotonically down to the floor
    assert all(a >= b - 1e-12 for a, b in zip(lrs[3:], lrs[4:]))
    assert abs(lrs[-1] - 1e-3) < 1e-4  # floor = lr * min_ratio


def test_grad_clip_coefficient_folded_into_adamw(tmp_path):
    """grad_clip folds the clip coefficient into the fused AdamW
    grad_scale (one less HBM pass): the scale passed to opt.step must
    equal 1/(world*accum) * min(1, clip/(norm_of_avg_grads + 1e-6))."""
    import pytest
    import torch
    from torch_on_k8s_amd.engine.trainer import Trainer, TrainerConfig
    from torch_on_k8s_amd.parallel.env import DistContext

    cfg = TrainerConfig(model="llama-tiny", micro_batch=1, seq_len=32,
                        dtype="fp32", grad_clip=0.5)
    ctx = DistContext(device=torch.device("cpu"))
    tr = Trainer(cfg, ctx)

    captured = {}
    real_step = tr.opt.step

    def spy(grad_scale=None, lr=None):
        captured["scale"] = grad_scale
        return real_step(grad_scale=grad_scale, lr=lr)

    tr.opt.step = spy
    tr.train_step()
    # recompute from the post-backward grads (opt.step hasn't zeroed them)
    total = torch.zeros(())
    for b in tr.fb.buckets:
        total += b.flat_grad.float().pow(2).sum()
    norm = total.sqrt().item()  # world=1, accum=1: sum == avg
    expected = min(1.0, 0.5 / (norm + 1e-6))
    assert captured["scale"] == pytest.approx(expected, rel=1e-5)

    # clip large enough to be inactive -> plain 1/(world*accum)
    cfg2 = TrainerConfig(model="llama-tiny", micro_batch=1, seq_len=32,
                         dtype="fp32", grad_clip=1e9)
    tr2 = Trainer(cfg2, ctx)
    captured2 = {}
    real2 = tr2.opt.step
    tr2.opt.step = lambda grad_scale=None, lr=None: (
        captured2.update(scale=grad_scale), real2(grad_scale=grad_scale,
                                                  lr=lr))[1]
    tr2.train_step()
    assert captured2["scale"] == pytest.approx(1.0)
