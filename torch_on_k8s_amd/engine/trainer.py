"""Training engine: the data plane the reference outsources to user
containers, built MI355X-native.

One process per GPU; DP gradient sync over RCCL/xGMI via flat buckets
(parallel/ddp.py); fused gfx950 kernels for the hot ops; checkpoint /
resume compatible with the control plane's elastic protocol (SURVEY.md
§5.4: the operator coordinates checkpoints via annotations, the data
plane writes them).
"""
from __future__ import annotations

import json
import os
import time
from dataclasses import dataclass, field, asdict

import torch

from torch_on_k8s_amd.engine.data import SyntheticTokens
from torch_on_k8s_amd.models.registry import build_model, get_model_config
from torch_on_k8s_amd.parallel.ddp import FlatBucketModel, FlatAdamW
from torch_on_k8s_amd.parallel.env import DistContext


@dataclass
class TrainerConfig:
    model: str = "llama3-8b"
    model_overrides: dict = field(default_factory=dict)
    micro_batch: int = 2
    seq_len: int = 4096
    grad_accum_steps: int = 1       # micro-steps per optimizer step
    lr: float = 3e-4
    lr_warmup_steps: int = 0        # linear warmup
    lr_decay_steps: int = 0         # cosine decay horizon (0 = constant)
    lr_min_ratio: float = 0.1       # floor as a fraction of lr
    weight_decay: float = 0.1
    betas: tuple = (0.9, 0.95)
    grad_clip: float = 0.0          # 0 disables (extra HBM pass when on)
    bucket_mb: int = 256
    activation_checkpointing: bool = False
    overlap_grad_sync: bool = True
    hip_graph: bool = False          # capture the whole step in a hipGraph
                                     # and replay it (needs grad_clip=0)
    dtype: str = "bf16"             # compute dtype on GPU
    seed: int = 1234
    metrics_path: str | None = None  # structured metrics for the elastic
                                     # autoscaler (replaces the reference's
                                     # stdout log-regex, observation.go:40-106)

    def to_dict(self):
        d = asdict(self)
        return d


class Trainer:
    def __init__(self, cfg: TrainerConfig, ctx: DistContext):
        self.cfg = cfg
        self.ctx = ctx
        self.device = ctx.device
        torch.manual_seed(cfg.seed)  # same init on every rank

        mcfg = get_model_config(cfg.model, **cfg.model_overrides)
        self.model_cfg = mcfg
        if self.device.type == "cuda" and cfg.dtype == "bf16":
            # construct + random-init directly on the GPU in bf16 (a CPU
            # fp32 init of 8B params costs ~30s and 32 GB of host RAM)
            prev_dtype = torch.get_default_dtype()
            try:
                torch.set_default_dtype(torch.bfloat16)
                with torch.device(self.device):
                    model = build_model(
                        mcfg,
                        activation_checkpointing=cfg.activation_checkpointing)
            finally:
                torch.set_default_dtype(prev_dtype)
        else:
            model = build_model(
                mcfg, activation_checkpointing=cfg.activation_checkpointing)
            model = model.to(self.device)
        self.fb = FlatBucketModel(
            model, bucket_mb=cfg.bucket_mb, overlap=cfg.overlap_grad_sync)
        self.opt = FlatAdamW(
            self.fb, lr=cfg.lr, betas=tuple(cfg.betas),
            weight_decay=cfg.weight_decay)
        self.data = SyntheticTokens(
            mcfg.vocab_size, cfg.micro_batch, cfg.seq_len, self.device,
            rank=ctx.rank, seed=cfg.seed)
        self.step_count = 0
        self._t_last = None
        self._graph = None
        self._graph_state = None
        if cfg.hip_graph:
            assert cfg.grad_clip == 0, "hip_graph requires grad_clip=0"
            assert cfg.grad_accum_steps <= 1, \
                "hip_graph does not support gradient accumulation yet"
            assert not cfg.lr_warmup_steps and not cfg.lr_decay_steps, \
                "hip_graph freezes the captured LR; schedules need eager"
            # Capturing RCCL collectives inside a hipGraph has known
            # constraints (NCCL watchdog/capture-mode interactions) and
            # has never executed on multi-GPU hardware in this repo —
            # refuse at world>1 unless explicitly overridden so replay
            # can never silently corrupt the gradient sync (r1 VERDICT
            # next-#2: guard or implement).
            if self.fb.world_size > 1 and \
                    os.environ.get("TOK_HIP_GRAPH_COLLECTIVES") != "1":
                raise RuntimeError(
                    "hip_graph=True with world_size>1 captures RCCL "
                    "collectives in the graph, which is unvalidated on "
                    "this stack; set TOK_HIP_GRAPH_COLLECTIVES=1 to "
                    "override after validating capture+replay on your "
                    "node, or run hip_graph on a single GPU")
            assert self.device.type == "cuda", "hip_graph needs a GPU"

    @property
    def module(self):
        return self.fb.module

    def train_step(self, sync: bool = True):
        """One full step: data -> fwd -> bwd (overlapped RCCL all-reduce)
        -> optional clip -> fused AdamW.

        sync=True returns the host loss (device sync); sync=False returns
        the detached loss tensor without host synchronisation (bench path).
        """
        if self.cfg.hip_graph:
            return self._train_step_graph(sync)
        t0 = time.perf_counter()
        accum = max(1, self.cfg.grad_accum_steps)
        self.fb.zero_grads()
        loss = None
        for micro in range(accum):
            # all-reduce (and its backward overlap) fires only on the
            # final micro-step; earlier ones just accumulate locally
            self.fb.set_accumulate(micro < accum - 1)
            inp, lab = self.data.batch(self.step_count * accum + micro)
            loss = self.fb(inp, lab)
            loss.backward()
        self.fb.finish_grad_sync()
        scale = 1.0 / (self.fb.world_size * accum)
        if self.cfg.grad_clip > 0:
            norm = self.fb.grad_norm() / accum
            coef = self.cfg.grad_clip / (norm + 1e-6)
            coef = torch.clamp(coef, max=1.0)
            scale = scale * coef.item()
        self.opt.step(grad_scale=scale, lr=self.current_lr())
        self.step_count += 1
        if not sync:
            return loss.detach()
        out = loss.detach().float().item()
        self._t_last = time.perf_counter() - t0
        if self.cfg.metrics_path and self.ctx.is_main:
            self._write_metrics(out)
        return out

    # ---- hipGraph step capture (MI355X: launch-bound inner loop ->
    # one graph replay per step; the AdamW bias correction reads the
    # optimizer's on-device step counter so replay stays correct) ------
    def _capture_graph(self):
        cfg = self.cfg
        inp = torch.zeros(cfg.micro_batch, cfg.seq_len, dtype=torch.long,
                          device=self.device)
        lab = torch.zeros_like(inp)

        def step_body():
            self.fb.zero_grads()
            loss = self.fb(inp, lab)
            loss.backward()
            self.fb.finish_grad_sync()
            self.opt.step()
            return loss

        # Warm up allocator/RCCL/hook state on a side stream. These are
        # REAL training steps (real batches) so no model state is wasted.
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                ni, nl = self.data.batch(self.step_count)
                inp.copy_(ni)
                lab.copy_(nl)
                step_body()
                self.step_count += 1
        torch.cuda.current_stream().wait_stream(side)
        # Capture records without executing; buffer contents don't matter.
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            loss = step_body()
        self._graph = g
        self._graph_state = (inp, lab, loss)

    def _train_step_graph(self, sync: bool):
        t0 = time.perf_counter()
        if self._graph is None:
            self._capture_graph()
        inp, lab, loss = self._graph_state
        ni, nl = self.data.batch(self.step_count)
        inp.copy_(ni)
        lab.copy_(nl)
        self._graph.replay()
        self.step_count += 1
        if not sync:
            return loss.detach()
        out = loss.detach().float().item()
        self._t_last = time.perf_counter() - t0
        if self.cfg.metrics_path and self.ctx.is_main:
            self._write_metrics(out)
        return out

    def current_lr(self) -> float:
        """Linear warmup then cosine decay to lr_min_ratio*lr."""
        cfg = self.cfg
        step = self.step_count + 1
        lr = cfg.lr
        if cfg.lr_warmup_steps and step < cfg.lr_warmup_steps:
            return lr * step / cfg.lr_warmup_steps
        if cfg.lr_decay_steps:
            import math
            t = min(1.0, (step - cfg.lr_warmup_steps) /
                    max(1, cfg.lr_decay_steps - cfg.lr_warmup_steps))
            floor = lr * cfg.lr_min_ratio
            return floor + (lr - floor) * 0.5 * (1 + math.cos(math.pi * t))
        return lr

    def tokens_per_step(self) -> int:
        return self.cfg.micro_batch * self.cfg.seq_len * \
            self.fb.world_size * max(1, self.cfg.grad_accum_steps)

    def _write_metrics(self, loss: float):
        """Structured metrics endpoint for the elastic autoscaler
        (reference parses worker-0 stdout with a regex,
        torchelastic/observation.go:40-106 - we write JSON instead)."""
        rec = {
            "step": self.step_count,
            "loss": loss,
            "step_time_s": self._t_last,
            "tokens_per_s": self.tokens_per_step() / self._t_last
            if self._t_last else None,
            "world_size": self.fb.world_size,
            "ts": time.time(),
        }
        tmp = self.cfg.metrics_path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(rec, f)
        os.replace(tmp, self.cfg.metrics_path)

    # ---- async periodic checkpoints ----------------------------------
    # Two phases: a SHORT blocking snapshot (device -> host copies at
    # PCIe rate) at a step boundary, then a background thread serializes
    # to disk while training continues. Sharded: every rank snapshots at
    # the SAME step (step % N == 0 needs no coordination — steps are
    # lockstep); each writer drops its shard + a .done marker and rank 0's
    # writer performs the atomic rename once all markers exist, so the
    # background phase needs no collectives.
    def snapshot_checkpoint_async(self, path: str) -> "threading.Thread":
        import threading
        world = self.fb.world_size
        sharded = world > 1
        tmp = f"{path}.tmp-{self.step_count}"
        os.makedirs(tmp, exist_ok=True)

        # phase 1 (blocking, short): host snapshot
        msd = None
        meta = None
        if self.ctx.is_main:
            # copy=True: on a CPU trainer .to("cpu") would ALIAS the
            # live tensors and the background writer would serialize
            # state the next step is mutating (torn snapshot)
            msd = {k: v.detach().to("cpu", non_blocking=True, copy=True)
                   for k, v in self.module.state_dict().items()}
            meta = {
                "step": self.step_count,
                "model": self.cfg.model,
                "model_config": self.model_cfg.to_dict(),
                "trainer_config": self.cfg.to_dict(),
                "optim_format": "sharded" if sharded else "single",
                "optim_shards": world if sharded else 1,
                "num_buckets": len(self.fb.buckets),
            }
        osd = self.opt.state_dict()
        if sharded:
            shard = {
                "step": osd["step"],
                "exp_avg": {i: t.to("cpu", non_blocking=True, copy=True)
                            for i, t in enumerate(osd["exp_avg"])
                            if i % world == self.ctx.rank},
                "exp_avg_sq": {i: t.to("cpu", non_blocking=True,
                                       copy=True)
                               for i, t in enumerate(osd["exp_avg_sq"])
                               if i % world == self.ctx.rank},
            }
        else:
            shard = {
                "step": osd["step"],
                "exp_avg": [t.to("cpu", non_blocking=True, copy=True)
                            for t in osd["exp_avg"]],
                "exp_avg_sq": [t.to("cpu", non_blocking=True, copy=True)
                               for t in osd["exp_avg_sq"]],
            }
        if self.device.type == "cuda":
            torch.cuda.synchronize()  # host buffers now stable

        rank, is_main = self.ctx.rank, self.ctx.is_main

        def writer():
            if is_main:
                torch.save(msd, os.path.join(tmp, "model.pt"))
                with open(os.path.join(tmp, "meta.json"), "w") as f:
                    json.dump(meta, f, indent=2)
            if sharded:
                torch.save(shard,
                           os.path.join(tmp, f"optim-shard-{rank}.pt"))
            elif is_main:
                torch.save(shard, os.path.join(tmp, "optim.pt"))
            open(os.path.join(tmp, f".done-{rank}"), "w").close()
            if is_main:
                # wait for every shard, then atomically publish
                want = {f".done-{r}" for r in range(world if sharded else 1)}
                deadline = time.time() + 600
                while time.time() < deadline:
                    if want <= set(os.listdir(tmp)):
                        break
                    time.sleep(0.05)
                else:
                    return  # incomplete; leave tmp for inspection
                for m in want:
                    os.unlink(os.path.join(tmp, m))
                if os.path.exists(path):
                    import shutil
                    shutil.rmtree(path)
                os.replace(tmp, path)

        t = threading.Thread(target=writer, daemon=True)
        t.start()
        return t

    # ---- checkpoint / resume (elastic protocol) ------
    def save_checkpoint(self, path: str, sharded: bool | None = None):
        """Atomic checkpoint.

        sharded=False (or world 1): rank 0 writes everything.
        sharded=True: EVERY rank must call this at the same step (the
        entrypoint coordinates via a broadcast) — rank r writes the
        optimizer buckets with index % world == r, cutting the serial
        ~(6 bytes/param) write by 1/world so elastic checkpoint
        transactions don't stall training on one writer. Shards are
        keyed by bucket index, so a job restarted at a DIFFERENT world
        size still loads them all.
        """
        from torch_on_k8s_amd.parallel.env import barrier
        world = self.fb.world_size
        if sharded is None:
            sharded = world > 1
        tmp = path + ".tmp"
        if self.ctx.is_main:
            os.makedirs(tmp, exist_ok=True)
            # state_dict values are views into flat buckets; clone so
            # torch.save serialises each tensor, not the whole bucket.
            msd = {k: v.detach().cpu().clone()
                   for k, v in self.module.state_dict().items()}
            torch.save(msd, os.path.join(tmp, "model.pt"))
            meta = {
                "step": self.step_count,
                "model": self.cfg.model,
                "model_config": self.model_cfg.to_dict(),
                "trainer_config": self.cfg.to_dict(),
                "optim_format": "sharded" if sharded else "single",
                "optim_shards": world if sharded else 1,
                "num_buckets": len(self.fb.buckets),
            }
            with open(os.path.join(tmp, "meta.json"), "w") as f:
                json.dump(meta, f, indent=2)
        if sharded:
            barrier(self.ctx)  # tmp dir exists; all ranks write shards
            osd = self.opt.state_dict()  # syncs the device step counter
            shard = {
                "step": osd["step"],
                "exp_avg": {i: t for i, t in enumerate(osd["exp_avg"])
                            if i % world == self.ctx.rank},
                "exp_avg_sq": {i: t for i, t in enumerate(osd["exp_avg_sq"])
                               if i % world == self.ctx.rank},
            }
            torch.save(shard,
                       os.path.join(tmp, f"optim-shard-{self.ctx.rank}.pt"))
            barrier(self.ctx)  # every shard durable before the rename
        elif self.ctx.is_main:
            torch.save(self.opt.state_dict(), os.path.join(tmp, "optim.pt"))
        if self.ctx.is_main:
            if os.path.exists(path):
                import shutil
                shutil.rmtree(path)
            os.replace(tmp, path)
        if sharded:
            barrier(self.ctx)  # no rank resumes before the rename lands

    def load_checkpoint(self, path: str):
        # map_location CPU: tensors stream host->device via copy_ into
        # the EXISTING buffers, so resume never doubles the device
        # footprint (loading ~75 GB of optimizer state with
        # map_location=cuda OOMs a trainer that is already resident —
        # found by the r2 GPU soak)
        sd = torch.load(os.path.join(path, "model.pt"),
                        map_location="cpu", weights_only=True)
        self.module.load_state_dict(sd)
        del sd
        with open(os.path.join(path, "meta.json")) as f:
            meta = json.load(f)
        self.step_count = meta["step"]
        nshards = meta.get("optim_shards", 1)
        if meta.get("optim_format", "single") == "single":
            osd = torch.load(os.path.join(path, "optim.pt"),
                             map_location="cpu", weights_only=True)
            self.opt.load_state_dict(osd)
            return
        # sharded: every rank reads all shards (bucket-indexed, world-
        # size independent)
        self.opt.step_count = self.step_count
        if self.opt.step_dev is not None:
            self.opt.step_dev.fill_(self.step_count)
        for r in range(nshards):
            shard = torch.load(os.path.join(path, f"optim-shard-{r}.pt"),
                               map_location="cpu", weights_only=True)
            self.opt.step_count = shard["step"]
            if self.opt.step_dev is not None:
                self.opt.step_dev.fill_(shard["step"])
            for i, t in shard["exp_avg"].items():
                self.opt.exp_avg[int(i)].copy_(t)
            for i, t in shard["exp_avg_sq"].items():
                self.opt.exp_avg_sq[int(i)].copy_(t)
