"""Model serving: load a packaged ModelVersion (OCI artifact) or a raw
checkpoint and serve generation over HTTP.

The reference stops at packaging (ModelVersion image pushed by Kaniko,
modelversion_controller.go:286-406) — consumption is left to the user.
This closes the loop node-natively: the same OCI artifact the registry
builds is loadable here, and decode runs the MI355X serving path
(KV-cache + flash-decode kernel + self-feeding hipGraph,
models/llama.py generate()).

  # from a trained job's packaged version:
  python -m torch_on_k8s_amd.serve --workdir /var/run/torch-on-k8s-amd \
      --version my-model:mv-llama-dp8-00042 --port 8500
  # or from a checkpoint dir / random-init preset:
  python -m torch_on_k8s_amd.serve --ckpt /path/to/ckpt --port 8500
  python -m torch_on_k8s_amd.serve --preset llama-1b --port 8500

Endpoints:
  GET  /healthz                -> {"status": "ok", "model": ...}
  GET  /v1/models              -> loaded model metadata
  POST /v1/generate            -> {"prompt_ids": [[int,...],...],
                                   "max_new_tokens": 32,
                                   "temperature": 0.0}
                                  returns {"output_ids": [[...]], ...}

Prompts are token IDs: this framework is tokenizer-free by design
(synthetic-data training; no network for vocab downloads).

NOTE: no `from __future__ import annotations` here — FastAPI resolves
the request model from runtime annotations, and stringified annotations
of a function-local class break that resolution.
"""
import argparse
import json
import os
import threading
import time


def load_from_checkpoint(ckpt_dir: str, device, dtype=None):
    """Build the model recorded in a trainer checkpoint (meta.json
    carries the family preset + full config dict) and load weights."""
    import torch
    from torch_on_k8s_amd.models.registry import (build_model,
                                                  get_model_config)
    with open(os.path.join(ckpt_dir, "meta.json")) as f:
        meta = json.load(f)
    base = get_model_config(meta["model"])
    cfg = type(base)(**meta["model_config"])
    model = build_model(cfg)
    sd = torch.load(os.path.join(ckpt_dir, "model.pt"),
                    map_location="cpu", weights_only=True)
    model.load_state_dict(sd)
    if dtype is not None:
        model = model.to(dtype)
    model = model.to(device).eval()
    return model, meta


class InferenceServer:
    """One resident model + a lock (decode owns the GPU; requests are
    serialized — batch inside a request for throughput)."""

    def __init__(self, model, meta: dict):
        self.model = model
        self.meta = meta
        self.lock = threading.Lock()
        self.requests = 0
        self.tokens_out = 0

    @classmethod
    def from_version(cls, workdir: str, ref: str, device, dtype=None):
        """Load 'model:version' from the manager's model registry
        (extracts the OCI artifact; the final checkpoint lives at
        <rootfs>/final inside the layer)."""
        import tempfile
        from torch_on_k8s_amd.controlplane.modelregistry import (
            ModelRegistry, StorageProvider)
        model_name, _, version = ref.partition(":")
        reg = ModelRegistry(StorageProvider(os.path.join(workdir, "models")))
        # re-index existing artifacts on disk (daemon restart case)
        reg.reindex()
        dest = tempfile.mkdtemp(prefix="tok-serve-")
        root = reg.extract(model_name, version, dest)
        ckpt = os.path.join(root, "final")
        model, meta = load_from_checkpoint(ckpt, device, dtype)
        meta = dict(meta, model_version=ref)
        return cls(model, meta)

    @classmethod
    def from_checkpoint(cls, ckpt_dir: str, device, dtype=None):
        model, meta = load_from_checkpoint(ckpt_dir, device, dtype)
        return cls(model, meta)

    @classmethod
    def from_preset(cls, preset: str, device, dtype=None):
        from torch_on_k8s_amd.models.registry import (build_model,
                                                      get_model_config)
        cfg = get_model_config(preset)
        model = build_model(cfg)
        if dtype is not None:
            model = model.to(dtype)
        model = model.to(device).eval()
        return cls(model, {"model": preset, "model_config": cfg.to_dict(),
                           "step": 0})

    def generate(self, prompt_ids, max_new_tokens: int = 32,
                 temperature: float = 0.0):
        import torch
        dev = next(self.model.parameters()).device
        inp = torch.tensor(prompt_ids, dtype=torch.long, device=dev)
        t0 = time.perf_counter()
        with self.lock, torch.no_grad():
            out = self.model.generate(inp, max_new_tokens=max_new_tokens,
                                      temperature=temperature)
        dt = time.perf_counter() - t0
        new = out[:, inp.shape[1]:]
        self.requests += 1
        self.tokens_out += new.numel()
        return {
            "output_ids": out.tolist(),
            "new_ids": new.tolist(),
            "new_tokens": int(new.numel()),
            "latency_s": dt,
            "tokens_per_s": new.numel() / dt if dt > 0 else None,
        }


def build_app(srv: InferenceServer):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel, Field

    class GenerateRequest(BaseModel):
        prompt_ids: list[list[int]]
        max_new_tokens: int = Field(default=32, ge=1, le=4096)
        temperature: float = Field(default=0.0, ge=0.0, le=10.0)

    app = FastAPI(title="torch-on-k8s-amd serving")

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "model": srv.meta.get("model"),
                "requests": srv.requests, "tokens_out": srv.tokens_out}

    @app.get("/v1/models")
    def models():
        return {"model": srv.meta.get("model"),
                "model_version": srv.meta.get("model_version"),
                "trained_steps": srv.meta.get("step"),
                "config": srv.meta.get("model_config")}

    @app.get("/metrics")
    def metrics():
        """Prometheus exposition (serving-side observability)."""
        from fastapi.responses import PlainTextResponse
        lines = [
            "# TYPE tok_serve_requests_total counter",
            f"tok_serve_requests_total {srv.requests}",
            "# TYPE tok_serve_tokens_out_total counter",
            f"tok_serve_tokens_out_total {srv.tokens_out}",
        ]
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.post("/v1/generate")
    def generate(req: GenerateRequest):
        vocab = srv.meta.get("model_config", {}).get("vocab_size")
        if not req.prompt_ids or not all(req.prompt_ids):
            raise HTTPException(400, "prompt_ids must be non-empty")
        if len({len(p) for p in req.prompt_ids}) != 1:
            raise HTTPException(400, "all prompts must share one length")
        if vocab and any(t < 0 or t >= vocab
                         for p in req.prompt_ids for t in p):
            raise HTTPException(400, f"token id out of range [0,{vocab})")
        max_seq = srv.meta.get("model_config", {}).get("max_seq_len")
        if max_seq and len(req.prompt_ids[0]) + req.max_new_tokens > max_seq:
            raise HTTPException(
                400, f"prompt+max_new_tokens exceeds the model's trained "
                     f"context ({max_seq}); RoPE extrapolation beyond it "
                     f"degrades silently")
        return srv.generate(req.prompt_ids, req.max_new_tokens,
                            req.temperature)

    return app


def main():
    import torch
    ap = argparse.ArgumentParser(prog="torch-on-k8s-amd-serve")
    src = ap.add_mutually_exclusive_group(required=True)
    src.add_argument("--version", help="model:version from --workdir")
    src.add_argument("--ckpt", help="checkpoint directory")
    src.add_argument("--preset", help="random-init preset (smoke/dev)")
    ap.add_argument("--workdir", default="/tmp/torch-on-k8s-amd")
    ap.add_argument("--port", type=int, default=8500)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    args = ap.parse_args()

    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if (args.dtype == "bf16" and device == "cuda") \
        else None
    if args.version:
        srv = InferenceServer.from_version(args.workdir, args.version,
                                           device, dtype)
    elif args.ckpt:
        srv = InferenceServer.from_checkpoint(args.ckpt, device, dtype)
    else:
        srv = InferenceServer.from_preset(args.preset, device, dtype)
    import uvicorn
    uvicorn.run(build_app(srv), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
