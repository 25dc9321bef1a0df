#!/usr/bin/env python3
"""Attention kernel microbenchmark (dev tool, not the driver contract).

Measures fwd / bwd wall time and effective TFLOP/s of the gfx950 flash
attention at the flagship shape, A/B against torch SDPA. Within-process
interleaved rounds (guide §5.4 rule 24).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def flops(B, S, Hq, D, causal, bwd=False):
    f = 4 * B * Hq * S * S * D
    if causal:
        f //= 2
    return f * (2.5 if bwd else 1.0)


def time_fn(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--B", type=int, default=2)
    ap.add_argument("--S", type=int, default=4096)
    ap.add_argument("--Hq", type=int, default=32)
    ap.add_argument("--Hkv", type=int, default=8)
    ap.add_argument("--D", type=int, default=128)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--rounds", type=int, default=3)
    args = ap.parse_args()
    from torch_on_k8s_amd import ops

    B, S, Hq, Hkv, D = args.B, args.S, args.Hq, args.Hkv, args.D
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=dev)
    do = torch.randn(B, S, Hq, D, dtype=torch.bfloat16, device=dev)
    rep = Hq // Hkv
    qt = q.transpose(1, 2).contiguous()
    kt = k.transpose(1, 2).repeat_interleave(rep, dim=1).contiguous()
    vt = v.transpose(1, 2).repeat_interleave(rep, dim=1).contiguous()
    dot = do.transpose(1, 2).contiguous()

    ffw = flops(B, S, Hq, D, True)
    fbw = flops(B, S, Hq, D, True, bwd=True)

    def hip_fwd():
        o, lse = ops._C.attn_fwd(q, k, v, True)
        return o, lse

    o_, lse_ = hip_fwd()

    def hip_bwd():
        return ops._C.attn_bwd(q, k, v, o_, lse_, do, True)

    qs = qt.detach().requires_grad_(True)
    ks = kt.detach().requires_grad_(True)
    vs = vt.detach().requires_grad_(True)

    def sdpa_fwd():
        return torch.nn.functional.scaled_dot_product_attention(
            qt, kt, vt, is_causal=True)

    def sdpa_fwdbwd():
        o = torch.nn.functional.scaled_dot_product_attention(
            qs, ks, vs, is_causal=True)
        o.backward(dot)
        qs.grad = ks.grad = vs.grad = None

    for r in range(args.rounds):
        th_f = time_fn(hip_fwd, args.iters)
        th_b = time_fn(hip_bwd, args.iters)
        ts_f = time_fn(sdpa_fwd, args.iters)
        ts_fb = time_fn(sdpa_fwdbwd, args.iters)
        print(f"[round {r}] hip fwd {th_f*1e3:7.2f} ms {ffw/th_f/1e12:7.1f} TF"
              f" | hip bwd {th_b*1e3:7.2f} ms {fbw/th_b/1e12:7.1f} TF"
              f" | sdpa fwd {ts_f*1e3:7.2f} ms {ffw/ts_f/1e12:7.1f} TF"
              f" | sdpa f+b {ts_fb*1e3:7.2f} ms "
              f"{(ffw+fbw)/ts_fb/1e12:7.1f} TF", flush=True)

    # correctness spot-check vs sdpa
    o_ref = sdpa_fwd().transpose(1, 2)
    err = (o_.float() - o_ref.float()).abs().max().item()
    print(f"max |hip - sdpa| = {err:.4f}")


if __name__ == "__main__":
    main()
