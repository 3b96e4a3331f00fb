#!/usr/bin/env python3
"""Isolated flash-attention microbenchmark (fwd / bwd) on MI355X.

Shapes default to the 2.7B flagship: B=2, T=4096, Hq=Hkv=20, D=128.
Reports TFLOP/s against causal-attention flop counts (0.5 * 4*B*T^2*H*D per
matmul pair)."""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=2)
    p.add_argument("--T", type=int, default=4096)
    p.add_argument("--Hq", type=int, default=20)
    p.add_argument("--Hkv", type=int, default=20)
    p.add_argument("--D", type=int, default=128)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--what", choices=["fwd", "bwd", "both", "ablate",
                                      "ablate-bwd"], default="both")
    args = p.parse_args()

    from modalities_amd.ops.backend import hip_ext
    ext = hip_ext()
    dev = "cuda:0"
    torch.manual_seed(0)
    B, T, Hq, Hkv, D = args.B, args.T, args.Hq, args.Hkv, args.D
    q = torch.randn(B, T, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, T, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, T, Hkv, D, device=dev, dtype=torch.bfloat16)
    do = torch.randn(B, T, Hq, D, device=dev, dtype=torch.bfloat16)

    o, lse = ext.attn_fwd(q, k, v, True)
    torch.cuda.synchronize()

    # causal: ~half the T^2 space; fwd = 2 matmuls, bwd = 5 matmuls
    flops_fwd = 0.5 * 4 * B * T * T * Hq * D
    flops_bwd = 0.5 * 10 * B * T * T * Hq * D

    if args.what in ("fwd", "both"):
        t0 = time.perf_counter()
        for _ in range(args.iters):
            o, lse = ext.attn_fwd(q, k, v, True)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(f"fwd: {dt*1e3:8.3f} ms  {flops_fwd/dt/1e12:8.1f} TF/s")

    if args.what == "ablate":
        for mode, name in [(0, "full"), (1, "no-Vt-stage"), (2, "no-stage"),
                           (3, "no-softmax")]:
            ext.attn_fwd_ablate(q, k, v, mode)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                ext.attn_fwd_ablate(q, k, v, mode)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.iters
            print(f"ablate {name:12s}: {dt*1e3:8.3f} ms  "
                  f"{flops_fwd/dt/1e12:8.1f} TF/s")

    if args.what == "ablate-bwd":
        # dkdv-only flops: 3 of the 5 bwd matmuls (S recompute, dV, dK)
        flops_dkdv = 0.5 * 6 * B * T * T * Hq * D
        delta = (do.float() * o.float()).sum(-1).permute(0, 2, 1).contiguous()
        for mode, name in [(0, "full"), (1, "no-qt/dot-stage"),
                           (2, "no-stage"), (3, "no-lse/delta-loads"),
                           (4, "no-dkdv-mfma"), (5, "stage-only")]:
            ext.attn_bwd_dkdv_ablate(do, q, k, v, lse, delta, mode)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                ext.attn_bwd_dkdv_ablate(do, q, k, v, lse, delta, mode)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.iters
            print(f"dkdv ablate {name:20s}: {dt*1e3:8.3f} ms  "
                  f"{flops_dkdv/dt/1e12:8.1f} TF/s")

    if args.what in ("bwd", "both"):
        t0 = time.perf_counter()
        for _ in range(args.iters):
            dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, True)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(f"bwd: {dt*1e3:8.3f} ms  {flops_bwd/dt/1e12:8.1f} TF/s")


if __name__ == "__main__":
    main()
