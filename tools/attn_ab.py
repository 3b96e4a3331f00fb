"""A/B micro-benchmark: K1 attention v1 vs v2 on the bench shapes.

Usage (on a GPU box):  python tools/attn_ab.py [--iters 50]
Prints per-kernel times and effective TFLOP/s (causal-adjusted) for
fwd and bwd at the 2.7B bench shape (hd=128/20 heads) and the exact
reference 2.7B shape (hd=80/32 heads, v2 only).
"""

import argparse
import sys

import torch


def flops_attn(B, T, H, D, causal=True, n_matmul=2):
    f = 2.0 * B * H * T * T * D * n_matmul
    return f / 2 if causal else f


def timeit(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters  # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()

    from modalities_amd.ops.backend import hip_ext
    ext = hip_ext()
    dev = "cuda"

    shapes = [
        ("bench-2.7B (hd128)", 2, 4096, 20, 20, 128, (1, 2)),
        ("ref-2.7B (hd80)", 2, 4096, 32, 32, 80, (2,)),
        ("8B-class (hd128 gqa)", 1, 8192, 32, 8, 128, (1, 2)),
    ]
    for name, B, T, Hq, Hkv, D, impls in shapes:
        torch.manual_seed(0)
        q = torch.randn(B, T, Hq, D, device=dev).bfloat16()
        k = torch.randn(B, T, Hkv, D, device=dev).bfloat16()
        v = torch.randn(B, T, Hkv, D, device=dev).bfloat16()
        do = torch.randn(B, T, Hq, D, device=dev).bfloat16()
        print(f"== {name}: B{B} T{T} Hq{Hq} Hkv{Hkv} D{D}")
        outs = {}
        for impl in impls:
            fwd = ext.attn_fwd_v1 if impl == 1 else ext.attn_fwd_v2
            bwd = ext.attn_bwd_v1 if impl == 1 else ext.attn_bwd_v2
            o, lse = fwd(q, k, v, True, 0)
            dq, dk, dv = bwd(do, q, k, v, o, lse, True, 0)
            outs[impl] = (o, dq, dk, dv)
            tf = timeit(lambda: fwd(q, k, v, True, 0), args.iters)
            tb = timeit(lambda: bwd(do, q, k, v, o, lse, True, 0), args.iters)
            ffw = flops_attn(B, T, Hq, D, n_matmul=2) / (tf * 1e-3) / 1e12
            fbw = flops_attn(B, T, Hq, D, n_matmul=5) / (tb * 1e-3) / 1e12
            print(f"  v{impl}: fwd {tf:7.3f} ms ({ffw:6.1f} TF/s eff)   "
                  f"bwd {tb:7.3f} ms ({fbw:6.1f} TF/s eff)")
        if len(outs) == 2:
            for i, tag in enumerate(("o", "dq", "dk", "dv")):
                a, b = outs[1][i].float(), outs[2][i].float()
                md = (a - b).abs().max().item()
                print(f"  v1-v2 max|d{tag}| = {md:.4e}")
    print("done")


if __name__ == "__main__":
    sys.exit(main())
