#!/usr/bin/env python3
"""Measure fused-vs-separate GEMM shapes for the 2.7B block (qkv, SwiGLU W/V)."""
import sys, time
import torch

def t(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

dev = "cuda"
M, h = 8192, 2560
x = torch.randn(M, h, device=dev, dtype=torch.bfloat16)
wq = torch.randn(h, h, device=dev, dtype=torch.bfloat16)
wk = torch.randn(h, h, device=dev, dtype=torch.bfloat16)
wv = torch.randn(h, h, device=dev, dtype=torch.bfloat16)
wqkv = torch.randn(h, 3*h, device=dev, dtype=torch.bfloat16)
print(f"qkv separate: {t(lambda: (x@wq, x@wk, x@wv)):.3f} ms")
print(f"qkv fused   : {t(lambda: x@wqkv):.3f} ms")
hid = 6912
w1 = torch.randn(h, hid, device=dev, dtype=torch.bfloat16)
w2 = torch.randn(h, hid, device=dev, dtype=torch.bfloat16)
wf = torch.randn(h, 2*hid, device=dev, dtype=torch.bfloat16)
print(f"W/V separate: {t(lambda: (x@w1, x@w2)):.3f} ms")
print(f"W/V fused   : {t(lambda: x@wf):.3f} ms")
# lm_head shape for reference
wl = torch.randn(h, 50304, device=dev, dtype=torch.bfloat16)
print(f"lm_head     : {t(lambda: x@wl, 10):.3f} ms")
