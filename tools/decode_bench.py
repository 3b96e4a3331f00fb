"""KV-cache decode microbench: cached incremental decoding vs full-context
re-forward on the flagship 2.7B shape (GPU).

    PYTHONPATH=. python tools/decode_bench.py [--ctx 512 2048 3968]
"""

import argparse
import importlib
import time

import torch


def build_model():
    bench = importlib.import_module("bench")
    from modalities_amd.models.gpt2 import GPT2LLM
    cfg = bench.build_model_cfg("gpt2-2.7b")
    cfg.fused_qkv = True
    with torch.device("meta"):
        m = GPT2LLM(cfg)
    m = m.to_empty(device="cuda")
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.02)
    return m.to(torch.bfloat16).eval(), cfg


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ctx", type=int, nargs="+", default=[512, 2048, 3968])
    ap.add_argument("--iters", type=int, default=32)
    args = ap.parse_args()
    model, cfg = build_model()
    for ctx in args.ctx:
        ids = torch.randint(0, cfg.vocab_size, (1, ctx), device="cuda")
        with torch.no_grad():
            cache = model.new_kv_cache(1, max_len=ctx + args.iters + 8)
            out = model.forward_cached({"input_ids": ids}, cache)
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(args.iters):
                nxt = out["logits"][:, -1:].argmax(-1)
                out = model.forward_cached({"input_ids": nxt}, cache)
            torch.cuda.synchronize()
            cached_ms = (time.time() - t0) / args.iters * 1000

            full = ids
            out2 = model({"input_ids": full})
            torch.cuda.synchronize()
            t0 = time.time()
            n_ref = max(4, args.iters // 4)
            for _ in range(n_ref):
                nxt = out2["logits"][:, -1:].argmax(-1)
                full = torch.cat([full, nxt], 1)
                out2 = model({"input_ids": full})
            torch.cuda.synchronize()
            refwd_ms = (time.time() - t0) / n_ref * 1000
        print(f"ctx {ctx}: cached {cached_ms:.1f} ms/tok  "
              f"re-forward {refwd_ms:.1f} ms/tok  ({refwd_ms/cached_ms:.1f}x)")


if __name__ == "__main__":
    main()
