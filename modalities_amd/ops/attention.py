"""Flash-style causal attention (K1): MFMA/LDS-tiled fwd+bwd, GQA-aware.

HIP kernels: csrc/attention_fwd.hip / csrc/attention_bwd.hip — online-softmax
tiling per the CDNA4 guide (swapped QK^T for lane-local softmax rows,
XOR-swizzled K LDS image, ds_read_b64_tr_b16 for V).

Replaces both attention paths of the reference (reference:
src/modalities/models/gpt2/gpt2_model.py:595-658 — flash_attn_func wheel and
torch SDPA) with a single native kernel; GQA without K/V head repetition
(the reference's PYTORCH_FLASH path materializes repeated KV heads,
gpt2_model.py:550-593).

Layout: q [B, T, Hq, D]; k, v [B, T, Hkv, D]; Hq % Hkv == 0. Causal only
(decoder LM pretraining); the CPU fallback handles the general case for
tests."""

import math

import torch

from modalities_amd.ops.backend import use_hip, hip_ext


def _attention_ref(q, k, v, causal=True):
    # q: [B, T, Hq, D], k/v: [B, S, Hkv, D] -> [B, T, Hq, D]; fp32 compute.
    B, T, Hq, D = q.shape
    S, Hkv = k.shape[1], k.shape[2]
    rep = Hq // Hkv
    qf = q.permute(0, 2, 1, 3).float()                      # [B,Hq,T,D]
    kf = k.permute(0, 2, 1, 3).float().repeat_interleave(rep, dim=1)
    vf = v.permute(0, 2, 1, 3).float().repeat_interleave(rep, dim=1)
    att = qf @ kf.transpose(-2, -1) / math.sqrt(D)          # [B,Hq,T,S]
    if causal:
        mask = torch.ones(T, S, dtype=torch.bool, device=q.device).tril(S - T)
        att = att.masked_fill(~mask, float("-inf"))
    att = att.softmax(-1)
    out = att @ vf                                          # [B,Hq,T,D]
    return out.permute(0, 2, 1, 3).to(q.dtype)


class _FlashAttnHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        o, lse = hip_ext().attn_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = hip_ext().attn_bwd(do.contiguous(), q, k, v, o, lse, ctx.causal)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True) -> torch.Tensor:
    """Causal flash attention. q: [B,T,Hq,D]; k,v: [B,S,Hkv,D]."""
    if use_hip(q, k, v):
        return _FlashAttnHip.apply(q.contiguous(), k.contiguous(), v.contiguous(), causal)
    return _attention_ref(q, k, v, causal)
