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


def _attention_ref(q, k, v, causal=True, q_offset=None):
    # q: [B, T, Hq, D], k/v: [B, S, Hkv, D] -> [B, T, Hq, D]; fp32 compute.
    # q_offset: global position of q row 0 relative to k row 0 (context
    # parallelism); default aligns the q block to the END of the keys.
    B, T, Hq, D = q.shape
    S, Hkv = k.shape[1], k.shape[2]
    rep = Hq // Hkv
    qf = q.permute(0, 2, 1, 3).float()                      # [B,Hq,T,D]
    kf = k.permute(0, 2, 1, 3).float().repeat_interleave(rep, dim=1)
    vf = v.permute(0, 2, 1, 3).float().repeat_interleave(rep, dim=1)
    att = qf @ kf.transpose(-2, -1) / math.sqrt(D)          # [B,Hq,T,S]
    if causal:
        diag = (S - T) if q_offset is None else q_offset
        mask = torch.ones(T, S, dtype=torch.bool, device=q.device).tril(diag)
        att = att.masked_fill(~mask, float("-inf"))
    att = att.softmax(-1)
    out = att @ vf                                          # [B,Hq,T,D]
    return out.permute(0, 2, 1, 3).to(q.dtype)


class _FlashAttnHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, q_offset):
        o, lse = hip_ext().attn_fwd(q, k, v, causal, q_offset)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.q_offset = q_offset
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = hip_ext().attn_bwd(do.contiguous(), q, k, v, o, lse,
                                        ctx.causal, ctx.q_offset)
        return dq, dk, dv, None, None


# ---------------------------------------------------------------------------
# Dispatcher-visible registration: a raw autograd.Function is opaque to
# torch dispatch, so selective-op activation checkpointing could never SAVE
# the attention output (it was always recomputed).
# Registering the op through torch.library lets the AC save-list match it.
# ---------------------------------------------------------------------------

_CUSTOM_OP_READY = False


def _ensure_custom_op():
    global _CUSTOM_OP_READY
    if _CUSTOM_OP_READY:
        return True
    try:
        @torch.library.custom_op("modalities_amd::flash_attention",
                                 mutates_args=())
        def _op(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                causal: bool, q_offset: int) -> tuple[torch.Tensor, torch.Tensor]:
            o, lse = hip_ext().attn_fwd(q, k, v, causal, q_offset)
            return o, lse

        @_op.register_fake
        def _(q, k, v, causal, q_offset):
            B, T, Hq, _ = q.shape
            return (torch.empty_like(q),
                    q.new_empty((B, Hq, T), dtype=torch.float32))

        def _setup(ctx, inputs, output):
            q, k, v, causal, q_offset = inputs
            ctx.save_for_backward(q, k, v, output[0], output[1])
            ctx.causal = causal
            ctx.q_offset = q_offset

        def _bwd(ctx, grad_o, grad_lse):
            q, k, v, o, lse = ctx.saved_tensors
            dq, dk, dv = hip_ext().attn_bwd(grad_o.contiguous(), q, k, v, o,
                                            lse, ctx.causal, ctx.q_offset)
            return dq, dk, dv, None, None

        _op.register_autograd(_bwd, setup_context=_setup)
        _CUSTOM_OP_READY = True
    except Exception:  # older torch without torch.library.custom_op
        _CUSTOM_OP_READY = False
    return _CUSTOM_OP_READY


from typing import Optional


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True,
                    q_offset: Optional[int] = None) -> torch.Tensor:
    """Causal flash attention. q: [B,T,Hq,D]; k,v: [B,S,Hkv,D].

    q_offset (context parallelism): global key position of q row 0; None =
    classic alignment (q block ends at the last key, i.e. offset S-T).
    The HIP kernels handle Tq != Tkv and offsets natively."""
    if use_hip(q, k, v):
        off = q_offset
        if off is None:
            off = k.shape[1] - q.shape[1]  # classic alignment (q at the end)
        if _ensure_custom_op():
            o, _ = torch.ops.modalities_amd.flash_attention(
                q.contiguous(), k.contiguous(), v.contiguous(), causal, off)
            return o
        return _FlashAttnHip.apply(q.contiguous(), k.contiguous(),
                                   v.contiguous(), causal, off)
    if q.is_cuda:
        # SDPA fallback (MODALITIES_AMD_FORCE_EAGER debug path)
        B, T, Hq, D = q.shape
        S, Hkv = k.shape[1], k.shape[2]
        rep = Hq // Hkv
        qt = q.transpose(1, 2)
        kt = k.transpose(1, 2).repeat_interleave(rep, dim=1)
        vt = v.transpose(1, 2).repeat_interleave(rep, dim=1)
        diag = (S - T) if q_offset is None else q_offset
        mask = torch.ones(T, S, dtype=torch.bool, device=q.device).tril(
            diag if causal else S)
        y = torch.nn.functional.scaled_dot_product_attention(
            qt, kt, vt, attn_mask=mask)
        return y.transpose(1, 2)
    return _attention_ref(q, k, v, causal, q_offset=q_offset)


class _FusedQKVRopeAttn(torch.autograd.Function):
    """Fused split + RoPE + flash attention over a joint QKV activation
    [B, T, Cq+Ck+Cv] (HIP path only).

    Forward RoPEs the q/k column blocks straight out of the joint tensor
    (slice-aware rope kernel: no .contiguous() copies of the slices).
    Backward assembles the joint dqkv gradient in place: the attention
    backward stores dv directly into its column block, and the inverse-RoPE
    kernel scatters dq/dk into theirs — eliminating the per-layer backward
    torch.cat (measured 2.5 ms/step on the 2.7B config) plus the forward
    q/k slice copies."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, Hq, Hkv, D):
        ext = hip_ext()
        B, T, _ = qkv.shape
        C, KV = Hq * D, Hkv * D
        qkv = qkv.contiguous()
        q = ext.rope_fwd_slice(qkv, cos, sin, 0, Hq, D, False)
        k = ext.rope_fwd_slice(qkv, cos, sin, C, Hkv, D, False)
        # V stays a row-strided view into the joint activation: the v2
        # kernels read it in place (no per-layer 2*B*T*KV copy)
        v = qkv[..., C + KV:].view(B, T, Hkv, D)
        o, lse = ext.attn_fwd(q, k, v, True, 0)
        ctx.save_for_backward(q, k, v, o, lse, cos, sin)
        ctx.dims = (Hq, Hkv, D)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse, cos, sin = ctx.saved_tensors
        Hq, Hkv, D = ctx.dims
        C, KV = Hq * D, Hkv * D
        B, T = q.shape[0], q.shape[1]
        ext = hip_ext()
        dqkv = torch.empty(B, T, C + 2 * KV, dtype=q.dtype, device=q.device)
        dq, dk = ext.attn_bwd_qkvjoint(do.contiguous(), q, k, v, o, lse,
                                       True, 0, dqkv, C + KV)
        ext.rope_bwd_slice(dq, cos, sin, dqkv, 0)
        ext.rope_bwd_slice(dk, cos, sin, dqkv, C)
        return dqkv, None, None, None, None, None


def fused_qkv_rope_attention(qkv: torch.Tensor, cos: torch.Tensor,
                             sin: torch.Tensor, n_head_q: int, n_head_kv: int,
                             head_dim: int) -> torch.Tensor:
    """Causal flash attention taking the joint QKV projection output
    directly; returns [B, T, n_head_q, head_dim]. HIP/GPU only — callers
    fall back to split + rope_apply + flash_attention elsewhere."""
    return _FusedQKVRopeAttn.apply(qkv, cos, sin, n_head_q, n_head_kv,
                                   head_dim)
