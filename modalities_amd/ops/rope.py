"""Rotary position embedding (K4): fused rotate-half applied to q and k.

HIP kernel: csrc/rope.hip. cos/sin tables are precomputed on the HOST once
and cached (guide Appendix B: on-device trig turns the op VALU-bound).
Replaces the reference's elementwise RotaryTransform (reference:
src/modalities/models/gpt2/gpt2_model.py:114-229)."""

import torch

from modalities_amd.ops.backend import use_hip, hip_ext


def precompute_rope_cos_sin(seq_len: int, head_dim: int, base: float = 10000.0,
                            device=None, dtype=torch.float32):
    """Return (cos, sin) of shape [seq_len, head_dim//2] (f32)."""
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2, dtype=torch.float32,
                                            device=device) / head_dim))
    t = torch.arange(seq_len, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)  # [T, D/2]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def _rotate_half(x):
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


def _rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    # x: [B, T, H, D]; cos/sin: [T, D/2]
    T, D = x.shape[1], x.shape[-1]
    c = torch.cat([cos[:T], cos[:T]], dim=-1).view(1, T, 1, D).to(torch.float32)
    s = torch.cat([sin[:T], sin[:T]], dim=-1).view(1, T, 1, D).to(torch.float32)
    xf = x.float()
    return (xf * c + _rotate_half(xf) * s).to(x.dtype)


class _RopeHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        ctx.save_for_backward(cos, sin)
        return hip_ext().rope_fwd(x.contiguous(), cos, sin, False)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        # The transpose of a rotation is rotation by -theta.
        return hip_ext().rope_fwd(dy.contiguous(), cos, sin, True), None, None


def rope_apply(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """Apply rotate-half RoPE to x of shape [B, T, H, D]."""
    if use_hip(x):
        return _RopeHip.apply(x, cos, sin)
    return _rope_ref(x, cos, sin)
