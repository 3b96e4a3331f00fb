"""SwiGLU gate epilogue (K6): out = silu(gate) * up, fwd + bwd.

HIP kernel: csrc/silu_mul.hip (vectorized elementwise, HBM-bound).
Replaces the reference's eager silu/mul in SwiGLU (reference:
src/modalities/models/model.py:141-157)."""

import torch

from modalities_amd.ops.backend import use_hip, hip_ext


class _SiluMulHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ctx.save_for_backward(gate, up)
        return hip_ext().silu_mul_fwd(gate.contiguous(), up.contiguous())

    @staticmethod
    def backward(ctx, dout):
        gate, up = ctx.saved_tensors
        dgate, dup = hip_ext().silu_mul_bwd(dout.contiguous(), gate.contiguous(),
                                            up.contiguous())
        return dgate, dup


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if use_hip(gate, up):
        return _SiluMulHip.apply(gate, up)
    gf = gate.float()
    return (torch.nn.functional.silu(gf) * up.float()).to(gate.dtype)


class _SiluMulJointHip(torch.autograd.Function):
    """Joint-layout variant: wv = [.., 2H] (gate | up halves from ONE
    packed up-projection GEMM); backward produces the joint dwv buffer the
    packed GEMM backward consumes directly (no torch.cat)."""

    @staticmethod
    def forward(ctx, wv):
        wv = wv.contiguous()
        ctx.save_for_backward(wv)
        return hip_ext().silu_mul_joint_fwd(wv)

    @staticmethod
    def backward(ctx, dout):
        (wv,) = ctx.saved_tensors
        return hip_ext().silu_mul_joint_bwd(dout.contiguous(), wv)


def silu_mul_joint(wv: torch.Tensor) -> torch.Tensor:
    """out = silu(wv[.., :H]) * wv[.., H:] for a packed [.., 2H] tensor."""
    if use_hip(wv):
        return _SiluMulJointHip.apply(wv)
    h = wv.shape[-1] // 2
    return silu_mul(wv[..., :h].contiguous(), wv[..., h:].contiguous())
