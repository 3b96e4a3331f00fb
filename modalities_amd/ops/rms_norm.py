"""RMSNorm fwd/bwd (K5). HIP kernel: csrc/rms_norm.hip (wave-level reduction,
vectorized bf16x8 loads per guide G13). Replaces the reference's
nn.RMSNorm / custom RMSLayerNorm (reference: src/modalities/models/components/
layer_norms.py:9-65, used gpt2_model.py:923-930)."""

import torch

from modalities_amd.ops.backend import use_hip, hip_ext


def _rms_norm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * weight.float()).to(x.dtype)


class _RMSNormHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        shape = x.shape
        x2d = x.contiguous().view(-1, shape[-1])
        y, invrms = hip_ext().rmsnorm_fwd(x2d, weight, eps)
        ctx.save_for_backward(x2d, weight, invrms)
        ctx.shape = shape
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight, invrms = ctx.saved_tensors
        dx, dw = hip_ext().rmsnorm_bwd(dy.contiguous().view_as(x2d), x2d, weight, invrms)
        return dx.view(ctx.shape), dw.to(weight.dtype), None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if use_hip(x, weight):
        return _RMSNormHip.apply(x, weight, eps)
    return _rms_norm_ref(x, weight, eps)


class RMSNorm(torch.nn.Module):
    """Drop-in RMSNorm module backed by the HIP kernel on device."""

    def __init__(self, normalized_shape: int, eps: float = 1e-6, device=None, dtype=None):
        super().__init__()
        self.eps = eps
        self.normalized_shape = normalized_shape
        self.weight = torch.nn.Parameter(
            torch.empty(normalized_shape, device=device, dtype=dtype))
        self.reset_parameters()

    def reset_parameters(self):
        torch.nn.init.ones_(self.weight)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, self.weight, self.eps)

    def extra_repr(self):
        return f"{self.normalized_shape}, eps={self.eps}"
