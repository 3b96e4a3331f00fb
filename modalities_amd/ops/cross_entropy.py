"""Fused cross-entropy over the vocab (K8): streamed log-softmax + NLL
fwd/bwd without materializing a second fp32 logits-sized tensor.

HIP kernel: csrc/cross_entropy.hip. Replaces the reference's
nn.CrossEntropyLoss over [B*T, vocab] (reference:
src/modalities/loss_functions.py:33-52)."""

import torch

from modalities_amd.ops.backend import use_hip, hip_ext


class _FusedCEHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        losses, lse = hip_ext().cross_entropy_fwd(logits, targets, ignore_index)
        ctx.save_for_backward(logits, targets, lse)
        ctx.ignore_index = ignore_index
        n_valid = (targets != ignore_index).sum()
        ctx.n_valid = n_valid
        return losses.sum() / n_valid.clamp(min=1).to(losses.dtype)

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        scale = dloss.float() / ctx.n_valid.clamp(min=1).float()
        dlogits = hip_ext().cross_entropy_bwd(logits, targets, lse, scale,
                                              ctx.ignore_index)
        return dlogits, None, None


def fused_cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                        ignore_index: int = -100) -> torch.Tensor:
    """Mean CE over non-ignored targets. logits: [N, V]; targets: [N]."""
    logits = logits.reshape(-1, logits.shape[-1])
    targets = targets.reshape(-1)
    if use_hip(logits):
        return _FusedCEHip.apply(logits.contiguous(), targets.contiguous(), ignore_index)
    return torch.nn.functional.cross_entropy(logits.float(), targets,
                                             ignore_index=ignore_index)
