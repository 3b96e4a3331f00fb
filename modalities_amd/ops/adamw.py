"""Fused multi-tensor AdamW step (K9) + multi-tensor grad-norm/clip (K10).

HIP kernel: csrc/adamw.hip. Replaces torch.optim.AdamW(fused=True)
(reference: src/modalities/optimizers/optimizer_factory.py:38-50) and
clip_grads_with_norm_ (reference: training/gradient_clipping/
fsdp_gradient_clipper.py:144-229)."""

from typing import Iterable

import torch

from modalities_amd.ops.backend import hip_ext, hip_available


@torch.no_grad()
def fused_adamw_step(params: list, grads: list, exp_avgs: list, exp_avg_sqs: list,
                     step: int, lr: float, beta1: float, beta2: float,
                     eps: float, weight_decay: float) -> None:
    """Apply one AdamW update to flat fp32 param shards.

    params/exp_avg/exp_avg_sq: fp32; grads: fp32 (already unscaled/reduced).
    """
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    if hip_available() and params and params[0].is_cuda:
        hip_ext().fused_adamw(params, grads, exp_avgs, exp_avg_sqs,
                              lr, beta1, beta2, eps, weight_decay, bc1, bc2)
        return
    # torch reference path (CPU tests)
    for p, g, m, v in zip(params, grads, exp_avgs, exp_avg_sqs):
        p.mul_(1.0 - lr * weight_decay)
        m.mul_(beta1).add_(g, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        denom = (v / bc2).sqrt_().add_(eps)
        p.addcdiv_(m, denom, value=-lr / bc1)


@torch.no_grad()
def multi_tensor_l2norm(tensors: Iterable[torch.Tensor]) -> torch.Tensor:
    """Sum-of-squares -> local L2 norm over a list of tensors (one scalar)."""
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return torch.zeros((), dtype=torch.float32)
    if hip_available() and tensors[0].is_cuda:
        return hip_ext().multi_tensor_sqsum(list(tensors)).sqrt()
    acc = torch.zeros((), dtype=torch.float32, device=tensors[0].device)
    for t in tensors:
        acc += t.float().pow(2).sum()
    return acc.sqrt()


@torch.no_grad()
def multi_tensor_scale_(tensors: Iterable[torch.Tensor], scale: torch.Tensor) -> None:
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return
    if hip_available() and tensors[0].is_cuda and isinstance(scale, torch.Tensor):
        hip_ext().multi_tensor_scale(list(tensors), scale)
        return
    for t in tensors:
        t.mul_(scale.to(t.device) if isinstance(scale, torch.Tensor) else scale)
