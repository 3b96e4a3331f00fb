"""Activation checkpointing, 3 variants (capability parity with reference
src/modalities/training/activation_checkpointing/activation_checkpointing.py:47-199):

- full: every transformer block recomputes activations in backward
- selective layer: every k-th block checkpointed
- selective op: checkpoint blocks but SAVE the outputs of expensive ops
  (matmul/attention) so only cheap elementwise work recomputes

Applied in place on a GPT2-style model's `blocks` ModuleList before any
sharding wrap (our engine's forward hooks compose with the checkpoint
wrapper because gather happens in the pre-forward hook, which the
recomputation re-enters)."""

from enum import Enum
from functools import partial

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint


class ActivationCheckpointingVariant(str, Enum):
    FULL_ACTIVATION_CHECKPOINTING = "full_activation_checkpointing"
    SELECTIVE_LAYER_ACTIVATION_CHECKPOINTING = "selective_layer_activation_checkpointing"
    SELECTIVE_OP_ACTIVATION_CHECKPOINTING = "selective_op_activation_checkpointing"


# Ops whose outputs are worth saving (reference saved-op dict :70-86 lists
# mm, SDPA, reduce_scatter, max; ours maps to mm + our custom attention which
# is opaque to the dispatcher and therefore recomputed — acceptable because
# its save set (o, lse) is produced by the autograd.Function anyway).
_SAVE_LIST = None


def _get_save_list():
    global _SAVE_LIST
    if _SAVE_LIST is None:
        ops = torch.ops.aten
        _SAVE_LIST = {
            ops.mm.default,
            ops.addmm.default,
            ops._scaled_dot_product_flash_attention.default,
            ops._scaled_dot_product_efficient_attention.default,
            ops.max.default,
        }
        # the HIP flash attention (K1) registered through torch.library —
        # selective-op AC saves its output instead of recomputing the whole
        # attention
        from modalities_amd.ops.attention import _ensure_custom_op
        if _ensure_custom_op():
            _SAVE_LIST.add(torch.ops.modalities_amd.flash_attention.default)
    return _SAVE_LIST


class CheckpointedBlock(nn.Module):
    """Wraps a block so its forward runs under torch.utils.checkpoint."""

    def __init__(self, block: nn.Module, context_fn=None):
        super().__init__()
        self.block = block
        self._context_fn = context_fn

    def forward(self, *args, **kwargs):
        if not torch.is_grad_enabled():
            return self.block(*args, **kwargs)
        kw = dict(use_reentrant=False)
        if self._context_fn is not None:
            kw["context_fn"] = self._context_fn
        return checkpoint(self.block, *args, **kwargs, **kw)

    def forward_cached(self, *args, **kwargs):
        # incremental decoding is inference-only: no recompute needed
        return self.block.forward_cached(*args, **kwargs)


def apply_activation_checkpointing_(model,
                                    variant: ActivationCheckpointingVariant,
                                    layers_fqn: str = "blocks",
                                    every_k_layers: int = 1) -> None:
    """Replace the model's block modules with checkpointed wrappers."""
    parent = model
    parts = layers_fqn.split(".")
    for p in parts[:-1]:
        parent = getattr(parent, p)
    blocks: nn.ModuleList = getattr(parent, parts[-1])
    if not isinstance(blocks, nn.ModuleList):
        raise TypeError(f"{layers_fqn} is not an nn.ModuleList")

    context_fn = None
    if variant == ActivationCheckpointingVariant.SELECTIVE_OP_ACTIVATION_CHECKPOINTING:
        from torch.utils.checkpoint import (
            CheckpointPolicy, create_selective_checkpoint_contexts)

        def policy(ctx, op, *args, **kwargs):
            return (CheckpointPolicy.MUST_SAVE if op in _get_save_list()
                    else CheckpointPolicy.PREFER_RECOMPUTE)

        context_fn = partial(create_selective_checkpoint_contexts, policy)

    new_blocks = []
    for i, block in enumerate(blocks):
        selective_skip = (
            variant == ActivationCheckpointingVariant.SELECTIVE_LAYER_ACTIVATION_CHECKPOINTING
            and i % every_k_layers != 0)
        if selective_skip:
            new_blocks.append(block)
        else:
            new_blocks.append(CheckpointedBlock(block, context_fn=context_fn))
    setattr(parent, parts[-1], nn.ModuleList(new_blocks))
