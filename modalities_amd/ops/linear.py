"""TwoStreamLinear: nn.Linear with a two-stream backward (MI355X-native
addition, no reference analog).

Autograd runs a linear's input-gradient and weight-gradient GEMMs
sequentially on one stream; they are independent, and each alone leaves
tail waves of the 256-CU chip idle. This drop-in Linear computes the wgrad
(+ bias grad) on a side HIP stream overlapped with the dgrad, joined by
events before returning (the same fork/join pattern as the dq/dkdv overlap
in attention_bwd.hip, which measured -16% there). State-dict compatible
with nn.Linear; inactive on CPU.
"""

import os
from typing import Optional

import torch
import torch.nn as nn

_side_stream: Optional[torch.cuda.Stream] = None

# Overlap only pays when one GEMM underfills the 256-CU chip: for very large
# weights (lm_head: 50304x2560) both wgrad and dgrad already use
# chip-filling persistent kernels, and co-running them degrades each more
# than the overlap saves. Weights above this element count run sequentially.
_TWO_STREAM_MAX_WEIGHT = int(os.environ.get(
    "MODALITIES_AMD_TWO_STREAM_MAX_WEIGHT", 64 * 1024 * 1024))

# Same reasoning for the activation row count (tokens per micro-batch), but
# with a harder failure mode: above ~8k rows hipBLASLt selects persistent /
# Stream-K kernels for BOTH backward GEMMs, and two device-filling kernels
# with intra-kernel global synchronization co-running on concurrent HIP
# streams can wedge the device outright (spinning workgroups hold CUs the
# other kernel's unlaunched workgroups need — observed as a hard hang at
# micro-batch >= 3 x 4096 tokens on MI355X, see tools/debug_b4f.py).
# Shapes above this row bound run the standard sequential backward.
_TWO_STREAM_MAX_ROWS = int(os.environ.get(
    "MODALITIES_AMD_TWO_STREAM_MAX_ROWS", 8192))


def _get_side_stream() -> torch.cuda.Stream:
    global _side_stream
    if _side_stream is None:
        _side_stream = torch.cuda.Stream()
    return _side_stream


class _TwoStreamLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        if not dy.is_cuda or dy2.shape[0] > _TWO_STREAM_MAX_ROWS:
            dw = dy2.t() @ x2
            db = dy2.sum(0) if ctx.has_bias else None
            dx = (dy2 @ weight).view_as(x)
            return dx, dw, db
        main = torch.cuda.current_stream()
        side = _get_side_stream()
        side.wait_stream(main)
        with torch.cuda.stream(side):
            dw = dy2.t() @ x2
            db = dy2.sum(0) if ctx.has_bias else None
        dx = (dy2 @ weight).view_as(x)
        main.wait_stream(side)
        # dw/db were allocated on the side stream but are consumed on main;
        # the wait above orders that, record_stream keeps the allocator from
        # re-using them early on the side timeline.
        dw.record_stream(main)
        if db is not None:
            db.record_stream(main)
        return dx, dw, db


class TwoStreamLinear(nn.Linear):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and torch.is_grad_enabled() and (
                x.requires_grad or self.weight.requires_grad) and \
                self.weight.numel() <= _TWO_STREAM_MAX_WEIGHT:
            return _TwoStreamLinearFn.apply(x, self.weight, self.bias)
        return torch.nn.functional.linear(x, self.weight, self.bias)
