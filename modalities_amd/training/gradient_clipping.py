"""Gradient clippers (capability parity with reference
src/modalities/training/gradient_clipping/fsdp_gradient_clipper.py).

For XGMIShardedModel the clip is sharded-native (local multi-tensor sq-sum
over the fp32 grad shards + one scalar all-reduce, K10); for plain modules
torch.nn.utils.clip_grad_norm_ is used."""

from enum import Enum
from typing import Optional

import torch

from modalities_amd.parallel.fsdp import XGMIShardedModel


class GradientClippingMode(str, Enum):
    P2_NORM = "p2_norm"
    MAX_NORM = "max_norm"  # inf-norm
    VALUE = "value"


class GradientClipper:
    """Callable: model -> grad norm tensor (before clipping)."""

    def __init__(self, max_norm: Optional[float] = 1.0,
                 norm_type: GradientClippingMode = GradientClippingMode.P2_NORM,
                 device_mesh=None, pp_group=None):
        self.max_norm = max_norm
        self.norm_type = norm_type
        # PP: the grad norm combines across stages (reference
        # fsdp_gradient_clipper.py:166-169)
        if pp_group is None and device_mesh is not None:
            from modalities_amd.parallel.mesh import ParallelismDegrees
            dim = device_mesh.dims.get(ParallelismDegrees.PP)
            if dim is not None and dim.size > 1:
                pp_group = dim.group
        self.pp_group = pp_group

    def __call__(self, model) -> torch.Tensor:
        if isinstance(model, XGMIShardedModel):
            return model.clip_grad_norm_(self.max_norm, pp_group=self.pp_group)
        params = [p for p in model.parameters() if p.grad is not None]
        norm = 2.0 if self.norm_type == GradientClippingMode.P2_NORM else float("inf")
        if self.pp_group is not None:
            import torch.distributed as dist
            local = torch.stack([p.grad.float().pow(2).sum() for p in params]).sum()                 if params else torch.zeros(())
            dist.all_reduce(local, group=self.pp_group)
            total = local.sqrt()
            if self.max_norm is not None and params:
                clip = (self.max_norm / (total + 1e-6)).clamp(max=1.0)
                for p in params:
                    p.grad.mul_(clip)
            return total
        return torch.nn.utils.clip_grad_norm_(params, self.max_norm, norm_type=norm)


class DummyGradientClipper:
    """Measures nothing, clips nothing."""

    def __call__(self, model) -> torch.Tensor:
        return torch.tensor(-1.0)
