"""Optimizers for the sharded engine + plain models.

ShardedAdamW steps directly on the engine's flat fp32 master shards with the
fused HIP AdamW kernel (K9) and a per-element weight-decay mask — the flat
equivalent of the reference's regex weight-decay groups (reference:
src/modalities/optimizers/optimizer_factory.py:22-215)."""

from typing import Optional

import torch

from modalities_amd.parallel.fsdp import XGMIShardedModel


class ShardedAdamW(torch.optim.Optimizer):
    """AdamW over XGMIShardedModel master shards.

    - exp_avg / exp_avg_sq are fp32 shards (persisted in state for DCP).
    - weight decay is applied through each unit's wd_mask_shard so norm/bias
      elements get 0 decay inside the flat shard.
    """

    def __init__(self, sharded_model: XGMIShardedModel, lr: float = 3e-4,
                 betas: tuple[float, float] = (0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.1):
        self.sharded_model = sharded_model
        # clip_grad_norm_ checks this flag to defer the clip-coefficient
        # multiply into our fused kernel instead of a separate grad pass
        sharded_model._optimizer_consumes_grad_scale = True
        sharded_model._pending_grad_scale = None
        params = [u.master_shard for u in sharded_model.units]
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        for u in sharded_model.units:
            st = self.state[u.master_shard]
            st["step"] = 0
            st["exp_avg"] = torch.zeros_like(u.master_shard)
            st["exp_avg_sq"] = torch.zeros_like(u.master_shard)
        # Device-resident shared step counter (hipGraph-safe bias correction:
        # a replayed training step advances it on-device; host-computed bc
        # would freeze at capture). Lazily initialized from the CPU step so
        # warmstart loads are honored.
        self._step_dev = None

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        group = self.param_groups[0]
        lr, (beta1, beta2) = group["lr"], group["betas"]
        eps, wd = group["eps"], group["weight_decay"]
        from modalities_amd.ops.backend import hip_available, hip_ext
        on_gpu = hip_available() and self.sharded_model.units \
            and self.sharded_model.units[0].master_shard.is_cuda
        if on_gpu:
            if self._step_dev is None:
                start = max((self.state[u.master_shard].get("step", 0)
                             for u in self.sharded_model.units), default=0)
                self._step_dev = torch.tensor(
                    start, dtype=torch.int32,
                    device=self.sharded_model.units[0].master_shard.device)
            self._step_dev += 1  # on-device: advances under hipGraph replay
            # The shared device counter assumes every unit steps every call
            # (true for full-model backward: all grads reduce each step). A
            # unit skipping would silently diverge its bias correction from
            # the per-unit CPU counter persisted in checkpoints (ADVICE r1
            # #4) — fail loudly instead.
            fresh = [u.grad_fresh for u in self.sharded_model.units]
            if any(fresh) and not all(fresh):
                raise RuntimeError(
                    "fused AdamW devstep: mixed grad_fresh across units "
                    f"({sum(fresh)}/{len(fresh)}) — the shared device step "
                    "counter requires all units to step together")
        gscale = getattr(self.sharded_model, "_pending_grad_scale", None)
        self.sharded_model._pending_grad_scale = None
        for ui, u in enumerate(self.sharded_model.units):
            if not u.grad_fresh:
                continue  # no grads reduced for this unit this step
            st = self.state[u.master_shard]
            st["step"] += 1
            step = st["step"]
            m, v = st["exp_avg"], st["exp_avg_sq"]
            g = u.grad_shard
            p = u.master_shard
            if on_gpu and u.bf16_shard.dtype == torch.bfloat16:
                # fused publish: the kernel writes the bf16 working shard in
                # the same pass (saves a full param re-read + cast)
                hip_ext().fused_adamw_masked_devstep(
                    p, g, m, v, u.wd_mask_shard, self._step_dev,
                    u.bf16_shard, gscale, lr, beta1, beta2, eps, wd)
                if u.full_buf is not None and self.sharded_model.world > 1:
                    u.free_full()  # stale gathered params
                self.sharded_model.prefetch_unit_gather(ui)
            elif on_gpu:  # fp32 working copy on GPU: unfused publish below
                if gscale is not None:
                    g = g.mul(gscale)
                bc1 = 1.0 - beta1 ** step
                bc2 = 1.0 - beta2 ** step
                hip_ext().fused_adamw_masked(p, g, m, v, u.wd_mask_shard,
                                             lr, beta1, beta2, eps, wd,
                                             bc1, bc2)
                u.publish_master()
            else:
                if gscale is not None:
                    g = g.mul(gscale)
                bc1 = 1.0 - beta1 ** step
                bc2 = 1.0 - beta2 ** step
                p.mul_(1.0 - lr * wd * u.wd_mask_shard)
                m.mul_(beta1).add_(g, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                denom = (v / bc2).sqrt().add_(eps)
                p.add_(-lr / bc1 * m / denom)
        if on_gpu:
            # stale gathered buffers were already freed per unit (before the
            # next-step prefetch gathers, which must NOT be freed here)
            pass
        else:
            self.sharded_model.publish_master()
            # same next-step gather prefetch as the GPU path (synchronous on
            # CPU/gloo, which lets the world-2 tests exercise the identical
            # collective ordering the RCCL path uses)
            for ui in range(len(self.sharded_model.units)):
                self.sharded_model.prefetch_unit_gather(ui)
        return loss

    def zero_grad(self, set_to_none: bool = True):
        self.sharded_model.zero_grad_shards()


def get_adam_w(wrapped_model, lr: float = 3e-4, betas=(0.9, 0.95), eps: float = 1e-8,
               weight_decay: float = 0.1, weight_decay_groups_excluded=None,
               foreach: Optional[bool] = None, fused: Optional[bool] = None):
    """Factory: sharded models get ShardedAdamW; plain modules get
    torch.optim.AdamW with decay/no-decay param groups."""
    if isinstance(wrapped_model, XGMIShardedModel):
        return ShardedAdamW(wrapped_model, lr=lr, betas=betas, eps=eps,
                            weight_decay=weight_decay)
    decay, no_decay = [], []
    for name, p in wrapped_model.named_parameters():
        if not p.requires_grad:
            continue
        (no_decay if p.ndim < 2 or "norm" in name.lower() or name.endswith(".bias")
         else decay).append(p)
    groups = [{"params": decay, "weight_decay": weight_decay},
              {"params": no_decay, "weight_decay": 0.0}]
    return torch.optim.AdamW(groups, lr=lr, betas=betas, eps=eps,
                             foreach=foreach, fused=False)
