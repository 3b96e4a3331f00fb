"""LR schedulers (capability parity with reference
src/modalities/optimizers/lr_schedulers.py:8-65)."""

import math

import torch
from torch.optim.lr_scheduler import LambdaLR, SequentialLR


class DummyLRScheduler(torch.optim.lr_scheduler.LRScheduler):
    """Constant LR, no-op scheduler."""

    def get_lr(self):
        return [group["lr"] for group in self.optimizer.param_groups]


def get_linear_warmup_cosine_annealing(optimizer, num_warmup_steps: int,
                                       num_total_steps: int, min_lr_ratio: float = 0.1):
    """Linear warmup then cosine decay to min_lr_ratio * base_lr."""

    def warmup(step):
        return (step + 1) / max(1, num_warmup_steps)

    def cosine(step):
        t = step / max(1, num_total_steps - num_warmup_steps)
        t = min(max(t, 0.0), 1.0)
        return min_lr_ratio + (1 - min_lr_ratio) * 0.5 * (1 + math.cos(math.pi * t))

    return SequentialLR(optimizer,
                        [LambdaLR(optimizer, warmup), LambdaLR(optimizer, cosine)],
                        milestones=[num_warmup_steps])


def get_cosine_annealing(optimizer, t_max: int, eta_min: float = 0.0):
    return torch.optim.lr_scheduler.CosineAnnealingLR(optimizer, T_max=t_max,
                                                      eta_min=eta_min)


def get_constant_lr(optimizer, factor: float = 1.0, total_iters: int = 0):
    return torch.optim.lr_scheduler.ConstantLR(optimizer, factor=factor,
                                               total_iters=total_iters)


def get_step_lr(optimizer, step_size: int, gamma: float = 0.1):
    return torch.optim.lr_scheduler.StepLR(optimizer, step_size=step_size, gamma=gamma)


def get_linear_lr(optimizer, start_factor: float = 1.0, end_factor: float = 0.0,
                  total_iters: int = 100):
    return torch.optim.lr_scheduler.LinearLR(optimizer, start_factor=start_factor,
                                             end_factor=end_factor, total_iters=total_iters)


def get_onecycle_lr(optimizer, max_lr: float, total_steps: int, pct_start: float = 0.3):
    return torch.optim.lr_scheduler.OneCycleLR(optimizer, max_lr=max_lr,
                                               total_steps=total_steps, pct_start=pct_start)
