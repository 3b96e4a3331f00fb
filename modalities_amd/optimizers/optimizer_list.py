"""Optimizer/scheduler lists for multi-part (pipeline-parallel) models
(capability parity with reference src/modalities/optimizers/
optimizer_list.py:16-52): a single optimizer/scheduler facade over one
instance per model part."""

from typing import Iterable

import torch


class OptimizersList(torch.optim.Optimizer):
    def __init__(self, optimizers: list[torch.optim.Optimizer]):
        if not optimizers:
            raise ValueError("OptimizersList needs at least one optimizer")
        self.optimizers = list(optimizers)
        # note: deliberately NOT calling super().__init__ — this is a facade

    @property
    def param_groups(self):
        return [g for opt in self.optimizers for g in opt.param_groups]

    def step(self, closure=None):
        for opt in self.optimizers:
            opt.step()

    def zero_grad(self, set_to_none: bool = True):
        for opt in self.optimizers:
            opt.zero_grad()

    def state_dict(self) -> dict:
        return {"optimizers": [opt.state_dict() for opt in self.optimizers]}

    def load_state_dict(self, state_dict: dict) -> None:
        for opt, sd in zip(self.optimizers, state_dict["optimizers"]):
            opt.load_state_dict(sd)


class SchedulerList:
    def __init__(self, schedulers: Iterable):
        self.schedulers = list(schedulers)

    def step(self):
        for s in self.schedulers:
            s.step()

    def state_dict(self) -> dict:
        return {"schedulers": [s.state_dict() for s in self.schedulers]}

    def load_state_dict(self, state_dict: dict) -> None:
        for s, sd in zip(self.schedulers, state_dict["schedulers"]):
            s.load_state_dict(sd)

    def get_last_lr(self):
        return [lr for s in self.schedulers for lr in s.get_last_lr()]
