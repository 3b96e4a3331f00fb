"""General utilities (capability parity with reference
src/modalities/util.py:26-291): rank-0 printing, trainable-param counting
(local/global, sharded- and PP-aware), TimeRecorder."""

import time
from enum import Enum
from typing import Optional

import torch
import torch.distributed as dist


def print_rank_0(*args, **kwargs) -> None:
    if not (dist.is_available() and dist.is_initialized()) or dist.get_rank() == 0:
        print(*args, **kwargs)


def warn_rank_0(msg: str) -> None:
    import warnings
    if not (dist.is_available() and dist.is_initialized()) or dist.get_rank() == 0:
        warnings.warn(msg)


def get_local_number_of_trainable_parameters(model: torch.nn.Module) -> int:
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    if isinstance(model, XGMIShardedModel):
        return sum(u.shard_numel for u in model.units)
    return sum(p.numel() for p in model.parameters() if p.requires_grad)


def get_total_number_of_trainable_parameters(model: torch.nn.Module,
                                             pp_group=None) -> int:
    """Global parameter count: sharded models sum their flat units over the
    shard group; PP models all-reduce stage counts over the pp group
    (reference util.py:152-240)."""
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    if isinstance(model, XGMIShardedModel):
        total = sum(u.total_numel for u in model.units)
    else:
        total = get_local_number_of_trainable_parameters(model)
        if pp_group is not None and dist.is_initialized():
            t = torch.tensor([total], dtype=torch.long)
            dist.all_reduce(t, group=pp_group)
            total = int(t.item())
    return total


class TimeRecorderStates(str, Enum):
    STOPPED = "stopped"
    RUNNING = "running"


class TimeRecorder:
    """Accumulating stopwatch (reference util.py:247-291)."""

    def __init__(self):
        self._delta = 0.0
        self._start: Optional[float] = None
        self._state = TimeRecorderStates.STOPPED

    def start(self) -> None:
        if self._state == TimeRecorderStates.RUNNING:
            raise RuntimeError("TimeRecorder already running")
        self._start = time.perf_counter()
        self._state = TimeRecorderStates.RUNNING

    def stop(self) -> None:
        if self._state != TimeRecorderStates.RUNNING:
            raise RuntimeError("TimeRecorder is not running")
        self._delta += time.perf_counter() - self._start
        self._state = TimeRecorderStates.STOPPED

    def reset(self) -> None:
        if self._state == TimeRecorderStates.RUNNING:
            raise RuntimeError("Cannot reset a running TimeRecorder")
        self._delta = 0.0

    @property
    def delta_t(self) -> float:
        return self._delta

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, *a):
        self.stop()
        return False

    def __repr__(self):
        return f"TimeRecorder(delta_t={self._delta:.6f}s)"
