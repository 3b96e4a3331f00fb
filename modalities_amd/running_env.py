"""Process/device runtime: RCCL process-group init context manager.

Capability parity with the reference CudaEnv (reference:
src/modalities/running_env/cuda_env.py:15-67). On ROCm the "nccl" backend of
torch.distributed IS RCCL over xGMI; on CPU-only hosts (tests) gloo is used.
"""

import os
import warnings
from datetime import timedelta
from enum import Enum

import torch
import torch.distributed as dist


class ProcessGroupBackendType(str, Enum):
    nccl = "nccl"   # RCCL on ROCm
    gloo = "gloo"   # CPU / tests


class MixedPrecisionSettings(str, Enum):
    BF_16 = "BF_16"
    FP_16 = "FP_16"
    FP_32 = "FP_32"

    @property
    def dtype(self) -> torch.dtype:
        return {"BF_16": torch.bfloat16, "FP_16": torch.float16,
                "FP_32": torch.float32}[self.value]


class DistEnv:
    """Context manager: init_process_group + device selection.

    Reads RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT from the
    environment (torchrun contract)."""

    def __init__(self, process_group_backend: ProcessGroupBackendType
                 = ProcessGroupBackendType.nccl, timeout_s: int = 600):
        if isinstance(process_group_backend, str):
            process_group_backend = ProcessGroupBackendType(process_group_backend)
        self.backend = process_group_backend
        self.timeout_s = timeout_s
        self.local_rank = int(os.environ.get("LOCAL_RANK", 0))

    def __enter__(self) -> "DistEnv":
        backend = self.backend.value
        if backend == "nccl" and not torch.cuda.is_available():
            warnings.warn("CUDA/ROCm unavailable; falling back to gloo backend")
            backend = "gloo"
        launched_by_torchrun = "RANK" in os.environ and "WORLD_SIZE" in os.environ
        if not dist.is_initialized() and launched_by_torchrun:
            dist.init_process_group(backend, timeout=timedelta(seconds=self.timeout_s))
        if torch.cuda.is_available():
            torch.cuda.set_device(self.local_rank)
        return self

    def __exit__(self, exc_type, exc, tb):
        if exc_type is torch.cuda.OutOfMemoryError:
            torch.cuda.empty_cache()
        if dist.is_initialized():
            dist.destroy_process_group()
        return False


# Backwards-friendly alias matching the reference's name
CudaEnv = DistEnv


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def global_rank() -> int:
    return dist.get_rank() if is_dist() else 0


def world_size() -> int:
    return dist.get_world_size() if is_dist() else 1


def device_for_rank() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
    return torch.device("cpu")


class Reducer:
    """All-reduce helper (reference: running_env/fsdp/reducer.py:7-17)."""

    @staticmethod
    def reduce(tensor: torch.Tensor, operation=None, post_processing_fun=None):
        if is_dist():
            op = operation if operation is not None else dist.ReduceOp.SUM
            dist.all_reduce(tensor, op=op)
        if post_processing_fun is not None:
            tensor = post_processing_fun(tensor)
        return tensor
