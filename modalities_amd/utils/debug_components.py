"""Debugging components (capability parity with reference
src/modalities/utils/debug_components.py:9-60 and the
"debugging_enriched" model wrapper, model_factory.py:410-592): NaN/inf
forward hooks, per-module tensor-stats JSONL dumps, deterministic mode."""

import json
from pathlib import Path
from typing import Optional

import torch
import torch.nn as nn


def enable_deterministic_mode(seed: int = 0) -> None:
    """Deterministic kernels + seeded RNG (reference deterministic-CUDA
    context)."""
    import os
    torch.manual_seed(seed)
    os.environ.setdefault("CUBLAS_WORKSPACE_CONFIG", ":4096:8")
    torch.use_deterministic_algorithms(True, warn_only=True)


def _iter_tensors(obj):
    if isinstance(obj, torch.Tensor):
        yield obj
    elif isinstance(obj, (tuple, list)):
        for o in obj:
            yield from _iter_tensors(o)
    elif isinstance(obj, dict):
        for o in obj.values():
            yield from _iter_tensors(o)


class NaNDetectionHook:
    """Raises on the first NaN/inf in any module output."""

    def __init__(self, module_name: str):
        self.module_name = module_name

    def __call__(self, module, args, output):
        for t in _iter_tensors(output):
            if t.is_floating_point() and not torch.isfinite(t).all():
                bad = (~torch.isfinite(t)).sum().item()
                raise RuntimeError(
                    f"Non-finite values ({bad} elements) in output of "
                    f"{self.module_name} ({type(module).__name__})")


class TensorStatsHook:
    """Appends per-forward output stats to a JSONL file (reference
    tensor-stats dump hooks, model_factory.py:410-592)."""

    def __init__(self, module_name: str, out_path: Path):
        self.module_name = module_name
        self.out_path = Path(out_path)
        self._step = 0

    def __call__(self, module, args, output):
        self._step += 1
        for i, t in enumerate(_iter_tensors(output)):
            if not t.is_floating_point():
                continue
            tf = t.detach().float()
            rec = {"module": self.module_name, "step": self._step, "tensor": i,
                   "shape": list(t.shape), "mean": tf.mean().item(),
                   "std": tf.std().item() if tf.numel() > 1 else 0.0,
                   "absmax": tf.abs().max().item()}
            with self.out_path.open("a") as f:
                f.write(json.dumps(rec) + "\n")


def register_nan_hooks(model: nn.Module,
                       module_name_filter: Optional[str] = None) -> int:
    """Register NaN detection on matching modules; returns hook count."""
    n = 0
    for name, module in model.named_modules():
        if module_name_filter and module_name_filter not in name:
            continue
        module.register_forward_hook(NaNDetectionHook(name or "<root>"))
        n += 1
    return n


def register_tensor_stats_hooks(model: nn.Module, out_path: Path,
                                module_name_filter: Optional[str] = None) -> int:
    n = 0
    for name, module in model.named_modules():
        if module_name_filter and module_name_filter not in name:
            continue
        module.register_forward_hook(TensorStatsHook(name or "<root>", out_path))
        n += 1
    return n


def get_debugging_enriched_model(model: nn.Module, log_dir: Path,
                                 nan_hooks: bool = True,
                                 tensor_stats: bool = False,
                                 module_name_filter: Optional[str] = None
                                 ) -> nn.Module:
    """Reference model_factory.get_debugging_enriched_model analog."""
    log_dir = Path(log_dir)
    log_dir.mkdir(parents=True, exist_ok=True)
    if nan_hooks:
        register_nan_hooks(model, module_name_filter)
    if tensor_stats:
        register_tensor_stats_hooks(model, log_dir / "tensor_stats.jsonl",
                                    module_name_filter)
    return model
