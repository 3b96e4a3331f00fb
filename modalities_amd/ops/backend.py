"""MI355X-native addition (no reference analog): HIP extension loading and dispatch policy.

The extension is built IN-TREE (modalities_amd/ops/_hip_ops.so) by
``modalities_amd.ops.build.build_extension()`` (driven from __graft_entry__)
so the .so travels with repo snapshots to GPU boxes. gfx950-only; no Triton,
no CUDA shims, no multi-backend dispatch.
"""

import importlib
import os
import sys
from pathlib import Path

import torch

_EXT = None
_TRIED = False

EXT_NAME = "_hip_ops"
EXT_DIR = Path(__file__).parent


def hip_ext():
    """Return the loaded HIP extension module, or None if unavailable."""
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    so_glob = sorted(EXT_DIR.glob(f"{EXT_NAME}*.so"))
    if not so_glob:
        return None
    try:
        # Import as a proper submodule so torch extension registration works.
        if str(EXT_DIR) not in sys.path:
            sys.path.insert(0, str(EXT_DIR))
        _EXT = importlib.import_module(EXT_NAME)
    except Exception as e:  # pragma: no cover - load failure is environmental
        if torch.cuda.is_available():
            raise RuntimeError(
                f"HIP op extension exists at {so_glob[0]} but failed to load: {e}"
            ) from e
        _EXT = None
    return _EXT


def hip_available() -> bool:
    return torch.cuda.is_available() and hip_ext() is not None


def require_hip(op_name: str):
    """Return the extension; raise loudly if we're on GPU without it."""
    ext = hip_ext()
    if ext is None:
        raise RuntimeError(
            f"modalities_amd op {op_name!r} was called with device tensors but the "
            f"HIP extension (modalities_amd/ops/{EXT_NAME}.so) is not built. "
            f"Run `python -c 'import __graft_entry__; __graft_entry__.build()'` "
            f"or `python -m modalities_amd.ops.build`. Refusing to silently fall "
            f"back to eager PyTorch on a GPU."
        )
    return ext


def use_hip(*tensors) -> bool:
    """True if these tensors should go down the HIP kernel path."""
    on_dev = all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_dev:
        return False
    if os.environ.get("MODALITIES_AMD_FORCE_EAGER") == "1":
        return False
    require_hip("op")  # loud failure if missing
    return True
