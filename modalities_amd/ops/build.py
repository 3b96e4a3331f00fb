"""MI355X-native addition (no reference analog): Build the HIP op extension IN-TREE (modalities_amd/ops/_hip_ops.so).

Drives hipcc directly (native HIP source, no hipify) with torch's include/
lib paths; gfx950 only. The .so lands next to this file so repo snapshots
carry it to GPU boxes. Usage: python -m modalities_amd.ops.build
"""

import subprocess
import sys
import sysconfig
from pathlib import Path

import torch
from torch.utils import cpp_extension

OPS_DIR = Path(__file__).parent
CSRC = OPS_DIR / "csrc"
OUT = OPS_DIR / "_hip_ops.so"

SOURCES = [
    "rms_norm.hip",
    "elementwise.hip",
    "cross_entropy.hip",
    "adamw.hip",
    "attention_fwd.hip",
    "attention_bwd.hip",
    "attention_v2.hip",
    "bindings.cpp",
]


def build_extension(verbose: bool = True, arch: str = "gfx950") -> Path:
    torch_lib = Path(torch.__file__).parent / "lib"
    includes = cpp_extension.include_paths() + [sysconfig.get_paths()["include"]]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)

    srcs = [str(CSRC / s) for s in SOURCES]
    newest_src = max((CSRC / s).stat().st_mtime for s in SOURCES)
    newest_src = max(newest_src, (CSRC / "common.h").stat().st_mtime)
    if OUT.exists() and OUT.stat().st_mtime > newest_src:
        if verbose:
            print(f"[build] {OUT} up to date")
        return OUT

    cmd = (
        ["hipcc", f"--offload-arch={arch}", "-O3", "-std=c++17", "-fPIC",
         "-shared", "-DTORCH_EXTENSION_NAME=_hip_ops",
         f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
         "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
         "-Wno-unused-result", "-fno-gpu-rdc",
         "-x", "hip"]
        + [f"-I{p}" for p in includes]
        + srcs
        + [f"-L{torch_lib}", "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10",
           "-lc10_hip", "-ltorch_python", "-lamdhip64",
           f"-Wl,-rpath,{torch_lib}", "-o", str(OUT)]
    )
    if verbose:
        print("[build]", " ".join(cmd))
    result = subprocess.run(cmd, capture_output=True, text=True)
    if result.returncode != 0:
        sys.stderr.write(result.stdout[-4000:] if result.stdout else "")
        sys.stderr.write(result.stderr[-8000:] if result.stderr else "")
        raise RuntimeError(f"hipcc failed with code {result.returncode}")
    if verbose and result.stderr:
        print(result.stderr[-2000:])
    return OUT


if __name__ == "__main__":
    build_extension()
    print(f"built {OUT}")
