"""modalities_amd — an MI355X-native distributed LLM pretraining framework.

A from-scratch framework with the capabilities of Modalities/modalities
(reference: /root/reference, pure-Python orchestration over CUDA/NCCL deps),
re-designed for AMD MI355X (gfx950, CDNA4):

- PyTorch-ROCm orchestration; one process per GPU over RCCL (xGMI intra-node).
- Hand-written HIP/CDNA4 kernels (MFMA + LDS tiling) for the hot ops in
  ``modalities_amd.ops`` (flash attention, RMSNorm, RoPE, SwiGLU, fused
  cross-entropy, fused AdamW, multi-tensor grad clip).
- An explicit sharded-data-parallel engine (``modalities_amd.parallel.fsdp``)
  built on bucketed RCCL all-gather / reduce-scatter with HIP-stream overlap,
  sized for 7x153 GB/s xGMI links and 288 GB HBM3E per GPU.
- YAML-driven component registry compatible with the reference's
  component_key/variant_key config namespace (reference:
  src/modalities/registry/components.py:187-531).
"""

__version__ = "0.1.0"

def version() -> str:
    return __version__
