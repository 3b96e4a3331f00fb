"""MI355X-native op library: HIP/CDNA4 kernels with torch reference fallbacks.

Replaces the dependency-provided kernels the reference framework leans on
(reference: SURVEY.md §2.4 K1-K11 — flash-attn wheel, cuBLAS epilogues,
torch SDPA/LayerNorm/CE/AdamW) with hand-written gfx950 HIP kernels:

- K1  flash_attention   : MFMA/LDS-tiled causal attention fwd+bwd, GQA-aware
- K4  rope              : fused rotate-half RoPE on q,k
- K5  rms_norm          : wave-level reduction norm fwd+bwd
- K6  silu_mul          : SwiGLU gate epilogue
- K8  cross_entropy     : fused log-softmax + NLL over the vocab
- K9  fused_adamw       : multi-tensor AdamW update
- K10 grad norm/clip    : multi-tensor L2 norm + scale

Dispatch policy: on ROCm devices the HIP extension is REQUIRED — a CUDA
(ROCm) tensor reaching an op without the extension raises, so GPU runs can
never silently fall back to eager PyTorch. On CPU the ops run a plain
PyTorch implementation (used for tests and as the numerics reference).
"""

from modalities_amd.ops.backend import (  # noqa: F401
    hip_available,
    hip_ext,
    require_hip,
)
from modalities_amd.ops.rms_norm import rms_norm  # noqa: F401
from modalities_amd.ops.rope import precompute_rope_cos_sin, rope_apply  # noqa: F401
from modalities_amd.ops.swiglu import silu_mul, silu_mul_joint  # noqa: F401
from modalities_amd.ops.cross_entropy import fused_cross_entropy  # noqa: F401
from modalities_amd.ops.attention import flash_attention  # noqa: F401
from modalities_amd.ops.adamw import fused_adamw_step  # noqa: F401
