"""MFU calculation with MI355X peak-FLOPS entries.

Capability parity with the reference GPT2MFUCalculator (reference:
src/modalities/utils/mfu.py:150-197; flops/token = 6N + 12*L*s*h). The peak
table carries MI355X (gfx950): 2.5 PF dense bf16 (AMD's headline ~5 PF
includes 2:1 structured sparsity — never price against that), and keeps the
NVIDIA entries so published baselines can be re-expressed."""

from typing import Optional

import torch

# 16-bit dense peak FLOP/s per device
PEAK_FLOPS_16BIT = {
    "MI355X": 2.5e15,
    "MI350X": 2.3e15,
    "MI300X": 1.3e15,
    "A100": 312e12,
    "H100": 989e12,
    "GH200": 989e12,
    "B200": 2.25e15,
}


def detect_device_peak_flops() -> Optional[float]:
    if not torch.cuda.is_available():
        return None
    name = torch.cuda.get_device_name(0).upper()
    for key, peak in PEAK_FLOPS_16BIT.items():
        if key in name:
            return peak
    if "GFX950" in name or "355" in name:
        return PEAK_FLOPS_16BIT["MI355X"]
    if torch.version.hip is not None:
        # unknown ROCm device string: this framework targets MI355X
        return PEAK_FLOPS_16BIT["MI355X"]
    return None


class GPT2MFUCalculator:
    """theoretical_flops_per_token = 6N + 12*L*s*h (weight + attention)."""

    def __init__(self, n_layer: int, sequence_length: int, n_embd: int,
                 world_size: int, num_params: int,
                 peak_flops_per_device: Optional[float] = None):
        self.n_layer = n_layer
        self.sequence_length = sequence_length
        self.n_embd = n_embd
        self.world_size = world_size
        self.num_params = num_params
        self.peak = peak_flops_per_device or detect_device_peak_flops()
        self.flops_per_token = 6 * num_params + 12 * n_layer * sequence_length * n_embd

    def compute(self, num_samples_per_second: torch.Tensor) -> torch.Tensor:
        """num_samples_per_second: GLOBAL throughput -> global MFU in [0,1]."""
        if self.peak is None:
            return torch.tensor(-1.0)
        tokens_per_second = num_samples_per_second * self.sequence_length
        achieved = tokens_per_second * self.flops_per_token
        return achieved / (self.world_size * self.peak)
