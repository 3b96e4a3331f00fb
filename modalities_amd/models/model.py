"""Model base classes (capability parity with reference
src/modalities/models/model.py:26-170)."""

from abc import abstractmethod
from typing import Optional

import torch
import torch.nn as nn

from modalities_amd.batch import DatasetBatch, InferenceResultBatch
from modalities_amd.ops import silu_mul
from modalities_amd.ops.linear import TwoStreamLinear

WeightDecayGroups = dict[str, list[str]]


class NNModel(nn.Module):
    """Base model carrying weight-decay group metadata (regex fragments over
    parameter names) consumed by the optimizer factory."""

    def __init__(self, seed: Optional[int] = None,
                 weight_decay_groups: Optional[WeightDecayGroups] = None):
        if seed is not None:
            torch.manual_seed(seed)
        self._weight_decay_groups = weight_decay_groups if weight_decay_groups else {}
        super().__init__()

    @property
    def weight_decay_groups(self) -> WeightDecayGroups:
        return self._weight_decay_groups

    @abstractmethod
    def forward(self, inputs: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        raise NotImplementedError

    def get_parameters(self) -> dict[str, torch.Tensor]:
        return {name: param for name, param in self.named_parameters()}


class SwiGLU(nn.Module):
    """SwiGLU MLP: W_2(silu(W x) * V x), hidden = 256-rounded 2/3 * 4 * d
    for even TP sharding (reference: src/modalities/models/model.py:116-142).
    The silu*mul epilogue runs as a single HIP kernel on device (K6)."""

    def __init__(self, n_embd: int, ffn_hidden: int, bias: bool = False):
        super().__init__()
        self.hidden_dim = self._get_hidden_dim(ffn_hidden)
        self.W = TwoStreamLinear(n_embd, self.hidden_dim, bias=bias)
        self.V = TwoStreamLinear(n_embd, self.hidden_dim, bias=bias)
        self.W_2 = TwoStreamLinear(self.hidden_dim, n_embd, bias=bias)

    @staticmethod
    def _get_hidden_dim(ffn_hidden: int) -> int:
        # 2/3 * ffn_hidden, rounded up to a multiple of 256.
        return 256 * ((int(2 * ffn_hidden / 3) + 256 - 1) // 256)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.W_2(silu_mul(self.W(x), self.V(x)))


def model_predict_batch(model: nn.Module, batch: DatasetBatch) -> InferenceResultBatch:
    forward_result = model(batch.samples)
    return InferenceResultBatch(targets=batch.targets, predictions=forward_result)
