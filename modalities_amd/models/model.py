"""Model base classes (capability parity with reference
src/modalities/models/model.py:26-170)."""

from abc import abstractmethod
from typing import Optional

import torch
import torch.nn as nn

from modalities_amd.batch import DatasetBatch, InferenceResultBatch
from modalities_amd.ops import silu_mul, silu_mul_joint
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

    MI355X-first layout: ``packed=True`` fuses the W and V up-projections
    into ONE [n_embd, 2*hidden] GEMM (half the up-projection launches at a
    better hipBLASLt shape); the silu*mul epilogue reads the joint output
    and its backward writes one joint gradient buffer (K6 joint kernel).
    ``packed=False`` keeps the reference's separate W/V parameters."""

    def __init__(self, n_embd: int, ffn_hidden: int, bias: bool = False,
                 packed: bool = True):
        super().__init__()
        self.hidden_dim = self._get_hidden_dim(ffn_hidden)
        self.packed = packed
        if packed:
            self.Wv = TwoStreamLinear(n_embd, 2 * self.hidden_dim, bias=bias)
        else:
            self.W = TwoStreamLinear(n_embd, self.hidden_dim, bias=bias)
            self.V = TwoStreamLinear(n_embd, self.hidden_dim, bias=bias)
        self.W_2 = TwoStreamLinear(self.hidden_dim, n_embd, bias=bias)

    @staticmethod
    def _get_hidden_dim(ffn_hidden: int) -> int:
        # 2/3 * ffn_hidden, rounded up to a multiple of 256.
        return 256 * ((int(2 * ffn_hidden / 3) + 256 - 1) // 256)

    # gate/up views of the packed weight (conversion / tests / TP introspect
    # the reference's separate-W/V naming through these)
    @property
    def W_weight(self) -> torch.Tensor:
        return self.Wv.weight[:self.hidden_dim] if self.packed else self.W.weight

    @property
    def V_weight(self) -> torch.Tensor:
        return self.Wv.weight[self.hidden_dim:] if self.packed else self.V.weight

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.packed:
            return self.W_2(silu_mul_joint(self.Wv(x)))
        return self.W_2(silu_mul(self.W(x), self.V(x)))


def model_predict_batch(model: nn.Module, batch: DatasetBatch) -> InferenceResultBatch:
    forward_result = model(batch.samples)
    return InferenceResultBatch(targets=batch.targets, predictions=forward_result)
