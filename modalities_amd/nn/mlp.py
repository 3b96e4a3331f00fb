"""Generic MLP (capability parity with reference src/modalities/nn/mlp.py)."""

from typing import Callable, Optional

import torch
import torch.nn as nn


class MLP(nn.Module):
    def __init__(self, in_features: int, hidden_features: Optional[int] = None,
                 out_features: Optional[int] = None, bias: bool = True,
                 dropout: float = 0.0,
                 act_fn: Callable[[], nn.Module] = nn.GELU):
        super().__init__()
        out_features = out_features or in_features
        hidden_features = hidden_features or 4 * in_features
        self.fc1 = nn.Linear(in_features, hidden_features, bias=bias)
        self.act = act_fn()
        self.fc2 = nn.Linear(hidden_features, out_features, bias=bias)
        self.drop = nn.Dropout(dropout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.drop(self.fc2(self.act(self.fc1(x))))
