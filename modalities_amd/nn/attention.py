"""Generic multi-head attention with causal / non-causal / cross modes
(capability parity with reference src/modalities/nn/attention.py:26-115).
Used by the vision / multimodal models; the LM hot path uses the fused K1
kernel in models/gpt2.py instead."""

from enum import Enum
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


class AttentionType(str, Enum):
    CAUSAL_SELF_ATTENTION = "causal_self_attention"
    NON_CAUSAL_SELF_ATTENTION = "non_causal_self_attention"
    CROSS_ATTENTION = "cross_attention"


class AttentionConfig:
    def __init__(self, attention_engine_type: str = "pytorch_flash_attention"):
        self.attention_engine_type = attention_engine_type


class MultiHeadAttention(nn.Module):
    def __init__(self, n_embd: int, n_head: int, bias: bool = True,
                 dropout: float = 0.0,
                 attention_type: AttentionType = AttentionType.NON_CAUSAL_SELF_ATTENTION,
                 attention_config: Optional[AttentionConfig] = None):
        super().__init__()
        if n_embd % n_head:
            raise ValueError("n_embd must be divisible by n_head")
        self.n_head = n_head
        self.head_dim = n_embd // n_head
        self.attention_type = attention_type
        self.wq = nn.Linear(n_embd, n_embd, bias=bias)
        self.wk = nn.Linear(n_embd, n_embd, bias=bias)
        self.wv = nn.Linear(n_embd, n_embd, bias=bias)
        self.c_proj = nn.Linear(n_embd, n_embd, bias=bias)
        self.dropout = dropout
        self.resid_dropout = nn.Dropout(dropout)

    def forward(self, x: torch.Tensor,
                context: Optional[torch.Tensor] = None) -> torch.Tensor:
        kv_src = context if (context is not None
                             and self.attention_type == AttentionType.CROSS_ATTENTION) \
            else x
        B, T, C = x.shape
        S = kv_src.shape[1]
        q = self.wq(x).view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        k = self.wk(kv_src).view(B, S, self.n_head, self.head_dim).transpose(1, 2)
        v = self.wv(kv_src).view(B, S, self.n_head, self.head_dim).transpose(1, 2)
        y = F.scaled_dot_product_attention(
            q, k, v,
            dropout_p=self.dropout if self.training else 0.0,
            is_causal=self.attention_type == AttentionType.CAUSAL_SELF_ATTENTION)
        y = y.transpose(1, 2).reshape(B, T, C)
        return self.resid_dropout(self.c_proj(y))
