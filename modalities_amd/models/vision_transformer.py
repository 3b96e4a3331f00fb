"""Vision Transformer (capability parity with reference
src/modalities/models/vision_transformer/vision_transformer_model.py:13-299):
conv patch embedding, pre-norm encoder blocks, optional class token or mean
pooling head."""

from typing import Annotated, Optional

import torch
import torch.nn as nn
from pydantic import BaseModel, Field

from modalities_amd.models.model import NNModel
from modalities_amd.nn.attention import AttentionType, MultiHeadAttention
from modalities_amd.nn.mlp import MLP


class VisionTransformerConfig(BaseModel):
    sample_key: str = "images"
    prediction_key: str = "cls_token"
    img_size: Annotated[int, Field(gt=0)] = 224
    n_classes: Optional[int] = 1000
    n_layer: Annotated[int, Field(gt=0)] = 12
    n_head: Annotated[int, Field(gt=0)] = 8
    n_embd: Annotated[int, Field(gt=0)] = 768
    ffn_hidden: Annotated[int, Field(gt=0)] = 3072
    dropout: float = 0.0
    patch_size: Annotated[int, Field(gt=0)] = 16
    patch_stride: Annotated[int, Field(gt=0)] = 16
    n_img_channels: Annotated[int, Field(gt=0)] = 3
    add_cls_token: bool = True
    bias: bool = True


class ImagePatchEmbedding(nn.Module):
    def __init__(self, n_img_channels: int, n_embd: int, patch_size: int,
                 patch_stride: int, add_cls_token: bool):
        super().__init__()
        self.conv = nn.Conv2d(n_img_channels, n_embd, kernel_size=patch_size,
                              stride=patch_stride)
        self.cls_token = nn.Parameter(torch.zeros(1, 1, n_embd)) \
            if add_cls_token else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.conv(x)                      # [B, C, H', W']
        x = x.flatten(2).transpose(1, 2)      # [B, N, C]
        if self.cls_token is not None:
            cls = self.cls_token.expand(x.shape[0], -1, -1)
            x = torch.cat([cls, x], dim=1)
        return x


class VisionTransformerBlock(nn.Module):
    def __init__(self, n_embd: int, n_head: int, ffn_hidden: int, bias: bool,
                 dropout: float):
        super().__init__()
        self.norm1 = nn.LayerNorm(n_embd)
        self.attention = MultiHeadAttention(
            n_embd, n_head, bias=bias, dropout=dropout,
            attention_type=AttentionType.NON_CAUSAL_SELF_ATTENTION)
        self.norm2 = nn.LayerNorm(n_embd)
        self.mlp = MLP(n_embd, ffn_hidden, dropout=dropout, bias=bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.attention(self.norm1(x))
        x = x + self.mlp(self.norm2(x))
        return x


class VisionTransformer(NNModel):
    def __init__(self, sample_key: str = "images",
                 prediction_key: str = "cls_token", img_size: int = 224,
                 n_classes: Optional[int] = 1000, n_layer: int = 12,
                 n_head: int = 8, n_embd: int = 768, ffn_hidden: int = 3072,
                 dropout: float = 0.0, patch_size: int = 16,
                 patch_stride: int = 16, n_img_channels: int = 3,
                 add_cls_token: bool = True, bias: bool = True):
        super().__init__()
        self.sample_key = sample_key
        self.prediction_key = prediction_key
        self.embedding_fn = ImagePatchEmbedding(n_img_channels, n_embd,
                                                patch_size, patch_stride,
                                                add_cls_token)
        n_side = (img_size - patch_size) // patch_stride + 1
        self.block_size = n_side * n_side + int(add_cls_token)
        self.positional_embedding_fn = nn.Embedding(self.block_size, n_embd)
        self.dropout = nn.Dropout(dropout)
        self.blocks = nn.ModuleList([
            VisionTransformerBlock(n_embd, n_head, ffn_hidden, bias, dropout)
            for _ in range(n_layer)])
        self.norm = nn.LayerNorm(n_embd)
        self.has_cls_token = add_cls_token
        self.head = nn.Linear(n_embd, n_classes, bias=bias) \
            if n_classes is not None else None

    def forward_images(self, images: torch.Tensor) -> torch.Tensor:
        x = self.embedding_fn(images)
        pos = torch.arange(x.shape[1], device=x.device)
        x = self.dropout(x + self.positional_embedding_fn(pos))
        for block in self.blocks:
            x = block(x)
        return x

    def forward(self, inputs: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        x = self.forward_images(inputs[self.sample_key])
        x = self.norm(x)
        if self.has_cls_token:
            x = x[:, 0]
        else:
            x = x.mean(dim=1)
        if self.head is not None:
            x = self.head(x)
        return {self.prediction_key: x}
