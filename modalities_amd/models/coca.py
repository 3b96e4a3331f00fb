"""CoCa: Contrastive Captioner (capability parity with reference
src/modalities/models/coca/ — vision encoder + unimodal text decoder +
multimodal decoder with cross-attention + attention pooling; trained with
contrastive (NCE) + captioning losses)."""

from typing import Optional

import torch
import torch.nn as nn

from modalities_amd.models.model import NNModel
from modalities_amd.models.vision_transformer import VisionTransformer
from modalities_amd.nn.attention import AttentionType, MultiHeadAttention
from modalities_amd.nn.mlp import MLP


class AttentionPooling(nn.Module):
    """Pool a variable number of visual tokens into n_queries learned
    queries via cross attention (reference: coca/attention_pooling.py)."""

    def __init__(self, n_embd: int, n_head: int, n_queries: int = 256,
                 bias: bool = True):
        super().__init__()
        self.queries = nn.Parameter(torch.randn(1, n_queries, n_embd) * 0.02)
        self.attn = MultiHeadAttention(n_embd, n_head, bias=bias,
                                       attention_type=AttentionType.CROSS_ATTENTION)
        self.norm = nn.LayerNorm(n_embd)

    def forward(self, vision_tokens: torch.Tensor) -> torch.Tensor:
        q = self.queries.expand(vision_tokens.shape[0], -1, -1)
        return self.norm(self.attn(q, context=vision_tokens))


class TextDecoderBlock(nn.Module):
    def __init__(self, n_embd: int, n_head: int, ffn_hidden: int, bias: bool,
                 dropout: float, with_cross_attention: bool):
        super().__init__()
        self.norm1 = nn.LayerNorm(n_embd)
        self.attn = MultiHeadAttention(
            n_embd, n_head, bias=bias, dropout=dropout,
            attention_type=AttentionType.CAUSAL_SELF_ATTENTION)
        self.with_cross = with_cross_attention
        if with_cross_attention:
            self.norm_cross = nn.LayerNorm(n_embd)
            self.cross_attn = MultiHeadAttention(
                n_embd, n_head, bias=bias, dropout=dropout,
                attention_type=AttentionType.CROSS_ATTENTION)
        self.norm2 = nn.LayerNorm(n_embd)
        self.mlp = MLP(n_embd, ffn_hidden, dropout=dropout, bias=bias)

    def forward(self, x: torch.Tensor,
                context: Optional[torch.Tensor] = None) -> torch.Tensor:
        x = x + self.attn(self.norm1(x))
        if self.with_cross and context is not None:
            x = x + self.cross_attn(self.norm_cross(x), context=context)
        x = x + self.mlp(self.norm2(x))
        return x


class TextDecoder(nn.Module):
    """Token embedding + causal blocks (optionally with cross attention)
    (reference: coca/text_decoder.py, multi_modal_decoder.py)."""

    def __init__(self, vocab_size: int, block_size: int, n_layer: int,
                 n_embd: int, n_head: int, ffn_hidden: int, bias: bool,
                 dropout: float, with_cross_attention: bool,
                 with_embeddings: bool = True):
        super().__init__()
        self.block_size = block_size
        self.wte = nn.Embedding(vocab_size, n_embd) if with_embeddings else None
        self.wpe = nn.Embedding(block_size, n_embd) if with_embeddings else None
        self.drop = nn.Dropout(dropout)
        self.blocks = nn.ModuleList([
            TextDecoderBlock(n_embd, n_head, ffn_hidden, bias, dropout,
                             with_cross_attention)
            for _ in range(n_layer)])

    def forward(self, x: torch.Tensor,
                context: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self.wte is not None:
            pos = torch.arange(x.shape[1], device=x.device)
            x = self.drop(self.wte(x) + self.wpe(pos))
        for block in self.blocks:
            x = block(x, context=context)
        return x


class CoCa(NNModel):
    """Vision encoder -> attention pooling (contrastive query + caption
    queries); text: unimodal decoder stack then multimodal stack with cross
    attention; outputs logits + normalized embeddings for the NCE loss
    (reference: coca/coca_model.py)."""

    def __init__(self, prediction_key: str = "logits",
                 vision_embd_prediction_key: str = "vision_embeddings",
                 text_embd_prediction_key: str = "text_embeddings",
                 vision_cls_prediction_key: str = "vision_cls",
                 text_cls_prediction_key: str = "text_cls",
                 vision_sample_key: str = "images",
                 text_sample_key: str = "input_ids",
                 vocab_size: int = 50304, text_block_size: int = 256,
                 n_layer_text: int = 4, n_layer_multimodal_text: int = 4,
                 n_head: int = 8, n_embd: int = 512, ffn_hidden: int = 2048,
                 bias: bool = True, dropout: float = 0.0,
                 n_pool_head: int = 8, n_vision_queries: int = 64,
                 vision_encoder_config: Optional[dict] = None):
        super().__init__()
        self.prediction_key = prediction_key
        self.vision_cls_prediction_key = vision_cls_prediction_key
        self.text_cls_prediction_key = text_cls_prediction_key
        self.vision_sample_key = vision_sample_key
        self.text_sample_key = text_sample_key

        vcfg = dict(sample_key=vision_sample_key, prediction_key="tokens",
                    n_classes=None, img_size=64, n_layer=2, n_head=n_head,
                    n_embd=n_embd, ffn_hidden=ffn_hidden, patch_size=16,
                    patch_stride=16, add_cls_token=False)
        vcfg.update(vision_encoder_config or {})
        self.vision_encoder = VisionTransformer(**vcfg)
        # n_queries + 1: the extra query becomes the contrastive cls token
        self.attn_pool = AttentionPooling(n_embd, n_pool_head,
                                          n_queries=n_vision_queries + 1,
                                          bias=bias)
        self.text_decoder = TextDecoder(
            vocab_size, text_block_size, n_layer_text, n_embd, n_head,
            ffn_hidden, bias, dropout, with_cross_attention=False)
        self.multimodal_decoder = TextDecoder(
            vocab_size, text_block_size, n_layer_multimodal_text, n_embd,
            n_head, ffn_hidden, bias, dropout, with_cross_attention=True,
            with_embeddings=False)
        self.norm = nn.LayerNorm(n_embd)
        self.lm_head = nn.Linear(n_embd, vocab_size, bias=False)
        self.logit_scale = nn.Parameter(torch.tensor(2.6592))  # ln(1/0.07)

    def _encode_vision(self, images: torch.Tensor):
        tokens = self.vision_encoder.forward_images(images)
        pooled = self.attn_pool(tokens)                  # [B, nq+1, C]
        vision_cls = torch.nn.functional.normalize(pooled[:, -1], dim=-1)
        return pooled[:, :-1], vision_cls

    def _encode_text(self, input_ids: torch.Tensor):
        x = self.text_decoder(input_ids)
        # last token embedding = contrastive text cls
        text_cls = torch.nn.functional.normalize(x[:, -1], dim=-1)
        return x, text_cls

    def forward(self, inputs: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        vision_context, vision_cls = self._encode_vision(
            inputs[self.vision_sample_key])
        text_hidden, text_cls = self._encode_text(inputs[self.text_sample_key])
        x = self.multimodal_decoder(text_hidden, context=vision_context)
        logits = self.lm_head(self.norm(x))
        return {self.prediction_key: logits,
                self.vision_cls_prediction_key: vision_cls,
                self.text_cls_prediction_key: text_cls,
                "logit_scale": self.logit_scale.exp()}
