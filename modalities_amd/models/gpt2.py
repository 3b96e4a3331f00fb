"""GPT2/Llama-style decoder LM, MI355X-first.

Capability parity with the reference GPT2LLM (reference:
src/modalities/models/gpt2/gpt2_model.py:816-1020 — GQA with separate q/k/v
projections, RoPE, optional QK-norm, pre-norm blocks, SwiGLU or GELU MLP,
ABSOLUTE/NOPE/ROTARY positional modes, weight tying, divisibility checks),
but the hot path is built on the modalities_amd HIP ops: K1 flash attention
(GQA without KV repetition), K4 fused RoPE, K5 RMSNorm, K6 SwiGLU epilogue.
Attention head_dim is sized for MFMA tiles (multiples of 16; 64/128 fast)."""

import math
from enum import Enum
from typing import Annotated, Optional

import torch
import torch.nn as nn
from pydantic import BaseModel, Field, model_validator

from modalities_amd.models.model import NNModel, SwiGLU
from modalities_amd.ops import flash_attention, precompute_rope_cos_sin, rope_apply
from modalities_amd.ops.attention import fused_qkv_rope_attention
from modalities_amd.ops.backend import use_hip
from modalities_amd.ops.linear import TwoStreamLinear
from modalities_amd.ops.rms_norm import RMSNorm


class PositionTypes(str, Enum):
    ABSOLUTE = "ABSOLUTE"
    NOPE = "NOPE"  # NOPE == rely on RoPE in attention (or none at all)


class AttentionImplementation(str, Enum):
    HIP_FLASH = "hip_flash"   # K1 HIP kernel on device; fp32 torch ref on CPU
    PYTORCH_FLASH = "pytorch_flash"  # torch SDPA (debug/comparison path)
    MANUAL = "manual"         # O(T^2) eager fallback (debug)


class ActivationType(str, Enum):
    GELU = "gelu"
    SWIGLU = "swiglu"


class QueryKeyValueTransformType(str, Enum):
    ROTARY = "RotaryTransform"
    IDENTITY = "IdentityTransform"


class LayerNormVariant(str, Enum):
    RMS_NORM = "rms_norm"
    LAYER_NORM = "layer_norm"


class LayerNormConfig(BaseModel):
    variant: LayerNormVariant = LayerNormVariant.RMS_NORM
    eps: float = 1e-6
    bias: bool = True  # layer_norm only


def make_norm(cfg: LayerNormConfig, ndim: int) -> nn.Module:
    if cfg.variant == LayerNormVariant.RMS_NORM:
        return RMSNorm(ndim, eps=cfg.eps)
    return nn.LayerNorm(ndim, eps=cfg.eps, bias=cfg.bias)


class GPT2LLMConfig(BaseModel):
    sample_key: str = "input_ids"
    prediction_key: str = "logits"
    use_meta_device: bool = False
    poe_type: PositionTypes = PositionTypes.NOPE
    sequence_length: Annotated[int, Field(gt=0)] = 4096
    vocab_size: Annotated[int, Field(gt=0)] = 50304
    n_layer: Annotated[int, Field(gt=0)] = 12
    n_head_q: Annotated[int, Field(gt=0)] = 12
    n_head_kv: Annotated[int, Field(gt=0)] = 12
    n_embd: Annotated[int, Field(gt=0)] = 768
    ffn_hidden: Annotated[int, Field(gt=0)] = 3072
    dropout: Annotated[float, Field(ge=0.0)] = 0.0
    bias: bool = False
    attention_implementation: AttentionImplementation = AttentionImplementation.HIP_FLASH
    activation_type: ActivationType = ActivationType.SWIGLU
    # MI355X-first: pack the SwiGLU W/V up-projections into one GEMM
    packed_swiglu: bool = True
    qkv_transform: QueryKeyValueTransformType = QueryKeyValueTransformType.ROTARY
    rope_base: float = 10000.0
    use_weight_tying: bool = False
    use_qk_norm: bool = False
    # Single [h, h + 2*kv] projection instead of separate q/k/v GEMMs —
    # one bigger hipBLASLt GEMM per block side (~0.13 ms/layer fwd on the
    # 2.7B shapes). MI355X-first option; off by default for weight-name
    # compatibility (TP sharding requires the unfused layout).
    fused_qkv: bool = False
    attention_norm_config: LayerNormConfig = LayerNormConfig()
    ffn_norm_config: LayerNormConfig = LayerNormConfig()
    lm_head_norm_config: LayerNormConfig = LayerNormConfig()
    seed: Optional[int] = None

    @model_validator(mode="after")
    def _validate(self):
        if self.n_head_q % self.n_head_kv != 0:
            raise ValueError("n_head_q must be divisible by n_head_kv")
        if self.n_embd % self.n_head_q != 0:
            raise ValueError("n_embd must be divisible by n_head_q")
        return self


class CausalSelfAttention(nn.Module):
    """Separate q/k/v projections (GQA), fused RoPE, optional QK-norm, flash
    attention via K1 (no KV-head repetition on device)."""

    def __init__(self, n_embd: int, n_head_q: int, n_head_kv: int, bias: bool,
                 dropout: float, attention_impl: AttentionImplementation,
                 use_qk_norm: bool, norm_cfg: LayerNormConfig,
                 fused_qkv: bool = False):
        super().__init__()
        self.n_head_q = n_head_q
        self.n_head_kv = n_head_kv
        self.head_dim = n_embd // n_head_q
        self.attention_impl = attention_impl
        self.fused_qkv = fused_qkv
        kv_dim = self.head_dim * n_head_kv
        if fused_qkv:
            self.qkv_attn = TwoStreamLinear(n_embd, n_embd + 2 * kv_dim, bias=bias)
        else:
            self.q_attn = TwoStreamLinear(n_embd, n_embd, bias=bias)
            self.k_attn = TwoStreamLinear(n_embd, kv_dim, bias=bias)
            self.v_attn = TwoStreamLinear(n_embd, kv_dim, bias=bias)
        self.c_proj = TwoStreamLinear(n_embd, n_embd, bias=bias)
        self.resid_dropout = nn.Dropout(dropout)
        self.dropout = dropout
        if dropout > 0.0 and attention_impl == AttentionImplementation.HIP_FLASH:
            # The K1 flash kernel has no dropout (pretraining runs use 0);
            # configs that DO ask for attention dropout fall back to the
            # SDPA path for the attention op (reference supports dropout in
            # its flash paths, gpt2_model.py:632-655) — everything else
            # (RoPE/norm/projection kernels) stays on the HIP path.
            import warnings
            warnings.warn(
                "attention dropout > 0: using the SDPA attention path "
                "instead of the K1 HIP flash kernel (K1 has no dropout)",
                stacklevel=3)
            self.attention_impl = AttentionImplementation.PYTORCH_FLASH
            attention_impl = AttentionImplementation.PYTORCH_FLASH
        if use_qk_norm:
            self.q_norm = make_norm(norm_cfg, self.head_dim)
            self.k_norm = make_norm(norm_cfg, self.head_dim)
        else:
            self.q_norm = self.k_norm = None

    def forward(self, x: torch.Tensor, rope_cos: Optional[torch.Tensor],
                rope_sin: Optional[torch.Tensor]) -> torch.Tensor:
        B, T, C = x.shape
        kv_dim = self.head_dim * self.n_head_kv
        if self.fused_qkv:
            qkv = self.qkv_attn(x)
            if (self.attention_impl == AttentionImplementation.HIP_FLASH
                    and self.q_norm is None and rope_cos is not None
                    and use_hip(qkv)):
                # fused split+RoPE+attention (joint dqkv assembly in bwd)
                y = fused_qkv_rope_attention(qkv, rope_cos, rope_sin,
                                             self.n_head_q, self.n_head_kv,
                                             self.head_dim)
                return self.resid_dropout(self.c_proj(y.reshape(B, T, C)))
            q, k, v = qkv.split([C, kv_dim, kv_dim], dim=-1)
            q = q.view(B, T, self.n_head_q, self.head_dim)
            k = k.view(B, T, self.n_head_kv, self.head_dim)
            v = v.view(B, T, self.n_head_kv, self.head_dim)
        else:
            q = self.q_attn(x).view(B, T, self.n_head_q, self.head_dim)
            k = self.k_attn(x).view(B, T, self.n_head_kv, self.head_dim)
            v = self.v_attn(x).view(B, T, self.n_head_kv, self.head_dim)
        if self.q_norm is not None:
            q = self.q_norm(q)
            k = self.k_norm(k)
        if rope_cos is not None:
            q = rope_apply(q, rope_cos, rope_sin)
            k = rope_apply(k, rope_cos, rope_sin)
        y = self._attend(q, k, v)
        y = y.reshape(B, T, C)
        return self.resid_dropout(self.c_proj(y))

    def forward_cached(self, x, rope_cos, rope_sin, cache_k, cache_v,
                       pos: int):
        """Incremental-decode step: project the T_new tokens at absolute
        positions [pos, pos+T_new), append K/V into the preallocated cache,
        attend q against the cache prefix with q_offset=pos (the same
        offset-causal contract the CP path uses). Inference-only
        (no grad); MI355X-first addition — the reference re-forwards the
        full context per generated token."""
        B, T, C = x.shape
        kv_dim = self.head_dim * self.n_head_kv
        if self.fused_qkv:
            qkv = self.qkv_attn(x)
            q, k, v = qkv.split([C, kv_dim, kv_dim], dim=-1)
        else:
            q, k, v = self.q_attn(x), self.k_attn(x), self.v_attn(x)
        q = q.view(B, T, self.n_head_q, self.head_dim)
        k = k.view(B, T, self.n_head_kv, self.head_dim)
        v = v.view(B, T, self.n_head_kv, self.head_dim)
        if self.q_norm is not None:
            q = self.q_norm(q)
            k = self.k_norm(k)
        if rope_cos is not None:
            cos, sin = rope_cos[pos:pos + T], rope_sin[pos:pos + T]
            q = rope_apply(q, cos, sin)
            k = rope_apply(k, cos, sin)
        cache_k[:, pos:pos + T] = k
        cache_v[:, pos:pos + T] = v
        k_full = cache_k[:, :pos + T]
        v_full = cache_v[:, :pos + T]
        if self.attention_impl == AttentionImplementation.HIP_FLASH and                 use_hip(q):
            y = flash_attention(q, k_full.contiguous(), v_full.contiguous(),
                                causal=True, q_offset=pos)
        else:
            rep = self.n_head_q // self.n_head_kv
            qt = q.transpose(1, 2)
            kt = k_full.transpose(1, 2).repeat_interleave(rep, dim=1)
            vt = v_full.transpose(1, 2).repeat_interleave(rep, dim=1)
            S = pos + T
            mask = (torch.arange(S, device=x.device)[None, :]
                    <= (pos + torch.arange(T, device=x.device))[:, None])
            y = torch.nn.functional.scaled_dot_product_attention(
                qt, kt, vt, attn_mask=mask)
            y = y.transpose(1, 2)
        return self.c_proj(y.reshape(B, T, C))

    def _attend(self, q, k, v):
        if self.attention_impl == AttentionImplementation.HIP_FLASH:
            return flash_attention(q, k, v, causal=True)
        # debug paths operate in [B, H, T, D]
        rep = self.n_head_q // self.n_head_kv
        qt = q.transpose(1, 2)
        kt = k.transpose(1, 2).repeat_interleave(rep, dim=1)
        vt = v.transpose(1, 2).repeat_interleave(rep, dim=1)
        if self.attention_impl == AttentionImplementation.PYTORCH_FLASH:
            y = torch.nn.functional.scaled_dot_product_attention(
                qt, kt, vt, is_causal=True,
                dropout_p=self.dropout if self.training else 0.0)
        else:  # MANUAL
            att = qt @ kt.transpose(-2, -1) / math.sqrt(self.head_dim)
            mask = torch.ones(qt.shape[2], kt.shape[2], dtype=torch.bool,
                              device=qt.device).tril()
            att = att.masked_fill(~mask, float("-inf")).softmax(-1)
            y = att @ vt
        return y.transpose(1, 2)


class TransformerMLP(nn.Module):
    def __init__(self, n_embd: int, ffn_hidden: int, bias: bool, dropout: float):
        super().__init__()
        self.c_fc = TwoStreamLinear(n_embd, ffn_hidden, bias=bias)
        self.gelu = nn.GELU(approximate="tanh")
        self.c_proj = TwoStreamLinear(ffn_hidden, n_embd, bias=bias)
        self.dropout = nn.Dropout(dropout)

    def forward(self, x):
        return self.dropout(self.c_proj(self.gelu(self.c_fc(x))))


class GPT2Block(nn.Module):
    """Pre-norm transformer block: x + attn(norm(x)); x + mlp(norm(x))."""

    def __init__(self, cfg: GPT2LLMConfig):
        super().__init__()
        self.attention_norm = make_norm(cfg.attention_norm_config, cfg.n_embd)
        self.attn = CausalSelfAttention(
            cfg.n_embd, cfg.n_head_q, cfg.n_head_kv, cfg.bias, cfg.dropout,
            cfg.attention_implementation, cfg.use_qk_norm,
            cfg.attention_norm_config, fused_qkv=cfg.fused_qkv)
        self.ffn_norm = make_norm(cfg.ffn_norm_config, cfg.n_embd)
        if cfg.activation_type == ActivationType.SWIGLU:
            self.mlp = SwiGLU(cfg.n_embd, cfg.ffn_hidden, cfg.bias,
                              packed=cfg.packed_swiglu)
        else:
            self.mlp = TransformerMLP(cfg.n_embd, cfg.ffn_hidden, cfg.bias, cfg.dropout)

    def forward(self, x, rope_cos, rope_sin):
        x = x + self.attn(self.attention_norm(x), rope_cos, rope_sin)
        x = x + self.mlp(self.ffn_norm(x))
        return x

    def forward_cached(self, x, rope_cos, rope_sin, cache_k, cache_v, pos):
        x = x + self.attn.forward_cached(self.attention_norm(x), rope_cos,
                                         rope_sin, cache_k, cache_v, pos)
        x = x + self.mlp(self.ffn_norm(x))
        return x


class GPT2LLM(NNModel):
    def __init__(self, config: GPT2LLMConfig):
        if isinstance(config, dict):
            config = GPT2LLMConfig(**config)
        weight_decay_groups = {
            "linear": [r"attn\..*attn\.weight", r"attn\.qkv_attn\.weight",
                       r"attn\.c_proj\.weight",
                       r"mlp\..*\.weight", r"lm_head\.weight"],
            "embedding": [r"wte\.weight", r"wpe\.weight"],
            "norm": [r"norm", r"ln_", r"\.bias"],
        }
        super().__init__(seed=config.seed, weight_decay_groups=weight_decay_groups)
        self.config = cfg = config
        self.sample_key = cfg.sample_key
        self.prediction_key = cfg.prediction_key

        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        if cfg.poe_type == PositionTypes.ABSOLUTE:
            self.wpe = nn.Embedding(cfg.sequence_length, cfg.n_embd)
        else:
            self.wpe = None
        self.drop = nn.Dropout(cfg.dropout)
        self.blocks = nn.ModuleList(GPT2Block(cfg) for _ in range(cfg.n_layer))
        self.lm_head_norm = make_norm(cfg.lm_head_norm_config, cfg.n_embd)
        self.lm_head = TwoStreamLinear(cfg.n_embd, cfg.vocab_size, bias=False)
        if cfg.use_weight_tying:
            self.lm_head.weight = self.wte.weight

        self._rope_cache: Optional[tuple[torch.Tensor, torch.Tensor]] = None

    # -- rope table ------------------------------------------------------
    def _rope(self, T: int, device):
        if self.config.qkv_transform != QueryKeyValueTransformType.ROTARY:
            return None, None
        head_dim = self.config.n_embd // self.config.n_head_q
        if (self._rope_cache is None or self._rope_cache[0].shape[0] < T
                or self._rope_cache[0].device != device):
            cos, sin = precompute_rope_cos_sin(
                max(T, self.config.sequence_length), head_dim,
                self.config.rope_base, device=device)
            self._rope_cache = (cos, sin)
        cos, sin = self._rope_cache
        return cos[:T], sin[:T]

    def forward_impl(self, inputs: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        input_ids = inputs[self.sample_key]
        B, T = input_ids.shape
        x = self.wte(input_ids)
        if self.wpe is not None:
            pos = torch.arange(T, dtype=torch.long, device=input_ids.device)
            x = x + self.wpe(pos)
        x = self.drop(x)
        rope_cos, rope_sin = self._rope(T, x.device)
        for block in self.blocks:
            x = block(x, rope_cos, rope_sin)
        x = self.lm_head_norm(x)
        logits = self.lm_head(x)
        return {self.prediction_key: logits}

    def forward(self, inputs: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        return self.forward_impl(inputs)

    # -- incremental decoding (KV cache) --------------------------------
    def new_kv_cache(self, batch_size: int, max_len: Optional[int] = None,
                     device=None, dtype=None) -> "KVCache":
        cfg = self.config
        max_len = max_len or cfg.sequence_length
        p = next(self.parameters())
        return KVCache(cfg.n_layer, batch_size, max_len,
                       cfg.n_head_kv, cfg.n_embd // cfg.n_head_q,
                       device or p.device, dtype or p.dtype)

    @torch.no_grad()
    def forward_cached(self, inputs: dict[str, torch.Tensor],
                       cache: "KVCache") -> dict[str, torch.Tensor]:
        """Forward only the NEW tokens; attends against the cached prefix.
        Advances cache.pos. Logits are returned for the new tokens only."""
        input_ids = inputs[self.sample_key]
        B, T = input_ids.shape
        pos = cache.pos
        if pos + T > cache.max_len:
            raise ValueError(f"KV cache overflow: {pos}+{T} > {cache.max_len}")
        x = self.wte(input_ids)
        if self.wpe is not None:
            p = torch.arange(pos, pos + T, dtype=torch.long,
                             device=input_ids.device)
            x = x + self.wpe(p)
        rope_cos, rope_sin = self._rope(cache.max_len, x.device)
        for i, block in enumerate(self.blocks):
            x = block.forward_cached(x, rope_cos, rope_sin,
                                     cache.k[i], cache.v[i], pos)
        cache.pos = pos + T
        x = self.lm_head_norm(x)
        return {self.prediction_key: self.lm_head(x)}


class KVCache:
    """Preallocated per-layer K/V buffers for incremental decoding."""

    def __init__(self, n_layer: int, batch_size: int, max_len: int,
                 n_head_kv: int, head_dim: int, device, dtype):
        self.max_len = max_len
        self.pos = 0
        self.k = [torch.zeros(batch_size, max_len, n_head_kv, head_dim,
                              device=device, dtype=dtype)
                  for _ in range(n_layer)]
        self.v = [torch.zeros(batch_size, max_len, n_head_kv, head_dim,
                              device=device, dtype=dtype)
                  for _ in range(n_layer)]

    def reset(self):
        self.pos = 0
