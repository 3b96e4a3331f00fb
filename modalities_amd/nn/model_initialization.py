"""Weight initialization (capability parity with reference
src/modalities/nn/model_initialization/: composed plain/scaled/scaled_embed
normal init selected by parameter-name regex, std="auto" = sqrt(2/(5*h)) per
arXiv:2312.16903; plus a Llama3/TorchTitan-style depth-scaled truncated
normal init, reference models/gpt2/llama3_like_initialization.py:21-181)."""

import math
import re
from typing import Optional, Union

import torch
import torch.nn as nn


class ModelInitializationIF:
    def initialize_in_place(self, model: nn.Module) -> None:
        raise NotImplementedError


class NamedParameterwiseNormalInitialization(ModelInitializationIF):
    """Normal(mean, std) on every parameter whose name fullmatches one of
    the regexes; biases matched by `bias_regexes` are zeroed."""

    def __init__(self, mean: float, std: float, parameter_name_regexes: list[str],
                 bias_regexes: Optional[list[str]] = None):
        self.mean = mean
        self.std = std
        self.weight_regexes = [re.compile(r) for r in parameter_name_regexes]
        self.bias_regexes = [re.compile(r) for r in (bias_regexes or [])]

    @torch.no_grad()
    def initialize_in_place(self, model: nn.Module) -> None:
        for name, p in model.named_parameters():
            name = name.replace("_orig_mod.", "")  # torch.compile FQN prefix
            if any(r.fullmatch(name) for r in self.weight_regexes):
                nn.init.normal_(p, mean=self.mean, std=self.std)
            if any(r.fullmatch(name) for r in self.bias_regexes):
                nn.init.zeros_(p)


def get_plain_initialization(mean: float, std: Union[float, str],
                             parameter_name_regexes: list[str],
                             hidden_dim: Optional[int] = None
                             ) -> NamedParameterwiseNormalInitialization:
    """std="auto" -> sqrt(2/(5*hidden_dim)) (arXiv:2312.16903)."""
    if std == "auto":
        if hidden_dim is None:
            raise ValueError("hidden_dim must be set when std='auto'")
        std = math.sqrt(2 / (5 * hidden_dim))
    elif hidden_dim is not None:
        raise ValueError("hidden_dim must not be set when std is a float")
    return NamedParameterwiseNormalInitialization(mean, float(std),
                                                  parameter_name_regexes)


def get_scaled_initialization(mean: float, std: float, num_layers: int,
                              parameter_name_regexes: list[str]
                              ) -> NamedParameterwiseNormalInitialization:
    """Projection-layer init scaled down with depth: std/sqrt(2*num_layers)
    (GPT-2 paper / arXiv:2312.16903)."""
    scaled = std / math.sqrt(2 * num_layers)
    return NamedParameterwiseNormalInitialization(mean, scaled,
                                                  parameter_name_regexes)


def get_scaled_embed_initialization(mean: float,
                                    parameter_name_regexes: list[str]
                                    ) -> NamedParameterwiseNormalInitialization:
    """Embedding init with std=0.4 (Le Scao et al. / reference default)."""
    return NamedParameterwiseNormalInitialization(mean, 0.4,
                                                  parameter_name_regexes)


class ComposedInitialization(ModelInitializationIF):
    """Run a list of initializations in order (later ones override earlier
    matches — reference composed_initialization.py:92-157)."""

    def __init__(self, model_initializers: list[ModelInitializationIF]):
        self.model_initializers = model_initializers

    def initialize_in_place(self, model: nn.Module) -> None:
        for init in self.model_initializers:
            init.initialize_in_place(model)


# Canonical GPT2 regex groups for the composed initializer (reference
# parameter_name_filters.py); our GPT2 uses q/k/v/c_proj + SwiGLU W/V/W_2.
GPT2_PLAIN_REGEXES = [
    r".*wte\.weight", r".*wpe\.weight",
    r".*q_attn\.weight", r".*k_attn\.weight", r".*v_attn\.weight",
    r".*qkv_attn\.weight",
    r".*mlp\.W\.weight", r".*mlp\.V\.weight", r".*mlp\.Wv\.weight",
    r".*mlp\.c_fc\.weight",
    r".*lm_head\.weight",
]
GPT2_PROJECTION_REGEXES = [
    r".*attn\.c_proj\.weight", r".*mlp\.W_2\.weight", r".*mlp\.c_proj\.weight",
]


def get_composed_model_initializer(model_type: str = "gpt2",
                                   weight_init_type: str = "scaled",
                                   mean: float = 0.0,
                                   std: Union[float, str] = 0.02,
                                   num_layers: Optional[int] = None,
                                   hidden_dim: Optional[int] = None
                                   ) -> ComposedInitialization:
    """Reference composed_initialization.py semantics: plain init everywhere,
    then (for weight_init_type="scaled") depth-scaled init on projections,
    or (for "scaled_embed") additionally embeddings at std=0.4."""
    if model_type != "gpt2":
        raise ValueError(f"Unknown model_type {model_type!r}")
    base_std = std
    inits: list[ModelInitializationIF] = [
        get_plain_initialization(mean, base_std,
                                 GPT2_PLAIN_REGEXES + GPT2_PROJECTION_REGEXES,
                                 hidden_dim=hidden_dim if std == "auto" else None)
    ]
    if weight_init_type in ("scaled", "scaled_embed"):
        if num_layers is None:
            raise ValueError("num_layers required for scaled init")
        eff_std = math.sqrt(2 / (5 * hidden_dim)) if std == "auto" else float(std)
        inits.append(get_scaled_initialization(mean, eff_std, num_layers,
                                               GPT2_PROJECTION_REGEXES))
    if weight_init_type == "scaled_embed":
        inits.append(get_scaled_embed_initialization(
            mean, [r".*wte\.weight", r".*wpe\.weight"]))
    if weight_init_type not in ("plain", "scaled", "scaled_embed"):
        raise ValueError(f"Unknown weight_init_type {weight_init_type!r}")
    return ComposedInitialization(inits)


class Llama3LikeInitialization(ModelInitializationIF):
    """TorchTitan/Llama3-style init (reference
    llama3_like_initialization.py:21-181): truncated normal with per-group
    std; final projections scaled by depth; norms reset to ones/zeros."""

    def __init__(self, n_embd: int, n_layer: int,
                 use_scaled_init: bool = True, cutoff_factor: float = 3.0):
        self.n_embd = n_embd
        self.n_layer = n_layer
        self.use_scaled_init = use_scaled_init
        self.cutoff_factor = cutoff_factor

    @torch.no_grad()
    def initialize_in_place(self, model: nn.Module) -> None:
        in_std = self.n_embd ** -0.5
        depth_std = in_std / math.sqrt(2 * self.n_layer) \
            if self.use_scaled_init else in_std
        cf = self.cutoff_factor
        for name, p in model.named_parameters():
            name = name.replace("_orig_mod.", "")
            if name.endswith(".bias"):
                nn.init.zeros_(p)
            elif "norm" in name.lower():
                nn.init.ones_(p)
            elif re.search(r"(c_proj|W_2)\.weight$", name):
                nn.init.trunc_normal_(p, mean=0.0, std=depth_std,
                                      a=-cf * depth_std, b=cf * depth_std)
            elif re.search(r"(wte|wpe)\.weight$", name) or "lm_head" in name:
                nn.init.trunc_normal_(p, mean=0.0, std=in_std,
                                      a=-cf * in_std, b=cf * in_std)
            elif p.ndim >= 2:
                nn.init.trunc_normal_(p, mean=0.0, std=in_std,
                                      a=-cf * in_std, b=cf * in_std)
