"""modalities_amd -> HuggingFace checkpoint conversion (capability parity
with reference src/modalities/conversion/gpt2/convert_gpt2.py:36-60 and
conversion_model.py:31-69).

Our GPT2/Llama-style architecture (RMSNorm + RoPE + GQA + SwiGLU, no biases)
maps 1:1 onto transformers' LlamaForCausalLM, so the export produces a
standard Llama checkpoint (no custom modeling files needed) with
logit-equality verification (reference conversion_model.py:71-90)."""

from pathlib import Path

import torch

from modalities_amd.models.gpt2 import (ActivationType, GPT2LLM, GPT2LLMConfig,
                                        LayerNormVariant, PositionTypes)


def check_converted_model(hf_model, modalities_model: GPT2LLM, num_testruns: int = 1,
                          vocab_size: int = 50304, seq_len: int = 64,
                          atol: float = 1e-4) -> None:
    hf_model.eval()
    modalities_model.eval()
    with torch.no_grad():
        for _ in range(num_testruns):
            ids = torch.randint(0, vocab_size, (1, seq_len))
            ours = modalities_model({modalities_model.sample_key: ids})[
                modalities_model.prediction_key].float()
            theirs = hf_model(ids).logits.float()
            if not torch.allclose(ours, theirs, atol=atol, rtol=1e-3):
                diff = (ours - theirs).abs().max().item()
                raise AssertionError(f"converted model logits differ (max abs "
                                     f"diff {diff})")


def convert_model_checkpoint(modalities_model: GPT2LLM):
    """Build an HF LlamaForCausalLM with our weights. Returns the HF model."""
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg: GPT2LLMConfig = modalities_model.config
    if cfg.activation_type != ActivationType.SWIGLU:
        raise ValueError("HF conversion requires SwiGLU MLP (Llama-style)")
    if cfg.poe_type != PositionTypes.NOPE:
        raise ValueError("HF conversion requires NOPE/RoPE positions")
    for nc in (cfg.attention_norm_config, cfg.ffn_norm_config,
               cfg.lm_head_norm_config):
        if nc.variant != LayerNormVariant.RMS_NORM:
            raise ValueError("HF conversion requires RMSNorm")
    hidden = modalities_model.blocks[0].mlp.hidden_dim
    head_dim = cfg.n_embd // cfg.n_head_q

    hf_cfg = LlamaConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.n_embd,
        intermediate_size=hidden,
        num_hidden_layers=cfg.n_layer,
        num_attention_heads=cfg.n_head_q,
        num_key_value_heads=cfg.n_head_kv,
        head_dim=head_dim,
        max_position_embeddings=cfg.sequence_length,
        rms_norm_eps=cfg.attention_norm_config.eps,
        rope_theta=cfg.rope_base,
        attention_bias=cfg.bias,
        mlp_bias=cfg.bias,
        tie_word_embeddings=cfg.use_weight_tying,
    )
    hf = LlamaForCausalLM(hf_cfg)
    sd = {}
    src = modalities_model.state_dict()
    sd["model.embed_tokens.weight"] = src["wte.weight"]
    head_dim_kv = cfg.n_embd // cfg.n_head_q * cfg.n_head_kv
    for i in range(cfg.n_layer):
        p, hp = f"blocks.{i}", f"model.layers.{i}"
        sd[f"{hp}.input_layernorm.weight"] = src[f"{p}.attention_norm.weight"]
        if cfg.fused_qkv:
            w = src[f"{p}.attn.qkv_attn.weight"]
            sd[f"{hp}.self_attn.q_proj.weight"] = w[:cfg.n_embd]
            sd[f"{hp}.self_attn.k_proj.weight"] = w[cfg.n_embd:cfg.n_embd + head_dim_kv]
            sd[f"{hp}.self_attn.v_proj.weight"] = w[cfg.n_embd + head_dim_kv:]
        else:
            sd[f"{hp}.self_attn.q_proj.weight"] = src[f"{p}.attn.q_attn.weight"]
            sd[f"{hp}.self_attn.k_proj.weight"] = src[f"{p}.attn.k_attn.weight"]
            sd[f"{hp}.self_attn.v_proj.weight"] = src[f"{p}.attn.v_attn.weight"]
        sd[f"{hp}.self_attn.o_proj.weight"] = src[f"{p}.attn.c_proj.weight"]
        sd[f"{hp}.post_attention_layernorm.weight"] = src[f"{p}.ffn_norm.weight"]
        if f"{p}.mlp.Wv.weight" in src:  # packed layout: gate|up halves
            wv = src[f"{p}.mlp.Wv.weight"]
            sd[f"{hp}.mlp.gate_proj.weight"] = wv[:wv.shape[0] // 2]
            sd[f"{hp}.mlp.up_proj.weight"] = wv[wv.shape[0] // 2:]
        else:
            sd[f"{hp}.mlp.gate_proj.weight"] = src[f"{p}.mlp.W.weight"]
            sd[f"{hp}.mlp.up_proj.weight"] = src[f"{p}.mlp.V.weight"]
        sd[f"{hp}.mlp.down_proj.weight"] = src[f"{p}.mlp.W_2.weight"]
        if cfg.bias:
            for ours, theirs in [("q_attn", "q_proj"), ("k_attn", "k_proj"),
                                 ("v_attn", "v_proj"), ("c_proj", "o_proj")]:
                sd[f"{hp}.self_attn.{theirs}.bias"] = src[f"{p}.attn.{ours}.bias"]
    sd["model.norm.weight"] = src["lm_head_norm.weight"]
    sd["lm_head.weight"] = src["lm_head.weight"]
    missing, unexpected = hf.load_state_dict(sd, strict=False)
    real_missing = [m for m in missing if "rotary_emb" not in m]
    if real_missing or unexpected:
        raise RuntimeError(f"state dict mismatch: missing={real_missing}, "
                           f"unexpected={unexpected}")
    return hf


def convert_gpt2_to_hf(config_path: Path, output_dir: Path,
                       prediction_key: str = "logits", verify: bool = True):
    """Load a modalities_amd model per config, convert, verify, save
    (reference convert_gpt2.py:36-60)."""
    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.config.yaml_loader import load_app_config_dict
    from modalities_amd.registry.components import get_default_registry

    config_dict = load_app_config_dict(Path(config_path))
    factory = ComponentFactory(get_default_registry())
    model = factory.build_component_by_key(config_dict, "model")
    hf = convert_model_checkpoint(model)
    if verify:
        check_converted_model(hf, model, vocab_size=model.config.vocab_size)
    output_dir = Path(output_dir)
    output_dir.mkdir(parents=True, exist_ok=True)
    hf.save_pretrained(output_dir)

    # ship a loadable tokenizer with the export when the config names an
    # SP tokenizer (reference convert_gpt2.py + conversion_tokenizer.py)
    tok = config_dict.get("tokenizer")
    if tok and tok.get("variant_key") == "pretrained_sp_tokenizer":
        from modalities_amd.conversion.convert_tokenizer import convert_tokenizer
        convert_tokenizer(tok["config"]["tokenizer_model_file"], str(output_dir))
    return hf
