"""GPT2 model variant coverage (reference model:
tests/models/test_causal_self_attention.py, test_rotary_qkv_transform.py,
layer norm tests): positional modes, GELU vs SwiGLU, QK-norm, weight tying,
GQA, activation checkpointing equivalence, debug hooks, profilers."""

import pytest
import torch

from modalities_amd.models.gpt2 import (ActivationType, AttentionImplementation,
                                        GPT2LLM, GPT2LLMConfig, PositionTypes)

VOCAB = 128


def cfg(**kw):
    d = dict(vocab_size=VOCAB, n_layer=2, n_head_q=4, n_head_kv=2, n_embd=64,
             ffn_hidden=256, sequence_length=32, seed=1, dropout=0.0)
    d.update(kw)
    return GPT2LLMConfig(**d)


def run_fwd(model, seed=9, batch=2, seqlen=16):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, VOCAB, (batch, seqlen), generator=g)
    return model({"input_ids": ids})["logits"]


@pytest.mark.parametrize("poe", [PositionTypes.ABSOLUTE, PositionTypes.NOPE])
@pytest.mark.parametrize("act", [ActivationType.GELU, ActivationType.SWIGLU])
def test_gpt2_variants_forward_backward(poe, act):
    model = GPT2LLM(cfg(poe_type=poe, activation_type=act))
    out = run_fwd(model)
    assert out.shape == (2, 16, VOCAB)
    out.float().sum().backward()
    assert model.wte.weight.grad is not None
    if poe == PositionTypes.ABSOLUTE:
        assert model.wpe is not None and model.wpe.weight.grad is not None


def test_gpt2_qk_norm_and_weight_tying():
    model = GPT2LLM(cfg(use_qk_norm=True, use_weight_tying=True))
    assert model.lm_head.weight is model.wte.weight
    out = run_fwd(model)
    assert torch.isfinite(out.float()).all()


def test_attention_impls_agree():
    """pytorch_flash (SDPA) and manual O(T^2) paths must agree; both serve
    as CPU references for the HIP kernel (tested on GPU)."""
    m1 = GPT2LLM(cfg(attention_implementation=AttentionImplementation.PYTORCH_FLASH))
    m2 = GPT2LLM(cfg(attention_implementation=AttentionImplementation.MANUAL))
    m2.load_state_dict(m1.state_dict())
    m1.eval(), m2.eval()
    o1, o2 = run_fwd(m1), run_fwd(m2)
    torch.testing.assert_close(o1, o2, rtol=1e-4, atol=1e-5)


def test_causality():
    model = GPT2LLM(cfg())
    model.eval()
    g = torch.Generator().manual_seed(4)
    ids = torch.randint(0, VOCAB, (1, 16), generator=g)
    out1 = model({"input_ids": ids})["logits"]
    ids2 = ids.clone()
    ids2[0, -1] = (ids2[0, -1] + 1) % VOCAB
    out2 = model({"input_ids": ids2})["logits"]
    torch.testing.assert_close(out1[0, :-1], out2[0, :-1], rtol=1e-4, atol=1e-5)


# ---- activation checkpointing ----------------------------------------------

@pytest.mark.parametrize("variant", ["full_activation_checkpointing",
                                     "selective_layer_activation_checkpointing",
                                     "selective_op_activation_checkpointing"])
def test_activation_checkpointing_grad_equivalence(variant):
    from modalities_amd.training.activation_checkpointing import (
        ActivationCheckpointingVariant, apply_activation_checkpointing_)
    torch.manual_seed(0)
    ref = GPT2LLM(cfg())
    torch.manual_seed(0)
    ckpt = GPT2LLM(cfg())
    apply_activation_checkpointing_(ckpt, ActivationCheckpointingVariant(variant),
                                    every_k_layers=2)
    g = torch.Generator().manual_seed(3)
    ids = torch.randint(0, VOCAB, (2, 17), generator=g)
    for model in (ref, ckpt):
        out = model({"input_ids": ids[:, :-1]})["logits"]
        loss = torch.nn.functional.cross_entropy(out.reshape(-1, VOCAB).float(),
                                                 ids[:, 1:].reshape(-1))
        loss.backward()
    for (n1, p1), (n2, p2) in zip(ref.named_parameters(), ckpt.named_parameters()):
        torch.testing.assert_close(p1.grad, p2.grad, rtol=1e-5, atol=1e-6,
                                   msg=lambda m: f"{n1}/{n2}: {m}")


def test_activation_checkpointing_composes_with_sharding_engine():
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    from modalities_amd.training.activation_checkpointing import (
        ActivationCheckpointingVariant, apply_activation_checkpointing_)
    torch.manual_seed(0)
    model = GPT2LLM(cfg())
    apply_activation_checkpointing_(
        model, ActivationCheckpointingVariant.FULL_ACTIVATION_CHECKPOINTING)
    sharded = XGMIShardedModel.from_transformer(model, torch.device("cpu"),
                                                param_dtype=torch.float32)
    g = torch.Generator().manual_seed(3)
    ids = torch.randint(0, VOCAB, (2, 17), generator=g)
    out = sharded({"input_ids": ids[:, :-1]})["logits"]
    loss = torch.nn.functional.cross_entropy(out.reshape(-1, VOCAB).float(),
                                             ids[:, 1:].reshape(-1))
    loss.backward()
    sharded.backward_epilogue()
    assert all(u.grad_fresh for u in sharded.units)


# ---- debug components -------------------------------------------------------

def test_nan_hook_raises():
    from modalities_amd.utils.debug_components import register_nan_hooks
    model = GPT2LLM(cfg())
    register_nan_hooks(model, module_name_filter="lm_head")
    with torch.no_grad():
        model.lm_head.weight[0, 0] = float("nan")
    with pytest.raises(RuntimeError, match="Non-finite"):
        run_fwd(model)


def test_tensor_stats_hook(tmp_path):
    import json

    from modalities_amd.utils.debug_components import register_tensor_stats_hooks
    model = GPT2LLM(cfg())
    n = register_tensor_stats_hooks(model, tmp_path / "stats.jsonl",
                                    module_name_filter="lm_head_norm")
    assert n == 1
    run_fwd(model)
    recs = [json.loads(ln) for ln in (tmp_path / "stats.jsonl").read_text().splitlines()]
    assert recs and recs[0]["module"] == "lm_head_norm"
    assert "mean" in recs[0] and "absmax" in recs[0]


# ---- profilers --------------------------------------------------------------

def test_steppable_profilers(tmp_path):
    from modalities_amd.utils.profilers import (SteppableKernelProfiler,
                                                SteppableNoProfiler, get_profiler)
    model = GPT2LLM(cfg())
    prof = SteppableKernelProfiler(tmp_path, wait=0, warmup=1, active=1, repeat=1)
    with prof:
        for _ in range(len(prof)):
            run_fwd(model)
            prof.step()
    assert list(tmp_path.glob("trace_step*.json"))
    assert list(tmp_path.glob("key_averages_step*.txt"))
    # rank filtering
    assert isinstance(get_profiler("kernel", tmp_path, global_rank=1,
                                   tracked_ranks=[0]), SteppableNoProfiler)


def test_fused_qkv_matches_unfused():
    torch.manual_seed(0)
    unfused = GPT2LLM(cfg())
    fused = GPT2LLM(cfg(fused_qkv=True))
    # copy unfused weights into the fused projection
    with torch.no_grad():
        for bu, bf_ in zip(unfused.blocks, fused.blocks):
            cat = torch.cat([bu.attn.q_attn.weight, bu.attn.k_attn.weight,
                             bu.attn.v_attn.weight], dim=0)
            bf_.attn.qkv_attn.weight.copy_(cat)
        sd = {k: v for k, v in unfused.state_dict().items()
              if "q_attn" not in k and "k_attn" not in k and "v_attn" not in k}
        fused.load_state_dict(sd, strict=False)
    unfused.eval(), fused.eval()
    o1, o2 = run_fwd(unfused), run_fwd(fused)
    torch.testing.assert_close(o1, o2, rtol=1e-4, atol=1e-5)


def test_packed_swiglu_matches_unpacked():
    """packed=True (one joint W|V GEMM + joint silu-mul) must match the
    reference separate-W/V layout bitwise given the same weights."""
    import torch

    from modalities_amd.models.model import SwiGLU
    torch.manual_seed(3)
    ref = SwiGLU(32, 128, packed=False)
    packed = SwiGLU(32, 128, packed=True)
    with torch.no_grad():
        packed.Wv.weight[:packed.hidden_dim].copy_(ref.W.weight)
        packed.Wv.weight[packed.hidden_dim:].copy_(ref.V.weight)
        packed.W_2.weight.copy_(ref.W_2.weight)
    x = torch.randn(2, 5, 32, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y1, y2 = ref(x), packed(x2)
    torch.testing.assert_close(y1, y2)
    y1.sum().backward()
    y2.sum().backward()
    torch.testing.assert_close(x.grad, x2.grad)
    torch.testing.assert_close(packed.Wv.weight.grad[:packed.hidden_dim],
                               ref.W.weight.grad)
    torch.testing.assert_close(packed.Wv.weight.grad[packed.hidden_dim:],
                               ref.V.weight.grad)
    torch.testing.assert_close(packed.W_weight, ref.W.weight)
    torch.testing.assert_close(packed.V_weight, ref.V.weight)


def test_attention_dropout_falls_back_to_sdpa():
    """dropout > 0 with the HIP flash impl routes the attention op to the
    SDPA path instead of erroring (reference flash paths support dropout)."""
    import warnings

    import torch

    from modalities_amd.models.gpt2 import (AttentionImplementation, GPT2LLM,
                                            GPT2LLMConfig)
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        model = GPT2LLM(GPT2LLMConfig(
            vocab_size=64, n_layer=1, n_head_q=2, n_head_kv=2, n_embd=32,
            ffn_hidden=128, sequence_length=16, dropout=0.1))
    assert any("SDPA" in str(x.message) for x in w)
    assert model.blocks[0].attn.attention_impl == \
        AttentionImplementation.PYTORCH_FLASH
    model.eval()
    out = model({"input_ids": torch.randint(0, 64, (2, 16))})
    assert out["logits"].shape == (2, 16, 64)


def test_torch_compile_blockwise_matches_eager():
    """get_compiled_model (reference tests/test_torch_compile.py behavior):
    per-block compile keeps outputs equal to eager and blocks compiled."""
    from modalities_amd.models.model_factory import ModelFactory

    torch.manual_seed(0)
    model = GPT2LLM(cfg())
    g = torch.Generator().manual_seed(4)
    ids = torch.randint(0, VOCAB, (2, 16), generator=g)
    with torch.no_grad():
        ref = model({"input_ids": ids})["logits"].clone()

    compiled = ModelFactory.get_compiled_model(model, block_names=["GPT2Block"])
    from torch._dynamo import OptimizedModule  # noqa: PLC0415
    n_compiled = sum(isinstance(m, OptimizedModule) or
                     getattr(m, "_compiled_call_impl", None) is not None
                     for m in compiled.modules())
    assert n_compiled >= 2, "expected every block compiled"
    with torch.no_grad():
        out = compiled({"input_ids": ids})["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-4, atol=2e-4)


def test_ac_variant_string_coercion_from_yaml():
    """YAML gives the AC variant as a string (enum name OR value); the
    factory must coerce it — a bad string must raise, and selective_layer
    with k=2 must leave alternate blocks unwrapped."""
    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.registry.components import get_default_registry

    def build(variant, k=1):
        c = {
            "model": {"component_key": "model", "variant_key": "gpt2",
                      "config": dict(sample_key="input_ids",
                                     prediction_key="logits", vocab_size=64,
                                     n_layer=4, n_head_q=2, n_head_kv=2,
                                     n_embd=32, ffn_hidden=64,
                                     sequence_length=16, seed=1)},
            "acm": {"component_key": "activation_checkpointed_model",
                    "variant_key": "default",
                    "config": {
                        "model": {"instance_key": "model",
                                  "pass_type": "BY_REFERENCE"},
                        "activation_checkpointing_variant": variant,
                        "every_k_layers": k}},
        }
        return ComponentFactory(get_default_registry()).build_component_by_key(
            c, "acm")

    m = build("SELECTIVE_LAYER_ACTIVATION_CHECKPOINTING", k=2)
    kinds = [type(b).__name__ for b in m.blocks]
    assert kinds == ["CheckpointedBlock", "GPT2Block",
                     "CheckpointedBlock", "GPT2Block"], kinds
    m2 = build("selective_op_activation_checkpointing")
    assert m2.blocks[0]._context_fn is not None
    with pytest.raises(KeyError):
        build("not_a_variant")
