"""Tests: number conversion, weight init, MFU, benchmarking sweeps,
experiment id, seeding, HF conversion."""

import math
from pathlib import Path

import pytest
import torch
import yaml

from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.nn.model_initialization import (
    Llama3LikeInitialization, get_composed_model_initializer,
    get_plain_initialization, get_scaled_initialization)
from modalities_amd.utils.benchmarking import (list_remaining_runs,
                                               prepare_sweep_configs)
from modalities_amd.utils.number_conversion import NumberConversion
from modalities_amd.utils.seeding import calculate_hashed_seed


def tiny_model():
    torch.manual_seed(0)
    return GPT2LLM(GPT2LLMConfig(vocab_size=128, n_layer=2, n_head_q=4,
                                 n_head_kv=2, n_embd=64, ffn_hidden=256,
                                 sequence_length=32))


# ---- number conversion ------------------------------------------------------

def test_number_conversion_roundtrip():
    nc = NumberConversion
    assert nc.get_num_samples_from_num_tokens(1000, 10) == 100
    assert nc.get_num_steps_from_num_tokens(
        dp_degree=2, local_micro_batch_size=4, global_num_tokens=64_000,
        sequence_length=100, gradient_accumulation_steps=2) == 40
    assert nc.get_num_tokens_from_num_steps(
        num_steps=40, dp_degree=2, local_micro_batch_size=4,
        sequence_length=100, gradient_accumulation_steps=2) == 64_000
    assert nc.get_local_num_batches_from_num_samples(4, 1000, 10) == 25


def test_number_conversion_checkpoint_path():
    p = Path("checkpoints/eid_x-seen_steps_160-seen_tokens_2621440"
             "-target_steps_320-target_tokens_5242880")
    nc = NumberConversion
    assert nc.get_num_seen_steps_from_checkpoint_path(p) == 160
    assert nc.get_global_num_seen_tokens_from_checkpoint_path(p) == 2621440
    assert nc.get_num_target_steps_from_checkpoint_path(p) == 320
    assert nc.get_global_num_target_tokens_from_checkpoint_path(p) == 5242880
    assert nc.get_last_step_from_checkpoint_path(p) == 159


def test_number_conversion_from_pbin(tmp_path):
    import numpy as np

    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [np.zeros(500, dtype=np.uint16)]
    p = tmp_path / "d.pbin"
    write_pbin(p, docs, 2)
    n = NumberConversion.get_num_tokens_from_packed_mem_map_dataset_continuous(
        p, sequence_length=10, num_ranks=1, local_micro_batch_size=7,
        gradient_accumulation_steps=1)
    # 499 usable samples of 10 -> 49 samples -> 7 steps of 7 -> 490 tokens
    assert n == 490


# ---- weight init ------------------------------------------------------------

def test_plain_init_applies_std():
    model = tiny_model()
    init = get_plain_initialization(0.0, 0.5, [r".*wte\.weight"])
    init.initialize_in_place(model)
    std = model.wte.weight.std().item()
    assert 0.4 < std < 0.6


def test_plain_init_auto_std():
    init = get_plain_initialization(0.0, "auto", [r".*wte\.weight"],
                                    hidden_dim=640)
    expected = math.sqrt(2 / (5 * 640))
    assert init.std == pytest.approx(expected)
    with pytest.raises(ValueError):
        get_plain_initialization(0.0, "auto", [])
    with pytest.raises(ValueError):
        get_plain_initialization(0.0, 0.02, [], hidden_dim=10)


def test_scaled_init_downscales_projections():
    model = tiny_model()
    init = get_scaled_initialization(0.0, 0.4, num_layers=2,
                                     parameter_name_regexes=[r".*c_proj\.weight"])
    init.initialize_in_place(model)
    std = model.blocks[0].attn.c_proj.weight.std().item()
    assert std == pytest.approx(0.4 / math.sqrt(4), rel=0.2)


def test_composed_init_runs():
    model = tiny_model()
    init = get_composed_model_initializer(weight_init_type="scaled_embed",
                                          std=0.02, num_layers=2)
    init.initialize_in_place(model)
    assert model.wte.weight.std().item() == pytest.approx(0.4, rel=0.1)
    proj_std = model.blocks[0].mlp.W_2.weight.std().item()
    assert proj_std == pytest.approx(0.02 / math.sqrt(4), rel=0.25)


def test_llama3_init():
    model = tiny_model()
    Llama3LikeInitialization(n_embd=64, n_layer=2).initialize_in_place(model)
    assert model.blocks[0].attn.q_attn.weight.std().item() == pytest.approx(
        64 ** -0.5, rel=0.25)
    assert torch.all(model.lm_head_norm.weight == 1.0)


# ---- seeding / benchmarking -------------------------------------------------

def test_hashed_seed_deterministic():
    a = calculate_hashed_seed(["exp1", "chunk0"])
    b = calculate_hashed_seed(["exp1", "chunk0"])
    c = calculate_hashed_seed(["exp1", "chunk1"])
    assert a == b != c
    assert 0 <= a < 2**32 - 1


def test_sweep_expansion_and_remaining(tmp_path):
    sweep = {
        "settings": {"cuda_env": {"world_size": {"sweep": [1, 2]}},
                     "training_target": {"num_target_steps": 4}},
        "lr": {"sweep": [0.1, 0.2]},
    }
    sweep_path = tmp_path / "sweep.yaml"
    sweep_path.write_text(yaml.safe_dump(sweep))
    out = tmp_path / "runs"
    n = prepare_sweep_configs(sweep_path, out)
    assert n == 4
    cfgs = list(out.glob("world_size_*/*/config.yaml"))
    assert len(cfgs) == 4
    # nothing run yet -> all remaining
    assert len(list_remaining_runs(out)) == 4
    # fake a complete run
    run_dir = cfgs[0].parent
    with open(run_dir / "evaluation_results.jsonl", "w") as f:
        f.write('{"num_train_steps_done": 4}\n')
    assert len(list_remaining_runs(out)) == 3


# ---- HF conversion ----------------------------------------------------------

def test_hf_conversion_logit_equality():
    pytest.importorskip("transformers")
    from modalities_amd.conversion.convert_gpt2 import (check_converted_model,
                                                        convert_model_checkpoint)
    torch.manual_seed(0)
    model = GPT2LLM(GPT2LLMConfig(
        vocab_size=128, n_layer=2, n_head_q=4, n_head_kv=2, n_embd=64,
        ffn_hidden=256, sequence_length=64, activation_type="swiglu"))
    hf = convert_model_checkpoint(model)
    check_converted_model(hf, model, num_testruns=2, vocab_size=128, seq_len=32)


# ---------------------------------------------------------------------------
# Init distribution statistics (reference tests/test_initialization_fsdpx.py:
# after composed init, each parameter group's empirical std must match its
# configured target).

def test_composed_init_distribution_statistics():
    import math

    import torch

    from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
    from modalities_amd.nn.model_initialization import (
        get_composed_model_initializer)

    torch.manual_seed(0)
    h, L = 256, 4
    cfg = GPT2LLMConfig(vocab_size=2048, n_layer=L, n_head_q=4, n_head_kv=4,
                        n_embd=h, ffn_hidden=4 * h, sequence_length=32,
                        dropout=0.0)
    model = GPT2LLM(cfg)
    init = get_composed_model_initializer(
        model_type="gpt2", weight_init_type="scaled", mean=0.0, std="auto",
        num_layers=L, hidden_dim=h)
    init.initialize_in_place(model)

    auto_std = math.sqrt(2 / (5 * h))
    proj_std = auto_std / math.sqrt(2 * L)

    def emp_std(t):
        return t.float().std().item()

    # embeddings at the plain/auto std
    assert abs(emp_std(model.wte.weight) - auto_std) / auto_std < 0.05
    # in-projections (qkv / mlp up) at the plain std
    qkv_w = model.blocks[0].attn.qkv_attn.weight if cfg.fused_qkv \
        else model.blocks[0].attn.q_attn.weight
    assert abs(emp_std(qkv_w) - auto_std) / auto_std < 0.05
    up = model.blocks[0].mlp.W_weight
    assert abs(emp_std(up) - auto_std) / auto_std < 0.05
    # OUT-projections scaled down by sqrt(2L)
    assert abs(emp_std(model.blocks[0].attn.c_proj.weight) - proj_std) \
        / proj_std < 0.05
    assert abs(emp_std(model.blocks[0].mlp.W_2.weight) - proj_std) \
        / proj_std < 0.05
    # norms untouched (ones)
    assert torch.all(model.blocks[0].attention_norm.weight == 1.0)
