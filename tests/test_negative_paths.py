"""Negative-path coverage: invalid configs, corrupt /
truncated .pbin files, checkpoint/model mismatches, and the per-rank JSON
error-log schema the CLI writes on failure. Each test asserts the specific
exception type and message."""

import json
import pickle
import struct
from pathlib import Path

import numpy as np
import pytest
import torch

from modalities_amd.dataloader.packed_data import (EmbeddedStreamData,
                                                   write_pbin)


# ---------------------------------------------------------------------------
# .pbin corruption
# ---------------------------------------------------------------------------

def _make_pbin(tmp_path, docs=((1, 2, 3), (4, 5))):
    p = tmp_path / "d.pbin"
    write_pbin(p, [np.asarray(d, dtype=np.uint16) for d in docs], 2)
    return p


def test_pbin_missing_file(tmp_path):
    with pytest.raises(FileNotFoundError, match="Packed data not found"):
        EmbeddedStreamData(tmp_path / "nope.pbin")


def test_pbin_truncated_data_section(tmp_path):
    p = _make_pbin(tmp_path)
    raw = p.read_bytes()
    (tmp_path / "t.pbin").write_bytes(raw[: EmbeddedStreamData.HEADER_SIZE_IN_BYTES + 3])
    with pytest.raises(ValueError, match="Truncated .pbin"):
        EmbeddedStreamData(tmp_path / "t.pbin")


def test_pbin_bad_token_size(tmp_path):
    p = _make_pbin(tmp_path)
    raw = bytearray(p.read_bytes())
    # token-size descriptor sits after the 8-byte data-length field
    raw[8:12] = struct.pack("<I", 7)
    (tmp_path / "b.pbin").write_bytes(bytes(raw))
    with pytest.raises(ValueError, match="token size 7"):
        EmbeddedStreamData(tmp_path / "b.pbin")


def test_pbin_corrupt_index(tmp_path):
    p = _make_pbin(tmp_path)
    raw = p.read_bytes()
    data_end = EmbeddedStreamData.HEADER_SIZE_IN_BYTES \
        + int.from_bytes(raw[:8], "little")
    (tmp_path / "c.pbin").write_bytes(raw[:data_end] + b"\x80garbage")
    with pytest.raises(ValueError, match="Corrupt .pbin document index"):
        EmbeddedStreamData(tmp_path / "c.pbin")


def test_dataset_block_size_too_large(tmp_path):
    from modalities_amd.dataloader.dataset import PackedMemMapDatasetContinuous
    p = _make_pbin(tmp_path)
    ds = PackedMemMapDatasetContinuous(p, sample_key="input_ids",
                                       block_size=1024)
    assert len(ds) == 0  # too few tokens for one block -> empty, not crash


# ---------------------------------------------------------------------------
# config / component factory
# ---------------------------------------------------------------------------

def test_unknown_component_variant():
    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.registry.components import get_default_registry
    factory = ComponentFactory(get_default_registry())
    cfg = {"loss_fn": {"component_key": "loss", "variant_key": "does_not_exist",
                       "config": {}}}
    with pytest.raises(ValueError, match="does_not_exist"):
        factory.build_component_by_key(cfg, "loss_fn")


def test_missing_reference_target():
    from pydantic import BaseModel, ConfigDict

    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.registry.components import get_default_registry

    class _Shell(BaseModel):
        model_config = ConfigDict(arbitrary_types_allowed=True)
        loss_fn: object
        other: object

    factory = ComponentFactory(get_default_registry())
    cfg = {"loss_fn": {"component_key": "loss", "variant_key": "clm_cross_entropy_loss",
                       "config": {"target_key": "t", "prediction_key": "p"}},
           "other": {"instance_key": "nonexistent", "pass_type": "BY_REFERENCE"}}
    with pytest.raises(KeyError, match="nonexistent"):
        factory.build_components(cfg, _Shell)


def test_inconsistent_training_target_raises():
    from modalities_amd.config.instantiation_models import TrainingSettings
    with pytest.raises(Exception, match="Inconsistent training target"):
        TrainingSettings(
            training_target={"num_target_steps": 10, "num_target_tokens": 999},
            step_profile={"gradient_accumulation_steps": 1,
                          "local_train_micro_batch_size": 1,
                          "sequence_length": 8})


def test_mesh_degree_mismatch_raises():
    from modalities_amd.parallel.mesh import DeviceMesh
    with pytest.raises(ValueError):
        DeviceMesh(world_size=4, rank=0, dp_shard=3, create_groups=False)


# ---------------------------------------------------------------------------
# checkpoint / model mismatches
# ---------------------------------------------------------------------------

def _tiny_engine(n_embd=32, n_layer=2, vocab=64):
    from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    torch.manual_seed(0)
    model = GPT2LLM(GPT2LLMConfig(
        vocab_size=vocab, n_layer=n_layer, n_head_q=2, n_head_kv=2,
        n_embd=n_embd, ffn_hidden=4 * n_embd, sequence_length=16, dropout=0.0))
    return XGMIShardedModel.from_transformer(model, torch.device("cpu"),
                                             param_dtype=torch.float32)


def test_checkpoint_meta_missing(tmp_path):
    from modalities_amd.checkpointing.app_state import AppState
    from modalities_amd.checkpointing.loading import ShardedCheckpointLoading
    from modalities_amd.optimizers.optimizer_factory import get_adam_w
    eng = _tiny_engine()
    app = AppState(eng, get_adam_w(eng, lr=1e-3))
    (tmp_path / "empty_ckpt").mkdir()
    with pytest.raises(FileNotFoundError):
        ShardedCheckpointLoading(0).load_checkpoint_(app,
                                                     tmp_path / "empty_ckpt")


def test_checkpoint_architecture_mismatch(tmp_path):
    """Loading a checkpoint saved for a DIFFERENT architecture must fail
    loudly with the offending unit named (not load garbage)."""
    from modalities_amd.checkpointing.app_state import AppState
    from modalities_amd.checkpointing.loading import ShardedCheckpointLoading
    from modalities_amd.checkpointing.saving import ShardedCheckpointSaving
    from modalities_amd.optimizers.optimizer_factory import get_adam_w
    from modalities_amd.training.progress import TrainingProgress

    eng = _tiny_engine(n_embd=32)
    opt = get_adam_w(eng, lr=1e-3)
    app = AppState(eng, opt)
    saving = ShardedCheckpointSaving(tmp_path, "exp", global_rank=0)
    progress = TrainingProgress(num_seen_steps_current_run=1,
                                num_seen_tokens_current_run=8,
                                num_target_steps=2, num_target_tokens=16)
    saving.save_checkpoint(app, progress)
    folder = saving._folder(progress)

    eng2 = _tiny_engine(n_embd=64)  # different width
    app2 = AppState(eng2, get_adam_w(eng2, lr=1e-3))
    with pytest.raises(ValueError, match="does not match the model"):
        ShardedCheckpointLoading(0).load_checkpoint_(app2, folder)


# ---------------------------------------------------------------------------
# CLI error-log schema
# ---------------------------------------------------------------------------

def test_cli_error_log_schema(tmp_path, monkeypatch):
    """Any exception inside a CLI entry point writes a per-rank JSON error
    log with rank/host/exception fields (reference __main__.py:726-749)."""
    from modalities_amd.__main__ import _exception_handling

    def boom():
        raise RuntimeError("synthetic failure xyz")

    with pytest.raises(RuntimeError, match="synthetic failure xyz"):
        _exception_handling(boom, error_log_dir=tmp_path)
    logs = list(tmp_path.glob("error_rank_*.json"))
    assert len(logs) == 1
    rec = json.loads(logs[0].read_text())
    for key in ("rank", "hostname", "exception_type", "message", "traceback"):
        assert key in rec, rec
    assert rec["exception_type"] == "RuntimeError"
    assert "synthetic failure xyz" in rec["message"]


def test_sampler_skip_past_end_raises():
    from modalities_amd.dataloader.samplers import ResumableDistributedSampler
    with pytest.raises(ValueError, match="nothing left"):
        ResumableDistributedSampler(dataset=list(range(10)), rank=0,
                                    num_replicas=2, skip_num_global_samples=10)
