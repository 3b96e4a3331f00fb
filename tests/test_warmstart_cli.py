"""Warmstart e2e through the full config/Main path (the reference's flagship
correctness test, reference tests/end2end_tests/test_fsdp_warmstart.py:54-160):
train 8 steps with checkpoints; restart from the step-4 checkpoint via a
warmstart config (${warmstart_env:...} resolver + number_conversion over the
checkpoint path); assert the logged losses of steps 5-8 are identical."""

import json
from pathlib import Path

import numpy as np
import pytest
import yaml

from modalities_amd.dataloader.packed_data import write_pbin
from modalities_amd.main import Main


def _prepare(tmp_path):
    rng = np.random.default_rng(7)
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)
    template = Path(__file__).parent / "configs" / "config_tiny_e2e.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    return text


def _losses_by_step(results_file: Path) -> dict[int, float]:
    out = {}
    for ln in Path(results_file).read_text().splitlines():
        r = json.loads(ln)
        if r.get("dataloader_tag") == "train":
            out[r["num_train_steps_done"]] = r["losses"]["CLMCrossEntropyLoss average"]
    return out


def test_warmstart_through_config_path(tmp_path):
    # ---- run A: uninterrupted 8 steps, checkpoints every 4 --------------
    text = _prepare(tmp_path)
    cfg_a = tmp_path / "a.yaml"
    cfg_a.write_text(text.replace("RESULTS_PATH_PLACEHOLDER",
                                  str(tmp_path / "a_results.jsonl")))
    main_a = Main(cfg_a, experiment_id="expA")
    main_a.run(main_a.build_components())
    losses_a = _losses_by_step(tmp_path / "a_results.jsonl")
    assert set(losses_a) == {2, 4, 6, 8}

    # the step-4 checkpoint of run A is the warmstart source
    ckpt_root = tmp_path / "ckpt" / "expA"
    step4 = [p for p in ckpt_root.iterdir() if "seen_steps_4" in p.name]
    assert step4, list(ckpt_root.iterdir())
    checkpoint_folder = step4[0]

    # ---- run B: fresh process-state, warmstart from step 4 --------------
    cfg_dict = yaml.safe_load(text.replace("RESULTS_PATH_PLACEHOLDER",
                                           str(tmp_path / "b_results.jsonl")))
    cfg_dict["app_state"] = {
        "component_key": "app_state", "variant_key": "sharded_warmstart",
        "config": {
            "model": {"instance_key": "wrapped_model", "pass_type": "BY_REFERENCE"},
            "optimizer": {"instance_key": "optimizer", "pass_type": "BY_REFERENCE"},
            "lr_scheduler": {"instance_key": "scheduler", "pass_type": "BY_REFERENCE"},
            "checkpoint_folder_path": "${warmstart_env:checkpoint_folder_path}",
        },
    }
    cfg_dict["settings"]["training_progress"] = {
        "num_seen_steps": {
            "component_key": "number_conversion",
            "variant_key": "num_seen_steps_from_checkpoint_path",
            "config": {"checkpoint_path": "${warmstart_env:checkpoint_folder_path}"},
        },
        "global_num_seen_tokens": {
            "component_key": "number_conversion",
            "variant_key": "global_num_seen_tokens_from_checkpoint_path",
            "config": {"checkpoint_path": "${warmstart_env:checkpoint_folder_path}"},
        },
    }
    cfg_dict["train_sampler"]["config"]["skip_num_global_samples"] = {
        "component_key": "number_conversion",
        "variant_key": "num_samples_from_num_steps",
        "config": {
            "num_steps": {
                "component_key": "number_conversion",
                "variant_key": "num_seen_steps_from_checkpoint_path",
                "config": {"checkpoint_path": "${warmstart_env:checkpoint_folder_path}"},
            },
            "dp_degree": 1, "local_micro_batch_size": 2,
            "gradient_accumulation_steps": 1,
        },
    }
    cfg_b = tmp_path / "b.yaml"
    cfg_b.write_text(yaml.safe_dump(cfg_dict, sort_keys=False))
    main_b = Main(cfg_b, experiment_id="expB", additional_resolver_funs={
        "warmstart_env": lambda key: {
            "checkpoint_folder_path": str(checkpoint_folder)}[key]})
    components_b = main_b.build_components()
    assert components_b.settings.training_progress.num_seen_steps == 4
    main_b.run(components_b)

    losses_b = _losses_by_step(tmp_path / "b_results.jsonl")
    assert set(losses_b) == {6, 8}
    for step in (6, 8):
        assert losses_b[step] == pytest.approx(losses_a[step], rel=1e-6), \
            (step, losses_a, losses_b)
