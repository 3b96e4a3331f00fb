"""Instruction-tuning data preparation (capability parity with reference
src/modalities/dataloader/apply_chat_template.py:15-148 and
create_instruction_tuning_data.py): apply a chat template to conversation
JSONL, split train/val/test, index and pack each split.

The reference uses a jinja2 sandbox; we use Python str.format on a
{role}/{content} template plus special begin/end-of-assistant markers that
the LossMaskingCollateFnWrapper later uses to mask non-assistant tokens."""

import json
import random
from pathlib import Path

import yaml


def apply_chat_template_to_conversation(conversation: list[dict],
                                        role_templates: dict[str, str],
                                        chat_template: str,
                                        assistant_role: str = "assistant",
                                        b_include_to_loss_token: str = "^",
                                        e_include_to_loss_token: str = "$") -> str:
    """conversation: list of {"role": ..., "content": ...}."""
    turns = []
    for msg in conversation:
        role, content = msg["role"], msg["content"]
        tmpl = role_templates.get(role, "{role}: {content}\n")
        turn = tmpl.format(role=role, content=content)
        if role == assistant_role:
            turn = f"{b_include_to_loss_token}{turn}{e_include_to_loss_token}"
        turns.append(turn)
    return chat_template.format(chat="".join(turns))


def apply_chat_template(config: dict) -> dict[str, Path]:
    """Read conversations JSONL, write templated JSONL per split.
    Config keys (settings): src_path, dst_dir, conversations_key,
    split_ratios {train,val,test}, seed; jinja2-free template under
    chat_template / role_templates."""
    settings = config["settings"]
    src = Path(settings["src_path"])
    dst_dir = Path(settings["dst_dir"])
    dst_dir.mkdir(parents=True, exist_ok=True)
    conversations_key = settings.get("conversations_key", "conversations")
    ratios = settings.get("split_ratios", {"train": 0.95, "val": 0.05, "test": 0.0})
    seed = settings.get("seed", 42)
    chat_template = config.get("chat_template", "{chat}")
    role_templates = config.get("role_templates",
                                {"user": "User: {content}\n",
                                 "assistant": "Assistant: {content}\n"})
    special = config.get("special_tokens", {})
    b_tok = special.get("b_include_to_loss_token", "^")
    e_tok = special.get("e_include_to_loss_token", "$")

    with src.open() as f:
        rows = [json.loads(ln) for ln in f if ln.strip()]
    rng = random.Random(seed)
    rng.shuffle(rows)
    n = len(rows)
    n_train = int(n * ratios.get("train", 0.95))
    n_val = int(n * ratios.get("val", 0.05))
    splits = {"train": rows[:n_train], "val": rows[n_train:n_train + n_val],
              "test": rows[n_train + n_val:]}
    out_paths = {}
    for split, split_rows in splits.items():
        if not split_rows:
            continue
        out = dst_dir / f"{split}.jsonl"
        with out.open("w") as f:
            for row in split_rows:
                text = apply_chat_template_to_conversation(
                    row[conversations_key], role_templates, chat_template,
                    b_include_to_loss_token=b_tok, e_include_to_loss_token=e_tok)
                f.write(json.dumps({"text": text}) + "\n")
        out_paths[split] = out
    return out_paths


def create_instruction_tuning_data(config_file_path: Path) -> dict[str, Path]:
    """Full pipeline: chat template -> index -> pack per split (reference
    create_instruction_tuning_data.py)."""
    from modalities_amd import api

    with open(config_file_path) as f:
        config = yaml.safe_load(f)
    jsonl_paths = apply_chat_template(config)
    packed = {}
    for split, jsonl_path in jsonl_paths.items():
        api.create_raw_data_index(jsonl_path,
                                  file_existence_policy=api.FileExistencePolicy.OVERRIDE)
        pack_cfg = {
            "settings": {
                "src_path": str(jsonl_path),
                "dst_path": str(jsonl_path.with_suffix(".pbin")),
                "jq_pattern": ".text",
                "eod_token": config.get("settings", {}).get("eod_token",
                                                            "<|endoftext|>"),
            },
            "tokenizer": config["tokenizer"],
        }
        api.pack_encoded_data(pack_cfg,
                              file_existence_policy=api.FileExistencePolicy.OVERRIDE)
        packed[split] = jsonl_path.with_suffix(".pbin")
    return packed
