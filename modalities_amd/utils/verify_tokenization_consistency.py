"""Tokenization-consistency verification (capability parity with reference
src/modalities/utils/verify_tokenization_consistency.py): check that the
multiprocess index -> tokenize -> pack pipeline produces exactly the token
stream that direct per-line tokenization yields, and that the byte-offset
index reproduces the raw lines."""

import json
import pickle
import tempfile
from pathlib import Path
from typing import Callable


def verify_index(src_path: Path, index_path: Path) -> None:
    """Every (offset, length) entry must reproduce its raw JSONL line."""
    src_path, index_path = Path(src_path), Path(index_path)
    raw = src_path.read_bytes()
    with open(index_path, "rb") as f:
        index = pickle.load(f)
    lines = [ln for ln in raw.split(b"\n") if ln.strip()]
    if len(index) != len(lines):
        raise AssertionError(
            f"index has {len(index)} entries for {len(lines)} non-empty lines")
    for i, (off, length) in enumerate(index):
        got = raw[off:off + length].strip(b"\n")
        if got.strip() != lines[i].strip():
            raise AssertionError(f"index entry {i} does not reproduce line "
                                 f"{i}: {got[:80]!r} != {lines[i][:80]!r}")


def verify_tokenization_consistency(src_path: Path, tokenizer_config: dict,
                                    eod_token: str = "<|endoftext|>",
                                    jq_pattern: str = ".text") -> int:
    """Index + pack `src_path` through the production pipeline
    (api.create_raw_data_index / api.pack_encoded_data), then re-read the
    .pbin and compare each document against direct tokenization of the
    same text. Returns the number of verified documents; raises
    AssertionError on any mismatch (reference
    verify_tokenization_consistency.py)."""
    from modalities_amd.api import (FileExistencePolicy, create_raw_data_index,
                                    pack_encoded_data)
    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.dataloader.dataset import PackedMemMapDatasetBase
    from modalities_amd.registry.components import get_default_registry

    src_path = Path(src_path)
    field = jq_pattern.lstrip(".")
    with tempfile.TemporaryDirectory() as td:
        index_path = Path(td) / "data.idx"
        pbin_path = Path(td) / "data.pbin"
        create_raw_data_index(src_path, index_path,
                              file_existence_policy=FileExistencePolicy.OVERRIDE)
        verify_index(src_path, index_path)
        cfg = {"settings": {"src_path": str(src_path),
                            "index_path": str(index_path),
                            "dst_path": str(pbin_path),
                            "jq_pattern": jq_pattern,
                            "eod_token": eod_token},
               "tokenizer": tokenizer_config}
        pack_encoded_data(cfg,
                          file_existence_policy=FileExistencePolicy.OVERRIDE)

        factory = ComponentFactory(get_default_registry())
        tokenizer = factory.build_component_by_key(cfg, "tokenizer")
        eod_id = tokenizer.get_token_id(eod_token)
        ds = PackedMemMapDatasetBase(pbin_path, sample_key="input_ids")
        n = 0
        with open(src_path, encoding="utf-8") as f:
            lines = [ln for ln in f if ln.strip()]
        di = 0
        for line in lines:
            text = json.loads(line)
            for part in field.split("."):
                if part:
                    text = text[part]
            expected = list(tokenizer.tokenize(text)) + [eod_id]
            if not expected[:-1]:
                continue  # empty tokenization is dropped by the packer
            got = ds[di]["input_ids"].tolist()
            if got != expected:
                raise AssertionError(
                    f"document {di}: packed tokens differ from direct "
                    f"tokenization ({got[:8]}... vs {expected[:8]}...)")
            di += 1
            n += 1
        if di != len(ds):
            raise AssertionError(
                f"packed file has {len(ds)} documents, expected {di}")
    return n
