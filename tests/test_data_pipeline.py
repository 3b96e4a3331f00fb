"""Data pipeline tests: indexing, packing, shuffling, chunking, filtering,
merging, instruction tuning (reference test model:
tests/dataloader/test_packed_dataset.py, test_end_to_end_indexation_and_
tokenization.py, preprocessing tests)."""

import json
import pickle
from pathlib import Path

import numpy as np
import pytest

from modalities_amd import api
from modalities_amd.dataloader.create_index import (IndexGenerator,
                                                    LargeFileLinesReader)
from modalities_amd.dataloader.packed_data import EmbeddedStreamData
from modalities_amd.preprocessing.shuffle_data import (
    create_filtered_tokenized_dataset, create_shuffled_dataset_chunk,
    shuffle_jsonl_data, shuffle_tokenized_data)
from modalities_amd.tokenization.tokenizer_wrapper import CharTokenizer


@pytest.fixture
def jsonl_corpus(tmp_path) -> Path:
    p = tmp_path / "corpus.jsonl"
    with p.open("w") as f:
        for i in range(20):
            f.write(json.dumps({"text": f"document number {i} " + "lorem " * i}) + "\n")
    return p


def test_create_index_and_reader(jsonl_corpus):
    idx_path = jsonl_corpus.with_suffix(".idx")
    n = IndexGenerator(jsonl_corpus).create_index(idx_path)
    assert n == 20
    reader = LargeFileLinesReader(jsonl_corpus, idx_path)
    assert len(reader) == 20
    assert json.loads(reader[3])["text"].startswith("document number 3")
    assert json.loads(reader[19])["text"].startswith("document number 19")
    reader.close()


def test_pack_encoded_data_end_to_end(jsonl_corpus, tmp_path):
    api.create_raw_data_index(jsonl_corpus)
    dst = tmp_path / "corpus.pbin"
    config = {
        "settings": {"src_path": str(jsonl_corpus), "dst_path": str(dst),
                     "jq_pattern": ".text", "eod_token": "<eod>"},
        "tokenizer": {"component_key": "tokenizer", "variant_key": "char",
                      "config": {}},
    }
    n = api.pack_encoded_data(config)
    assert n == 20
    data = EmbeddedStreamData(dst)
    assert len(data.index_base) == 20
    tok = CharTokenizer()
    # roundtrip doc 5: text + eod token
    start, length = data.index_base[5]
    ids = data.tokens(start, length).tolist()
    assert ids[-1] == tok.get_token_id("<eod>")
    assert tok.decode(ids[:-1]).startswith("document number 5")


def test_shuffle_tokenized_data_preserves_docs(tmp_path):
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [np.arange(i, i + 5, dtype=np.uint16) for i in range(10)]
    src = tmp_path / "a.pbin"
    write_pbin(src, docs, 2)
    dst = tmp_path / "b.pbin"
    shuffle_tokenized_data(src, dst, batch_size=3, seed=1)
    out = EmbeddedStreamData(dst)
    assert len(out.index_base) == 10
    originals = {tuple(d.tolist()) for d in docs}
    shuffled = [tuple(out.tokens(s, n).tolist()) for s, n in out.index_base]
    assert set(shuffled) == originals
    assert [tuple(d.tolist()) for d in docs] != shuffled  # actually shuffled


def test_shuffle_jsonl(tmp_path, jsonl_corpus):
    dst = tmp_path / "shuffled.jsonl"
    shuffle_jsonl_data(jsonl_corpus, dst, seed=3)
    src_lines = sorted(jsonl_corpus.read_text().splitlines())
    dst_lines = sorted(dst.read_text().splitlines())
    assert src_lines == dst_lines


def test_chunking_covers_all_docs(tmp_path):
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [np.arange(i, i + 4, dtype=np.uint16) for i in range(12)]
    src = tmp_path / "a.pbin"
    write_pbin(src, docs, 2)
    chunks = []
    for cid in range(3):
        out = tmp_path / f"chunk{cid}.pbin"
        create_shuffled_dataset_chunk([src], out, cid, 3, global_seed=5)
        data = EmbeddedStreamData(out)
        chunks.extend(tuple(data.tokens(s, n).tolist()) for s, n in data.index_base)
    assert sorted(chunks) == sorted(tuple(d.tolist()) for d in docs)


def test_filtering(tmp_path):
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [np.full(4, i, dtype=np.uint16) for i in range(10)]
    src = tmp_path / "a.pbin"
    write_pbin(src, docs, 2)
    dst = tmp_path / "f.pbin"
    kept = create_filtered_tokenized_dataset(
        src, dst, lambda i, toks: toks[0] % 2 == 0)
    assert kept == 5
    out = EmbeddedStreamData(dst)
    for s, n in out.index_base:
        assert out.tokens(s, n)[0] % 2 == 0


def test_merge_packed_data(tmp_path):
    from modalities_amd.dataloader.packed_data import write_pbin
    a = [np.arange(3, dtype=np.uint16), np.arange(4, dtype=np.uint16)]
    b = [np.arange(5, dtype=np.uint16)]
    pa, pb = tmp_path / "a.pbin", tmp_path / "b.pbin"
    write_pbin(pa, a, 2)
    write_pbin(pb, b, 2)
    merged = tmp_path / "m.pbin"
    api.merge_packed_data_files([pa, pb], merged)
    out = EmbeddedStreamData(merged)
    assert len(out.index_base) == 3
    assert out.tokens(*out.index_base[2]).tolist() == list(range(5))


def test_instruction_tuning_pipeline(tmp_path):
    src = tmp_path / "chats.jsonl"
    with src.open("w") as f:
        for i in range(10):
            f.write(json.dumps({"conversations": [
                {"role": "user", "content": f"question {i}"},
                {"role": "assistant", "content": f"answer {i}"},
            ]}) + "\n")
    cfg = {
        "settings": {"src_path": str(src), "dst_dir": str(tmp_path / "out"),
                     "split_ratios": {"train": 0.8, "val": 0.2, "test": 0.0},
                     "eod_token": "<eod>"},
        "tokenizer": {"component_key": "tokenizer", "variant_key": "char",
                      "config": {}},
    }
    import yaml
    cfg_path = tmp_path / "it.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    from modalities_amd.dataloader.instruction_tuning import \
        create_instruction_tuning_data
    packed = create_instruction_tuning_data(cfg_path)
    assert set(packed) == {"train", "val"}
    train = EmbeddedStreamData(packed["train"])
    assert len(train.index_base) == 8
    tok = CharTokenizer()
    ids = train.tokens(*train.index_base[0]).tolist()
    text = tok.decode(ids)
    # assistant turn is bracketed by the loss-mask markers
    assert "^" in text and "$" in text and "Assistant: answer" in text


def test_pbin_byte_compat_with_reference_layout(tmp_path):
    """Header = 8B data len + 4B token size; index pickled after data
    (reference create_packed_data.py:346-405)."""
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [np.asarray([1, 2, 3], dtype=np.uint16)]
    p = tmp_path / "x.pbin"
    write_pbin(p, docs, 2)
    raw = p.read_bytes()
    data_len = int.from_bytes(raw[:8], "little")
    token_size = int.from_bytes(raw[8:12], "little")
    assert data_len == 6 and token_size == 2
    assert raw[12:18] == np.asarray([1, 2, 3], dtype="<u2").tobytes()
    index = pickle.loads(raw[18:])
    assert index == [(0, 6)]


def test_memmap_dataset_tokenize_on_the_fly(tmp_path):
    """MemMapDataset (reference dataset.py:134-188): jsonl -> index ->
    per-item tokenize must match direct tokenization."""
    from modalities_amd.api import create_raw_data_index
    from modalities_amd.dataloader.dataset import MemMapDataset
    from modalities_amd.tokenization.tokenizer_wrapper import CharTokenizer

    src = tmp_path / "corpus.jsonl"
    texts = ["hello world", "lorem ipsum dolor", "a"]
    src.write_text("\n".join(json.dumps({"text": t}) for t in texts) + "\n")
    idx = tmp_path / "corpus.idx"
    create_raw_data_index(src, idx)
    tok = CharTokenizer()
    ds = MemMapDataset(src, tok, sample_key="input_ids", index_path=idx)
    assert len(ds) == 3
    for i, t in enumerate(texts):
        assert ds[i]["input_ids"].tolist() == list(tok.tokenize(t))
    with pytest.raises(IndexError):
        ds[3]


def test_verify_tokenization_consistency(tmp_path):
    from modalities_amd.utils.verify_tokenization_consistency import (
        verify_index, verify_tokenization_consistency)

    src = tmp_path / "corpus.jsonl"
    texts = ["hello world", "lorem ipsum dolor sit amet", "xyz"]
    src.write_text("\n".join(json.dumps({"text": t}) for t in texts) + "\n")
    n = verify_tokenization_consistency(
        src, tokenizer_config={"component_key": "tokenizer",
                               "variant_key": "char", "config": {}})
    assert n == 3

    # corrupt index must be detected
    import pickle
    from modalities_amd.api import create_raw_data_index
    idx = tmp_path / "corpus.idx"
    create_raw_data_index(src, idx)
    with open(idx, "rb") as f:
        index = pickle.load(f)
    index[1] = (index[1][0] + 2, index[1][1])
    bad = tmp_path / "bad.idx"
    with open(bad, "wb") as f:
        pickle.dump(index, f)
    with pytest.raises(AssertionError):
        verify_index(src, bad)
