"""Offline data shuffling / chunking / filtering (capability parity with
reference src/modalities/preprocessing/shuffle_data.py:9-120,
create_chunks.py:9-100 and dataloader/filter_packed_data.py:13-80):

- shuffle a .pbin at document level (index permutation + data rewrite)
- shuffle a JSONL file at line level
- deterministic chunking of tokenized (.pbin) or jsonl datasets
- filter a .pbin by a document predicate
"""

import pickle
import random
from pathlib import Path
from typing import Callable, Optional

import numpy as np

from modalities_amd.dataloader.packed_data import (
    DATA_SECTION_LENGTH_IN_BYTES, EmbeddedStreamData,
    TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES, write_pbin)


def _write_pbin_from_byte_docs(out_path: Path, docs: list[bytes],
                               token_size_in_bytes: int) -> None:
    index = []
    offset = 0
    with Path(out_path).open("wb") as f:
        f.write((0).to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))
        f.write(token_size_in_bytes.to_bytes(
            TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES, "little"))
        for raw in docs:
            f.write(raw)
            index.append((offset, len(raw)))
            offset += len(raw)
        f.write(pickle.dumps(index))
        f.seek(0)
        f.write(offset.to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))


def shuffle_tokenized_data(input_data_path: Path, output_data_path: Path,
                           batch_size: int = 1024, seed: Optional[int] = None) -> None:
    """Shuffle documents of a .pbin (reference shuffle_data.py:60-120).
    batch_size controls how many docs are materialized per write batch."""
    data = EmbeddedStreamData(Path(input_data_path))
    index = list(data.index_base)
    rng = random.Random(seed)
    rng.shuffle(index)
    docs = (bytes(data.data[start:start + length]) for start, length in index)
    # stream in batches to bound memory
    out_index: list[tuple[int, int]] = []
    offset = 0
    with Path(output_data_path).open("wb") as f:
        f.write((0).to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))
        f.write(data.token_size_in_bytes.to_bytes(
            TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES, "little"))
        batch: list[bytes] = []
        for raw in docs:
            batch.append(raw)
            if len(batch) >= batch_size:
                for r in batch:
                    f.write(r)
                    out_index.append((offset, len(r)))
                    offset += len(r)
                batch = []
        for r in batch:
            f.write(r)
            out_index.append((offset, len(r)))
            offset += len(r)
        f.write(pickle.dumps(out_index))
        f.seek(0)
        f.write(offset.to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))


def shuffle_jsonl_data(input_data_path: Path, output_data_path: Path,
                       seed: Optional[int] = None) -> None:
    with Path(input_data_path).open("rb") as f:
        lines = [ln for ln in f.read().splitlines() if ln.strip()]
    rng = random.Random(seed)
    rng.shuffle(lines)
    with Path(output_data_path).open("wb") as f:
        for ln in lines:
            f.write(ln + b"\n")


def create_shuffled_dataset_chunk(file_path_list: list[Path], output_chunk_file_path: Path,
                                  chunk_id: int, num_chunks: int,
                                  global_seed: Optional[int] = None) -> None:
    """Deterministic chunk: take the chunk_id-th slice of every input pbin's
    documents, concatenate, shuffle with a chunk-derived seed (reference
    create_chunks.py:9-100)."""
    from modalities_amd.utils.seeding import calculate_hashed_seed
    docs: list[bytes] = []
    token_size = None
    for p in file_path_list:
        data = EmbeddedStreamData(Path(p))
        token_size = data.token_size_in_bytes if token_size is None else token_size
        if data.token_size_in_bytes != token_size:
            raise ValueError("mixed token sizes across chunk inputs")
        index = data.index_base
        n = len(index)
        lo = chunk_id * n // num_chunks
        hi = (chunk_id + 1) * n // num_chunks
        for start, length in index[lo:hi]:
            docs.append(bytes(data.data[start:start + length]))
    if not docs:
        raise ValueError(f"Chunk {chunk_id}/{num_chunks} selects no documents")
    seed = calculate_hashed_seed([str(global_seed), str(chunk_id)]) \
        if global_seed is not None else None
    random.Random(seed).shuffle(docs)
    _write_pbin_from_byte_docs(output_chunk_file_path, docs, token_size)


def create_shuffled_jsonl_dataset_chunk(file_path_list: list[Path],
                                        output_chunk_file_path: Path, chunk_id: int,
                                        num_chunks: int,
                                        global_seed: Optional[int] = None) -> None:
    lines: list[bytes] = []
    for p in file_path_list:
        with Path(p).open("rb") as f:
            file_lines = [ln for ln in f.read().splitlines() if ln.strip()]
        n = len(file_lines)
        lo = chunk_id * n // num_chunks
        hi = (chunk_id + 1) * n // num_chunks
        lines.extend(file_lines[lo:hi])
    from modalities_amd.utils.seeding import calculate_hashed_seed
    seed = calculate_hashed_seed([str(global_seed), str(chunk_id)]) \
        if global_seed is not None else None
    random.Random(seed).shuffle(lines)
    with Path(output_chunk_file_path).open("wb") as f:
        for ln in lines:
            f.write(ln + b"\n")


def create_filtered_tokenized_dataset(input_data_path: Path, output_data_path: Path,
                                      filter_routine: Callable[[int, np.ndarray], bool]
                                      ) -> int:
    """Copy documents passing filter_routine(doc_idx, token_array) into a new
    .pbin (reference filter_packed_data.py:13-80). Returns kept count."""
    data = EmbeddedStreamData(Path(input_data_path))
    kept: list[bytes] = []
    for i, (start, length) in enumerate(data.index_base):
        tokens = data.tokens(start, length)
        if filter_routine(i, tokens):
            kept.append(bytes(data.data[start:start + length]))
    _write_pbin_from_byte_docs(output_data_path, kept, data.token_size_in_bytes)
    return len(kept)
