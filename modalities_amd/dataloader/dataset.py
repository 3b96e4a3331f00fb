"""Datasets over packed token files (capability parity with reference
src/modalities/dataloader/dataset.py:76-464)."""

from pathlib import Path

import numpy as np
import torch
from torch.utils.data import Dataset

from modalities_amd.dataloader.packed_data import EmbeddedStreamData


class PackedMemMapDatasetBase(Dataset):
    """Document-level dataset over a .pbin: item i = tokens of document i
    (reference: dataset.py:191-309)."""

    def __init__(self, raw_data_path: Path, sample_key: str, load_index: bool = True):
        self.sample_key = sample_key
        self._embedded_stream_data = EmbeddedStreamData(raw_data_path, load_index=load_index)
        self._token_dtype = self._embedded_stream_data.token_dtype

    @property
    def token_size_in_bytes(self) -> int:
        return self._embedded_stream_data.token_size_in_bytes

    @property
    def num_tokens(self) -> int:
        return self._embedded_stream_data.num_tokens

    def __len__(self) -> int:
        return len(self._embedded_stream_data.index_base)

    def _tokens(self, offset: int, length: int) -> np.ndarray:
        return self._embedded_stream_data.tokens(offset, length)

    def __getitem__(self, idx: int) -> dict[str, torch.Tensor]:
        offset, length = self._embedded_stream_data.index_base[idx]
        tokens = np.asarray(self._tokens(offset, length))
        return {self.sample_key: torch.from_numpy(tokens.astype(np.int64))}


class PackedMemMapDatasetContinuous(PackedMemMapDatasetBase):
    """Fixed-length blocks over the continuous token stream.

    With reuse_last_target=True (default), consecutive blocks overlap by one
    token so every token is predicted exactly once after the shift-by-one
    collate (reference: dataset.py:312-401)."""

    def __init__(self, raw_data_path: Path, sample_key: str, block_size: int,
                 reuse_last_target: bool = True):
        super().__init__(raw_data_path, sample_key, load_index=False)
        if block_size < 2:
            raise ValueError(f"block_size must be >= 2, got {block_size}")
        self.block_size = block_size
        self._reuse_last_target = reuse_last_target
        total_tokens = self._embedded_stream_data.num_tokens
        if reuse_last_target:
            # blocks of block_size overlapping by 1 token
            self._num_samples = max(0, (total_tokens - 1) // (block_size - 1))
        else:
            self._num_samples = total_tokens // block_size

    def __len__(self) -> int:
        return self._num_samples

    def __getitem__(self, idx: int) -> dict[str, torch.Tensor]:
        if not 0 <= idx < self._num_samples:
            raise IndexError(f"index {idx} out of range ({self._num_samples})")
        tsz = self.token_size_in_bytes
        if self._reuse_last_target:
            start_tok = idx * (self.block_size - 1)
        else:
            start_tok = idx * self.block_size
        tokens = self._tokens(start_tok * tsz, self.block_size * tsz)
        return {self.sample_key: torch.from_numpy(np.asarray(tokens).astype(np.int64))}


class PackedMemMapDatasetMegatron(PackedMemMapDatasetBase):
    """Document-boundary-aware packing: blocks never span documents; short
    trailing fragments are dropped (reference: dataset.py:404-437)."""

    def __init__(self, raw_data_path: Path, sample_key: str, block_size: int):
        super().__init__(raw_data_path, sample_key, load_index=True)
        self.block_size = block_size
        tsz = self.token_size_in_bytes
        self._blocks: list[tuple[int, int]] = []  # (byte offset, byte length)
        for offset, length in self._embedded_stream_data.index_base:
            n_tokens = length // tsz
            for b in range(n_tokens // block_size):
                self._blocks.append((offset + b * block_size * tsz, block_size * tsz))

    def __len__(self) -> int:
        return len(self._blocks)

    def __getitem__(self, idx: int) -> dict[str, torch.Tensor]:
        offset, length = self._blocks[idx]
        tokens = self._tokens(offset, length)
        return {self.sample_key: torch.from_numpy(np.asarray(tokens).astype(np.int64))}


class DummyDataset(Dataset):
    """Random-sample dataset from a sample-shape spec
    (reference: dataset.py:76-131)."""

    def __init__(self, num_samples: int, sample_definition: list):
        # sample_definition entries: dicts/tuples (sample_key, sample_shape, sample_type)
        self.num_samples = num_samples
        self._defs = []
        for d in sample_definition:
            if isinstance(d, dict):
                self._defs.append((d["sample_key"], tuple(d["sample_shape"]), d["sample_type"]))
            else:
                self._defs.append((d[0], tuple(d[1]), d[2]))

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int) -> dict[str, torch.Tensor]:
        out = {}
        for key, shape, typ in self._defs:
            if typ == "int":
                out[key] = torch.randint(0, 512, shape, dtype=torch.int64)
            elif typ == "float":
                out[key] = torch.randn(*shape)
            else:
                raise ValueError(f"Unknown sample_type {typ}")
        return out


class CombinedDataset(Dataset):
    """Concatenation of datasets with cumulative-size dispatch
    (reference: dataset.py:440-464)."""

    def __init__(self, datasets: list[Dataset]):
        self.datasets = datasets
        self._cumsum = np.cumsum([len(d) for d in datasets])

    def __len__(self):
        return int(self._cumsum[-1]) if len(self.datasets) else 0

    def __getitem__(self, idx: int):
        ds_idx = int(np.searchsorted(self._cumsum, idx, side="right"))
        local = idx - (int(self._cumsum[ds_idx - 1]) if ds_idx > 0 else 0)
        return self.datasets[ds_idx][local]


class SyntheticLMDataset(Dataset):
    """Deterministic synthetic token blocks for benchmarking (no disk IO):
    seeds per index so every rank/epoch sees stable data."""

    def __init__(self, num_samples: int, sequence_length: int, vocab_size: int,
                 sample_key: str = "input_ids", seed: int = 1234):
        self.num_samples = num_samples
        self.sequence_length = sequence_length
        self.vocab_size = vocab_size
        self.sample_key = sample_key
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int) -> dict[str, torch.Tensor]:
        g = torch.Generator().manual_seed(self.seed + idx)
        return {self.sample_key: torch.randint(0, self.vocab_size,
                                               (self.sequence_length + 1,), generator=g)}


class MemMapDataset(Dataset):
    """Tokenize-on-the-fly dataset over a JSONL file via the byte-offset
    index (capability parity with reference dataloader/dataset.py:134-188):
    each __getitem__ mmap-reads one line, extracts the text field
    (jq_pattern, simple ".field[.sub]" selectors) and tokenizes it."""

    def __init__(self, raw_data_path: Path, tokenizer, sample_key: str,
                 index_path: Path = None, jq_pattern: str = ".text"):
        import json

        from modalities_amd.dataloader.create_index import LargeFileLinesReader
        self.sample_key = sample_key
        self.reader = LargeFileLinesReader(Path(raw_data_path),
                                           index_path=index_path)
        self.tokenizer = tokenizer
        self._fields = [f for f in jq_pattern.split(".") if f]
        self._json = json

    def __len__(self) -> int:
        return len(self.reader)

    def __getitem__(self, idx: int) -> dict[str, torch.Tensor]:
        if idx >= len(self.reader):
            raise IndexError("Index out of bounds")
        obj = self._json.loads(self.reader[idx])
        for f in self._fields:
            obj = obj[f]
        ids = self.tokenizer.tokenize(obj)
        return {self.sample_key: torch.tensor(ids, dtype=torch.long)}
