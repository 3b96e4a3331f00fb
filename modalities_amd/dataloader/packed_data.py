"""The .pbin packed-token file format (byte-compatible with the reference).

Layout (reference: src/modalities/dataloader/create_packed_data.py:346-405):
  [ 8 B LE: data-section length in bytes ]
  [ 4 B LE: token size in bytes (1/2/4)  ]
  [ data section: tokens, little-endian  ]
  [ pickled list[(start_byte, len_bytes)] document index, offsets relative
    to the data-section start ]

This module provides reading (mmap), writing, merging and a multiprocess
tokenize->pack pipeline equivalent to the reference's PackedDataGenerator
(create_packed_data.py:138-324), implemented fresh for this framework.
"""

import multiprocessing as mp
import pickle
from pathlib import Path
from typing import Callable, Iterable, Optional

import numpy as np

DATA_SECTION_LENGTH_IN_BYTES = 8
TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES = 4
HEADER_SIZE_IN_BYTES = DATA_SECTION_LENGTH_IN_BYTES + TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES


def _np_dtype_for_token_size(token_size_in_bytes: int) -> np.dtype:
    try:
        return {1: np.dtype(np.uint8), 2: np.dtype(np.uint16), 4: np.dtype(np.uint32)}[
            token_size_in_bytes
        ].newbyteorder("<")
    except KeyError:
        raise ValueError(f"Unsupported token size {token_size_in_bytes} (need 1, 2 or 4)")


def token_size_for_vocab(vocab_size: int) -> int:
    if vocab_size <= 2**8:
        return 1
    if vocab_size <= 2**16:
        return 2
    return 4


class EmbeddedStreamData:
    """mmap view of a .pbin file: header, data section, document index."""

    HEADER_SIZE_IN_BYTES = HEADER_SIZE_IN_BYTES

    def __init__(self, data_path: Path, load_index: bool = True):
        self._data_path = Path(data_path)
        if not self._data_path.is_file():
            raise FileNotFoundError(f"Packed data not found at {self._data_path.absolute()}. "
                                    f"Create it with `modalities-amd data pack_encoded_data`.")
        file_size = self._data_path.stat().st_size
        with self._data_path.open("rb") as f:
            self.data_len = int.from_bytes(f.read(DATA_SECTION_LENGTH_IN_BYTES), "little")
            self.token_size_in_bytes = int.from_bytes(
                f.read(TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES), "little", signed=False)
            if self.token_size_in_bytes not in (1, 2, 4):
                raise ValueError(
                    f"Corrupt .pbin header in {self._data_path}: token size "
                    f"{self.token_size_in_bytes} not in (1, 2, 4)")
            if HEADER_SIZE_IN_BYTES + self.data_len > file_size:
                raise ValueError(
                    f"Truncated .pbin file {self._data_path}: header claims "
                    f"{self.data_len} data bytes but the file has only "
                    f"{file_size - HEADER_SIZE_IN_BYTES} after the header")
            if load_index:
                f.seek(HEADER_SIZE_IN_BYTES + self.data_len)
                try:
                    self._index_base: Optional[list[tuple[int, int]]] = \
                        pickle.loads(f.read())
                except Exception as e:
                    raise ValueError(
                        f"Corrupt .pbin document index in {self._data_path} "
                        f"(truncated file or bad header?): {e}") from e
            else:
                self._index_base = None
        self._data = np.memmap(self._data_path, mode="r", offset=HEADER_SIZE_IN_BYTES,
                               shape=(self.data_len,))

    @property
    def index_base(self) -> list[tuple[int, int]]:
        if self._index_base is None:
            raise ValueError("Index was not loaded (load_index=False).")
        return self._index_base

    @property
    def data(self) -> np.ndarray:
        """Raw byte view of the data section."""
        return self._data

    @property
    def token_dtype(self) -> np.dtype:
        return _np_dtype_for_token_size(self.token_size_in_bytes)

    @property
    def num_tokens(self) -> int:
        return self.data_len // self.token_size_in_bytes

    def tokens(self, offset_bytes: int, length_bytes: int) -> np.ndarray:
        buf = self._data[offset_bytes:offset_bytes + length_bytes]
        return np.frombuffer(buf, dtype=self.token_dtype)


def write_pbin(out_path: Path, documents: Iterable[np.ndarray], token_size_in_bytes: int) -> int:
    """Write documents (arrays of token ids) into a .pbin. Returns doc count."""
    dtype = _np_dtype_for_token_size(token_size_in_bytes)
    index: list[tuple[int, int]] = []
    offset = 0
    out_path = Path(out_path)
    with out_path.open("wb") as f:
        f.write((0).to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))
        f.write(token_size_in_bytes.to_bytes(TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES, "little"))
        for doc in documents:
            arr = np.asarray(doc).astype(dtype, copy=False)
            raw = arr.tobytes()
            f.write(raw)
            index.append((offset, len(raw)))
            offset += len(raw)
        f.write(pickle.dumps(index))
        # header fixup with the real data length (reference: :327-344)
        f.seek(0)
        f.write(offset.to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))
    return len(index)


def join_embedded_stream_data(stream_files: list[Path], target_file: Path) -> None:
    """Merge multiple .pbin files into one (reference: :407-458)."""
    streams = [EmbeddedStreamData(p) for p in stream_files]
    token_sizes = {s.token_size_in_bytes for s in streams}
    if len(token_sizes) != 1:
        raise ValueError(f"Cannot merge pbin files with mixed token sizes: {token_sizes}")
    token_size = token_sizes.pop()
    index: list[tuple[int, int]] = []
    offset = 0
    with Path(target_file).open("wb") as f:
        f.write((0).to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))
        f.write(token_size.to_bytes(TOKEN_SIZE_DESCRIPTOR_LENGTH_IN_BYTES, "little"))
        for s in streams:
            f.write(s.data.tobytes())
            for start, length in s.index_base:
                index.append((offset + start, length))
            offset += s.data_len
        f.write(pickle.dumps(index))
        f.seek(0)
        f.write(offset.to_bytes(DATA_SECTION_LENGTH_IN_BYTES, "little"))


# ---------------------------------------------------------------------------
# Tokenize -> pack pipeline (offline preprocessing)
# ---------------------------------------------------------------------------

def _process_lines(args):
    lines, tokenize_fn_factory, eod_token_id, token_size = args
    tokenize = tokenize_fn_factory()
    dtype = _np_dtype_for_token_size(token_size)
    out = []
    for line_id, text in lines:
        ids = tokenize(text)
        if not ids:
            continue
        arr = np.asarray(list(ids) + [eod_token_id], dtype=dtype)
        out.append((line_id, arr.tobytes()))
    return out


class PackedDataGenerator:
    """Tokenize a jsonl/text corpus into a .pbin.

    For simplicity and determinism the parallel path uses a process pool with
    ordered chunk results (the reference uses explicit reader/processor/writer
    processes with a reassembly dict, create_packed_data.py:172-283; the result
    is identical: documents appear in input order)."""

    def __init__(self, texts: Iterable[tuple[int, str]],
                 tokenize_fn_factory: Callable[[], Callable[[str], list[int]]],
                 eod_token_id: int, vocab_size: int, num_processes: int = 1):
        self._texts = texts
        self._tokenize_fn_factory = tokenize_fn_factory
        self._eod = eod_token_id
        self._token_size = token_size_for_vocab(vocab_size)
        self._num_processes = num_processes

    def run(self, out_path: Path, chunk_size: int = 512) -> int:
        def chunks():
            buf = []
            for item in self._texts:
                buf.append(item)
                if len(buf) >= chunk_size:
                    yield buf
                    buf = []
            if buf:
                yield buf

        def doc_stream():
            if self._num_processes <= 1:
                for chunk in chunks():
                    for _, raw in _process_lines(
                            (chunk, self._tokenize_fn_factory, self._eod, self._token_size)):
                        yield np.frombuffer(raw, dtype=_np_dtype_for_token_size(self._token_size))
            else:
                with mp.get_context("spawn").Pool(self._num_processes) as pool:
                    args = ((c, self._tokenize_fn_factory, self._eod, self._token_size)
                            for c in chunks())
                    for result in pool.imap(_process_lines, args):
                        for _, raw in result:
                            yield np.frombuffer(raw,
                                                dtype=_np_dtype_for_token_size(self._token_size))

        return write_pbin(out_path, doc_stream(), self._token_size)
