"""Raw-corpus indexing (capability parity with reference
src/modalities/dataloader/create_index.py:12-80 and
large_file_lines_reader.py:18-130): build a pickled list of
(start_byte, length_bytes) for every line of a JSONL file, and a mmap-backed
random-access line reader over it."""

import mmap
import pickle
from pathlib import Path
from typing import Optional


class IndexGenerator:
    def __init__(self, src_file: Path, drop_faulty_entries: bool = False):
        self.src_file = Path(src_file)
        self.drop_faulty_entries = drop_faulty_entries

    def create_index(self, target_path_index: Path) -> int:
        """Scan the file once, recording (start, len) of every non-empty
        line. Returns the number of indexed lines."""
        index: list[tuple[int, int]] = []
        with self.src_file.open("rb") as f:
            offset = 0
            for line in f:
                stripped = line.rstrip(b"\r\n")
                if stripped:
                    index.append((offset, len(stripped)))
                offset += len(line)
        with Path(target_path_index).open("wb") as f:
            pickle.dump(index, f)
        return len(index)


class LargeFileLinesReader:
    """Random access to lines of a large text file via its index + mmap."""

    def __init__(self, raw_data_path: Path, index_path: Optional[Path] = None,
                 encoding: str = "utf-8", use_sample_length_from_index: bool = True):
        self.raw_data_path = Path(raw_data_path)
        self.index_path = Path(index_path) if index_path is not None \
            else self.default_index_path(self.raw_data_path)
        self.encoding = encoding
        if not self.raw_data_path.is_file():
            raise FileNotFoundError(self.raw_data_path)
        if not self.index_path.is_file():
            raise FileNotFoundError(
                f"Index {self.index_path} not found; create it with "
                f"`modalities-amd data create_raw_index`")
        with self.index_path.open("rb") as f:
            self.index = pickle.load(f)
        self._f = self.raw_data_path.open("rb")
        self._mmap = mmap.mmap(self._f.fileno(), 0, access=mmap.ACCESS_READ)

    @staticmethod
    def default_index_path(raw_data_path: Path) -> Path:
        return raw_data_path.with_suffix(".idx")

    def __len__(self) -> int:
        return len(self.index)

    def __getitem__(self, key: int) -> str:
        offset, length = self.index[key]
        return self._mmap[offset:offset + length].decode(self.encoding)

    def close(self):
        self._mmap.close()
        self._f.close()
