"""Hashed seed derivation (capability parity with reference
src/modalities/utils/seeding.py:4-22): derive a deterministic child seed
from a base seed plus string tags."""

import hashlib


def calculate_hashed_seed(input_data: list[str], max_seed: int = 2**32 - 1) -> int:
    joined = "-".join(input_data)
    digest = hashlib.sha256(joined.encode("utf-8")).hexdigest()
    return int(digest, 16) % max_seed
