"""Experiment-ID generation synced across ranks (capability parity with
reference src/modalities/util.py:54-140: rank 0 generates
<timestamp>_<config-hash>, broadcasts as a fixed-width uint8 tensor)."""

import hashlib
from datetime import datetime
from pathlib import Path

import torch
import torch.distributed as dist


def generate_experiment_id(config_path: Path) -> str:
    stamp = datetime.now().strftime("%Y-%m-%d__%H-%M-%S")
    h = hashlib.sha256(str(config_path).encode()).hexdigest()[:6]
    return f"{stamp}_{h}"


def get_synced_experiment_id_of_run(config_path: Path, max_len: int = 1024) -> str:
    if not (dist.is_available() and dist.is_initialized()):
        return generate_experiment_id(config_path)
    rank = dist.get_rank()
    buf = torch.zeros(max_len, dtype=torch.uint8)
    if rank == 0:
        raw = generate_experiment_id(config_path).encode("utf-8")[:max_len]
        buf[:len(raw)] = torch.tensor(list(raw), dtype=torch.uint8)
    if torch.cuda.is_available():
        buf = buf.cuda()
    dist.broadcast(buf, src=0)
    raw = bytes(b for b in buf.cpu().tolist() if b != 0)
    return raw.decode("utf-8")
