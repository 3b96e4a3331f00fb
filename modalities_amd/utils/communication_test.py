"""RCCL/gloo communication self-test (capability parity with reference
src/modalities/utils/communication_test.py:8-37, extended: all_gather +
all_reduce + reduce_scatter sanity before training starts — the first-class
comm-level validation the reference lacks)."""

import torch
import torch.distributed as dist


def run_communication_test(device=None) -> None:
    if not dist.is_initialized():
        return
    rank = dist.get_rank()
    world = dist.get_world_size()
    dev = device if device is not None else (
        torch.device("cuda", torch.cuda.current_device())
        if torch.cuda.is_available() else torch.device("cpu"))
    backend = dist.get_backend()

    # all_gather of rank ids
    x = torch.tensor([rank, rank + 1, rank + 2, rank + 3], device=dev)
    gathered = [torch.zeros_like(x) for _ in range(world)]
    dist.all_gather(gathered, x)
    for r, g in enumerate(gathered):
        expected = torch.tensor([r, r + 1, r + 2, r + 3], device=dev)
        if not torch.equal(g, expected):
            raise RuntimeError(f"all_gather mismatch at rank {rank}: {g} != {expected}")

    # all_reduce sum
    y = torch.ones(8, device=dev)
    dist.all_reduce(y)
    if not torch.equal(y, torch.full((8,), float(world), device=dev)):
        raise RuntimeError(f"all_reduce mismatch at rank {rank}: {y}")

    # reduce_scatter (RCCL path; emulate on gloo)
    z = torch.arange(world * 2, dtype=torch.float32, device=dev)
    if backend != "gloo":
        out = torch.zeros(2, device=dev)
        dist.reduce_scatter_tensor(out, z)
        expected = z[rank * 2:(rank + 1) * 2] * world
    else:
        dist.all_reduce(z)
        out = z[rank * 2:(rank + 1) * 2]
        expected = torch.arange(world * 2, dtype=torch.float32,
                                device=dev)[rank * 2:(rank + 1) * 2] * world
    if not torch.equal(out, expected):
        raise RuntimeError(f"reduce_scatter mismatch at rank {rank}: {out} != {expected}")
    if rank == 0:
        print(f"communication test passed (backend={backend}, world={world})",
              flush=True)
