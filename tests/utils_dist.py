"""Multi-process test harness: run a function under N gloo-backed ranks on
CPU (mirrors the reference's MultiProcessingCudaEnv test pattern, reference:
tests/end2end_tests/custom_components.py:29-70)."""

import multiprocessing as mp
import os
import traceback

import torch.distributed as dist


def _worker(rank: int, world_size: int, port: int, fn, args, err_q, ret_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        result = fn(rank, world_size, *args)
        ret_q.put((rank, result))
    except Exception:
        err_q.put((rank, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world_size: int, port: int, args=(), timeout_s: int = 180):
    """Spawn world_size processes running fn(rank, world_size, *args).
    Returns {rank: result}. Raises on any rank failure."""
    ctx = mp.get_context("spawn")
    err_q = ctx.Queue()
    ret_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world_size, port, fn, args, err_q, ret_q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    import queue as queue_mod
    import time
    deadline = time.time() + timeout_s
    while len(results) < world_size and time.time() < deadline:
        if not err_q.empty():
            rank, tb = err_q.get()
            for p in procs:
                p.terminate()
            raise RuntimeError(f"rank {rank} failed:\n{tb}")
        try:
            rank, result = ret_q.get(timeout=1.0)
            results[rank] = result
        except queue_mod.Empty:
            pass
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if len(results) < world_size:
        if not err_q.empty():
            rank, tb = err_q.get()
            raise RuntimeError(f"rank {rank} failed:\n{tb}")
        raise TimeoutError(f"only {len(results)}/{world_size} ranks finished")
    return results
