"""70B memory math: estimator correctness vs real models,
the documented 288-GB verdicts, and a world-8 gloo dry run of the bench
code path (tiny model, full-shard + reshard + AC: the exact flags the 70B
run uses)."""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

repo = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(repo))


def _cfg(name):
    from bench import build_model_cfg
    return build_model_cfg(name)


def test_param_count_matches_real_model():
    from modalities_amd.models.gpt2 import GPT2LLM
    from modalities_amd.utils.memory_budget import gpt2_param_count
    cfg = _cfg("gpt2-tiny")
    model = GPT2LLM(cfg)
    real = sum(p.numel() for p in model.parameters())
    est = gpt2_param_count(cfg)
    assert abs(est - real) / real < 0.01, (est, real)


def test_70b_fits_world8_not_world1():
    from modalities_amd.utils.memory_budget import estimate_sharded_memory
    cfg = _cfg("gpt2-70b")
    hbm = 288_000_000_000  # 288 GB (268 GiB)
    e8 = estimate_sharded_memory(cfg, world=8, micro_batch=1, full_ac=True)
    assert 69e9 < e8.n_params < 70e9
    assert e8.total_bytes < 0.75 * hbm         # plenty of headroom at 8
    e1 = estimate_sharded_memory(cfg, world=1, micro_batch=1, full_ac=True)
    assert e1.total_bytes > 2 * hbm            # impossible on one GPU
    # no-reshard keeps all gathered weights resident -> over budget even at 8
    e8n = estimate_sharded_memory(cfg, world=8, micro_batch=1, full_ac=True,
                                  reshard_after_forward=False)
    assert e8n.total_bytes > hbm


def test_27b_and_8b_fit_single_gpu():
    from modalities_amd.utils.memory_budget import estimate_sharded_memory
    hbm = 288_000_000_000
    e = estimate_sharded_memory(_cfg("gpt2-2.7b"), world=1, micro_batch=2)
    assert e.total_bytes < 0.5 * hbm
    e = estimate_sharded_memory(_cfg("gpt2-8b"), world=1, micro_batch=1,
                                full_ac=True)
    assert e.total_bytes < 0.75 * hbm


@pytest.mark.slow
def test_bench_world8_gloo_dry_run():
    """The 70B launch path (full-shard + --reshard + --ac, torchrun world
    8) on the tiny model: proves the engine's unit sizing, gather/reduce
    ordering and the bench contract at the 70B world size on CPU."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29371", str(repo / "bench.py"), "--gpus", "8",
         "--steps", "2", "--warmup", "1", "--model", "gpt2-tiny",
         "--reshard", "--ac", "--micro-batch", "1", "--seq-len", "128"],
        capture_output=True, text=True, timeout=900, env=env, cwd=repo)
    assert r.returncode == 0, r.stderr[-3000:]
    import json
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 8 and out["steps"] == 2
