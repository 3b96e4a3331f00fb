"""HSDP (replicate x shard) engine test + distributed dataloader determinism
(reference model: tests/fsdp2_parallelization/test_full_and_hybrid_sharding.py
and tests/dataloader/distributed/)."""

import numpy as np
import pytest
import torch

from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from tests.conftest import find_free_port
from tests.utils_dist import run_distributed

VOCAB = 128


def tiny_cfg():
    return GPT2LLMConfig(vocab_size=VOCAB, n_layer=2, n_head_q=4, n_head_kv=2,
                         n_embd=64, ffn_hidden=256, sequence_length=32, seed=7)


def make_batch(seed, batch=4, seqlen=16):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, VOCAB, (batch, seqlen + 1), generator=g)
    return ids[:, :-1], ids[:, 1:]


def _hsdp_worker(rank, world):
    """world 2 as (dp_replicate=2, dp_shard=1): each rank holds full params;
    grads all-reduced over the replicate group. Must match single-process
    full-batch training."""
    import torch.distributed as dist

    from modalities_amd.optimizers.optimizer_factory import get_adam_w
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    from modalities_amd.parallel.mesh import DeviceMesh, ParallelismDegrees
    mesh = DeviceMesh(world, rank, dp_replicate=2, dp_shard=1)
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    shard_dim = mesh.dims[ParallelismDegrees.DP_SHARD]
    rep_dim = mesh.dims[ParallelismDegrees.DP_REPLICATE]
    sharded = XGMIShardedModel.from_transformer(
        model, torch.device("cpu"), process_group=shard_dim.group,
        rank=shard_dim.rank, world_size=shard_dim.size,
        param_dtype=torch.float32, replicate_group=rep_dim.group)
    opt = get_adam_w(sharded, lr=1e-3, weight_decay=0.0)
    losses = []
    for i in range(3):
        x, y = make_batch(100 + i)
        n = x.shape[0] // world
        xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
        out = sharded({"input_ids": xs})["logits"]
        loss = torch.nn.functional.cross_entropy(out.view(-1, VOCAB).float(),
                                                 ys.reshape(-1))
        loss.backward()
        sharded.backward_epilogue()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    return losses


def test_hybrid_sharding_matches_full_batch():
    from tests.test_fsdp_engine import eager_reference_losses
    ref = eager_reference_losses(steps=3)
    results = run_distributed(_hsdp_worker, world_size=2, port=find_free_port())
    merged = [0.5 * (a + b) for a, b in zip(results[0], results[1])]
    assert ref == pytest.approx(merged, rel=1e-4)


# ---- distributed dataloader determinism ------------------------------------

def _loader_worker(rank, world, pbin_path, skip):
    from torch.utils.data import BatchSampler

    from modalities_amd.dataloader.dataloader import (GPT2LLMCollateFn,
                                                      LLMDataLoader)
    from modalities_amd.dataloader.dataset import PackedMemMapDatasetContinuous
    from modalities_amd.dataloader.samplers import ResumableDistributedSampler
    ds = PackedMemMapDatasetContinuous(pbin_path, "input_ids", block_size=17)
    sampler = ResumableDistributedSampler(ds, rank=rank, num_replicas=world,
                                          shuffle=True, seed=42, drop_last=True,
                                          skip_num_global_samples=skip)
    bs = BatchSampler(sampler, batch_size=2, drop_last=True)
    loader = LLMDataLoader("train", ds, bs,
                           collate_fn=GPT2LLMCollateFn("input_ids", "target_ids"))
    return [batch.samples["input_ids"].numpy() for batch in loader]


@pytest.fixture
def big_pbin(tmp_path):
    from modalities_amd.dataloader.packed_data import write_pbin
    rng = np.random.default_rng(3)
    docs = [rng.integers(0, 200, size=300, dtype=np.uint8) for _ in range(8)]
    p = tmp_path / "big.pbin"
    write_pbin(p, docs, 1)
    return p


def test_distributed_sampler_partition_and_determinism(big_pbin):
    r1 = run_distributed(_loader_worker, world_size=2, port=find_free_port(),
                         args=(str(big_pbin), 0))
    r2 = run_distributed(_loader_worker, world_size=2, port=find_free_port(),
                         args=(str(big_pbin), 0))
    # deterministic across runs
    for rank in (0, 1):
        assert len(r1[rank]) == len(r2[rank]) > 0
        for a, b in zip(r1[rank], r2[rank]):
            np.testing.assert_array_equal(a, b)
    # ranks see disjoint samples
    seen0 = {a.tobytes() for batch in r1[0] for a in batch}
    seen1 = {a.tobytes() for batch in r1[1] for a in batch}
    assert not (seen0 & seen1)


def test_skip_samples_resumes_midstream(big_pbin):
    full = run_distributed(_loader_worker, world_size=2, port=find_free_port(),
                           args=(str(big_pbin), 0))
    skipped = run_distributed(_loader_worker, world_size=2, port=find_free_port(),
                              args=(str(big_pbin), 4))  # skip 2 global batches
    for rank in (0, 1):
        assert len(skipped[rank]) == len(full[rank]) - 1
        for a, b in zip(skipped[rank], full[rank][1:]):
            np.testing.assert_array_equal(a, b)
