"""Tests for the XGMI sharded data-parallel engine.

Correctness model: sharded training at any world size must match plain eager
fp32 single-process training (same seeds, same data) to tight tolerance,
since at param_dtype=fp32 the engine is numerically a re-bucketed DDP+ZeRO3.
"""

import pytest
import torch

from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.optimizers.optimizer_factory import get_adam_w
from modalities_amd.parallel.fsdp import XGMIShardedModel
from tests.utils_dist import run_distributed
from tests.conftest import find_free_port

VOCAB = 128


def tiny_cfg(**kw):
    d = dict(vocab_size=VOCAB, n_layer=2, n_head_q=4, n_head_kv=2, n_embd=64,
             ffn_hidden=256, sequence_length=32, seed=7)
    d.update(kw)
    return GPT2LLMConfig(**d)


def make_batch(seed, batch=4, seqlen=16):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, VOCAB, (batch, seqlen + 1), generator=g)
    return ids[:, :-1], ids[:, 1:]


def eager_reference_losses(steps=4, lr=1e-3):
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    opt = torch.optim.AdamW(model.parameters(), lr=lr, betas=(0.9, 0.95),
                            eps=1e-8, weight_decay=0.0)
    losses = []
    for i in range(steps):
        x, y = make_batch(100 + i)
        out = model({"input_ids": x})
        loss = torch.nn.functional.cross_entropy(
            out["logits"].view(-1, VOCAB).float(), y.reshape(-1))
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    return losses


def sharded_losses_one_proc(steps=4, lr=1e-3, blocks_per_unit=1, reshard=False):
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    sharded = XGMIShardedModel.from_transformer(
        model, torch.device("cpu"), blocks_per_unit=blocks_per_unit,
        param_dtype=torch.float32, reshard_after_forward=reshard)
    opt = get_adam_w(sharded, lr=lr, weight_decay=0.0)
    losses = []
    for i in range(steps):
        x, y = make_batch(100 + i)
        out = sharded({"input_ids": x})
        loss = torch.nn.functional.cross_entropy(
            out["logits"].view(-1, VOCAB).float(), y.reshape(-1))
        loss.backward()
        sharded.backward_epilogue()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    return losses


def test_sharded_matches_eager_world1():
    ref = eager_reference_losses()
    got = sharded_losses_one_proc()
    assert ref == pytest.approx(got, rel=1e-4), (ref, got)


def test_sharded_matches_eager_reshard_after_forward():
    ref = eager_reference_losses()
    got = sharded_losses_one_proc(reshard=True)
    assert ref == pytest.approx(got, rel=1e-4), (ref, got)


def test_sharded_matches_eager_grouped_blocks():
    ref = eager_reference_losses()
    got = sharded_losses_one_proc(blocks_per_unit=2)
    assert ref == pytest.approx(got, rel=1e-4), (ref, got)


def _dp2_worker(rank, world, steps, lr):
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    sharded = XGMIShardedModel.from_transformer(
        model, torch.device("cpu"), param_dtype=torch.float32)
    opt = get_adam_w(sharded, lr=lr, weight_decay=0.0)
    losses = []
    for i in range(steps):
        x, y = make_batch(100 + i)  # same global batch, split across ranks
        n = x.shape[0] // world
        xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
        out = sharded({"input_ids": xs})
        loss = torch.nn.functional.cross_entropy(
            out["logits"].view(-1, VOCAB).float(), ys.reshape(-1))
        loss.backward()
        sharded.backward_epilogue()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    return losses


def test_dp2_matches_eager_global_batch():
    """2-rank sharded training on split halves of the same global batch must
    match single-process eager training on the full batch."""
    ref = eager_reference_losses(steps=3)
    results = run_distributed(_dp2_worker, world_size=2, port=find_free_port(),
                              args=(3, 1e-3))
    # mean of the two ranks' micro losses == global loss
    merged = [0.5 * (a + b) for a, b in zip(results[0], results[1])]
    assert ref == pytest.approx(merged, rel=1e-4), (ref, merged)


def test_grad_accumulation_matches_full_batch():
    """Accumulated microbatch grads == full-batch grads (compared on the
    flat grad shards, before the optimizer amplifies rounding noise)."""
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    ref_sharded = XGMIShardedModel.from_transformer(model, torch.device("cpu"),
                                                    param_dtype=torch.float32)
    x, y = make_batch(55, batch=4)
    out = ref_sharded({"input_ids": x})
    loss = torch.nn.functional.cross_entropy(out["logits"].view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()
    ref_sharded.backward_epilogue()
    ref_grads = [u.grad_shard.clone() for u in ref_sharded.units]

    torch.manual_seed(0)
    model2 = GPT2LLM(tiny_cfg())
    acc_sharded = XGMIShardedModel.from_transformer(model2, torch.device("cpu"),
                                                    param_dtype=torch.float32)
    for mb in range(2):
        xs, ys = x[mb * 2:(mb + 1) * 2], y[mb * 2:(mb + 1) * 2]
        out = acc_sharded({"input_ids": xs})
        loss = torch.nn.functional.cross_entropy(
            out["logits"].view(-1, VOCAB).float(), ys.reshape(-1)) / 2
        loss.backward()
        acc_sharded.backward_epilogue()
    for rg, u in zip(ref_grads, acc_sharded.units):
        torch.testing.assert_close(rg, u.grad_shard, rtol=1e-4, atol=1e-5)


def test_clip_grad_norm_matches_torch():
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    x, y = make_batch(9)
    out = model({"input_ids": x})
    loss = torch.nn.functional.cross_entropy(out["logits"].view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()
    ref_norm = torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)

    torch.manual_seed(0)
    model2 = GPT2LLM(tiny_cfg())
    sharded = XGMIShardedModel.from_transformer(model2, torch.device("cpu"),
                                                param_dtype=torch.float32)
    out = sharded({"input_ids": x})
    loss = torch.nn.functional.cross_entropy(out["logits"].view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()
    sharded.backward_epilogue()
    got_norm = sharded.clip_grad_norm_(1.0)
    assert ref_norm.item() == pytest.approx(got_norm.item(), rel=1e-5)


def test_weight_tying_sharded():
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg(use_weight_tying=True))
    sharded = XGMIShardedModel.from_transformer(model, torch.device("cpu"),
                                                param_dtype=torch.float32)
    # tied param appears once in the flat units
    all_params = sum(len(u.param_infos) for u in sharded.units)
    n_unique = len({id(p) for p in model.parameters()})
    assert all_params == n_unique
    x, y = make_batch(3)
    out = sharded({"input_ids": x})
    loss = torch.nn.functional.cross_entropy(out["logits"].view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()
    sharded.backward_epilogue()
    opt = get_adam_w(sharded, lr=1e-3)
    opt.step()
