"""TP/SP correctness (reference model:
tests/fsdp2_parallelization/test_tensor_parallelism.py:42-115): sharded
weight shapes, TP2 forward == unsharded forward, TP2 backward grads match,
SP variant equivalence. Runs on CPU via gloo world 2."""

import pytest
import torch

from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from tests.conftest import find_free_port
from tests.utils_dist import run_distributed

VOCAB = 128


def tiny_cfg(**kw):
    d = dict(vocab_size=VOCAB, n_layer=2, n_head_q=4, n_head_kv=2, n_embd=64,
             ffn_hidden=256, sequence_length=32, seed=11, dropout=0.0)
    d.update(kw)
    return GPT2LLMConfig(**d)


def make_batch(seed=5, batch=2, seqlen=32):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, VOCAB, (batch, seqlen + 1), generator=g)
    return ids[:, :-1], ids[:, 1:]


def reference_fwd_bwd():
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    x, y = make_batch()
    out = model({"input_ids": x})["logits"]
    loss = torch.nn.functional.cross_entropy(out.view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()
    grads = {n: p.grad.clone() for n, p in model.named_parameters()}
    return out.detach(), loss.detach(), grads


def _tp_worker(rank, world, sequence_parallel):
    import torch.distributed as dist

    from modalities_amd.parallel.tp import get_gpt2_tensor_parallelized_model
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    model = get_gpt2_tensor_parallelized_model(
        model, group=dist.group.WORLD, tp_rank=rank, tp_size=world,
        sequence_parallel=sequence_parallel)
    x, y = make_batch()
    out = model({"input_ids": x})["logits"]
    loss = torch.nn.functional.cross_entropy(out.view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()
    # representative sharded + replicated grads
    g_q = model.blocks[0].attn.q_attn.weight.grad.clone()
    g_wte = model.wte.weight.grad.clone()
    shapes = {
        "q_attn": tuple(model.blocks[0].attn.q_attn.weight.shape),
        "c_proj": tuple(model.blocks[0].attn.c_proj.weight.shape),
        "W": tuple(model.blocks[0].mlp.W_weight.shape),
        "W_2": tuple(model.blocks[0].mlp.W_2.weight.shape),
    }
    return (out.detach().numpy(), loss.item(), shapes, g_q.numpy(), g_wte.numpy())


@pytest.mark.parametrize("sequence_parallel", [False, True],
                         ids=["tp", "tp_sp"])
def test_tp2_matches_unsharded(sequence_parallel):
    ref_out, ref_loss, ref_grads = reference_fwd_bwd()
    results = run_distributed(_tp_worker, world_size=2, port=find_free_port(),
                              args=(sequence_parallel,))
    for rank, (out, loss, shapes, g_q, g_wte) in results.items():
        out = torch.from_numpy(out)
        g_q = torch.from_numpy(g_q)
        g_wte = torch.from_numpy(g_wte)
        loss = torch.tensor(loss)
        # sharded shapes: q/k/v + up-proj split on dim 0; down-proj on dim 1
        assert shapes["q_attn"] == (32, 64)
        assert shapes["c_proj"] == (64, 32)
        assert shapes["W"][0] == shapes["W_2"][1]
        torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(loss, ref_loss, rtol=1e-5, atol=1e-6)
        # sharded grad equals the matching slice of the reference grad
        ref_gq = ref_grads["blocks.0.attn.q_attn.weight"]
        half = ref_gq.shape[0] // 2
        torch.testing.assert_close(g_q, ref_gq[rank * half:(rank + 1) * half],
                                   rtol=1e-4, atol=1e-5)
        # replicated param grads see the all-reduced (full) gradient
        torch.testing.assert_close(g_wte, ref_grads["wte.weight"],
                                   rtol=1e-4, atol=1e-5)


def _tp_dp_worker(rank, world):
    """TP=2 composed with the XGMI sharding engine over a 1-rank dp group
    (mesh world 2 = tp 2 x dp 1): end-to-end trainability."""
    import torch.distributed as dist

    from modalities_amd.optimizers.optimizer_factory import get_adam_w
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    from modalities_amd.parallel.tp import get_gpt2_tensor_parallelized_model
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    model = get_gpt2_tensor_parallelized_model(
        model, group=dist.group.WORLD, tp_rank=rank, tp_size=world)
    sharded = XGMIShardedModel.from_transformer(
        model, torch.device("cpu"), param_dtype=torch.float32,
        rank=0, world_size=1)
    opt = get_adam_w(sharded, lr=1e-3, weight_decay=0.0)
    losses = []
    for i in range(3):
        x, y = make_batch(100 + i)
        out = sharded({"input_ids": x})["logits"]
        loss = torch.nn.functional.cross_entropy(out.view(-1, VOCAB).float(),
                                                 y.reshape(-1))
        loss.backward()
        sharded.backward_epilogue()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    return losses


def test_tp2_with_sharding_engine_trains():
    results = run_distributed(_tp_dp_worker, world_size=2,
                              port=find_free_port())
    # both TP ranks see identical losses, and loss decreases
    assert results[0] == pytest.approx(results[1], rel=1e-5)
    assert results[0][-1] < results[0][0]


def _tp2_dp2_worker(rank, world):
    """TP=2 x DP_shard=2 over a world-4 DeviceMesh: per-step mean losses
    must match the single-process full-batch run (grads are MEAN over the
    dp group; both TP ranks of a dp replica see the same rows)."""
    import torch.distributed as dist

    from modalities_amd.optimizers.optimizer_factory import get_adam_w
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    from modalities_amd.parallel.mesh import DeviceMesh, ParallelismDegrees
    from modalities_amd.parallel.tp import get_gpt2_tensor_parallelized_model

    mesh = DeviceMesh(world_size=world, rank=rank, tp=2, dp_shard=2)
    tp = mesh.dims[ParallelismDegrees.TP]
    dp = mesh.dims[ParallelismDegrees.DP_SHARD]
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    model = get_gpt2_tensor_parallelized_model(
        model, group=tp.group, tp_rank=tp.rank, tp_size=tp.size)
    sharded = XGMIShardedModel.from_transformer(
        model, torch.device("cpu"), process_group=dp.group,
        param_dtype=torch.float32, rank=dp.rank, world_size=dp.size)
    opt = get_adam_w(sharded, lr=1e-3, weight_decay=0.0)
    losses = []
    for i in range(2):
        x, y = make_batch(200 + i, batch=4)
        rows = slice(dp.rank * 2, dp.rank * 2 + 2)
        out = sharded({"input_ids": x[rows]})["logits"]
        loss = torch.nn.functional.cross_entropy(
            out.reshape(-1, VOCAB).float(), y[rows].reshape(-1))
        loss.backward()
        sharded.backward_epilogue()
        opt.step()
        opt.zero_grad()
        g = loss.detach().clone()
        dist.all_reduce(g)  # mean over the 4 ranks == global-batch mean
        losses.append(g.item() / world)
    return losses


def test_tp2_dp2_matches_single_process():
    torch.manual_seed(0)
    ref_model = GPT2LLM(tiny_cfg())
    ref_opt = torch.optim.AdamW(ref_model.parameters(), lr=1e-3, betas=(0.9, 0.95),
                                eps=1e-8, weight_decay=0.0)
    ref_losses = []
    for i in range(2):
        x, y = make_batch(200 + i, batch=4)
        out = ref_model({"input_ids": x})["logits"]
        loss = torch.nn.functional.cross_entropy(out.reshape(-1, VOCAB).float(),
                                                 y.reshape(-1))
        loss.backward()
        ref_opt.step()
        ref_opt.zero_grad()
        ref_losses.append(loss.item())
    results = run_distributed(_tp2_dp2_worker, world_size=4,
                              port=find_free_port())
    for r in range(4):
        assert results[r] == pytest.approx(ref_losses, rel=2e-4), \
            (results[r], ref_losses)


# ---------------------------------------------------------------------------
# Vocab sharding: embedding rows + lm_head columns sharded
# over the tp group, loss via gather-free vocab-parallel cross-entropy
# (reference shards both but gathers logits, model_factory.py:657-766).

def _tp_vocab_worker(rank, world, fused_qkv, tying):
    import torch.distributed as dist

    from modalities_amd.parallel.tp import (get_gpt2_tensor_parallelized_model,
                                            vocab_parallel_cross_entropy)
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg(fused_qkv=fused_qkv, use_weight_tying=tying))
    model = get_gpt2_tensor_parallelized_model(
        model, group=dist.group.WORLD, tp_rank=rank, tp_size=world,
        shard_vocab=True)
    assert model.lm_head.weight.shape[0] == VOCAB // world
    assert model.wte.weight.shape[0] == VOCAB // world
    if tying:
        assert model.lm_head.weight is model.wte.weight
    x, y = make_batch()
    out = model({"input_ids": x})["logits"]      # [B, T, V/world]
    assert out.shape[-1] == VOCAB // world
    info = model._tp_vocab_info
    loss = vocab_parallel_cross_entropy(out.reshape(-1, out.shape[-1]),
                                        y.reshape(-1), info[0], info[1],
                                        info[2], info[3])
    loss.backward()
    g_wte = model.wte.weight.grad.clone()
    g_qw = (model.blocks[0].attn.qkv_attn.weight.grad.clone() if fused_qkv
            else model.blocks[0].attn.q_attn.weight.grad.clone())
    return (out.detach().numpy(), loss.item(), g_wte.numpy(), g_qw.numpy())


@pytest.mark.parametrize("fused_qkv,tying", [(False, False), (True, False),
                                             (False, True)])
def test_tp2_vocab_sharded_matches_unsharded(fused_qkv, tying):
    torch.manual_seed(0)
    ref_model = GPT2LLM(tiny_cfg(fused_qkv=fused_qkv, use_weight_tying=tying))
    x, y = make_batch()
    out = ref_model({"input_ids": x})["logits"]
    loss = torch.nn.functional.cross_entropy(out.view(-1, VOCAB).float(),
                                             y.reshape(-1))
    loss.backward()

    results = run_distributed(_tp_vocab_worker, world_size=2,
                              port=find_free_port(), args=(fused_qkv, tying))
    # logits: rank r holds vocab slice r
    for r in range(2):
        got = torch.from_numpy(results[r][0])
        ref_slice = out.detach()[..., r * VOCAB // 2:(r + 1) * VOCAB // 2]
        torch.testing.assert_close(got, ref_slice, rtol=2e-4, atol=2e-4)
        assert results[r][1] == pytest.approx(loss.item(), rel=1e-5)
    # wte grad: rank r holds rows slice r (colwise-replicated input grads
    # are identical across ranks)
    name = "wte.weight"
    ref_wte = dict(ref_model.named_parameters())[name].grad
    for r in range(2):
        got = torch.from_numpy(results[r][2])
        torch.testing.assert_close(got, ref_wte[r * VOCAB // 2:
                                                (r + 1) * VOCAB // 2],
                                   rtol=2e-4, atol=2e-4)
    # q (or fused qkv) projection grad: column-parallel rows per rank
    if fused_qkv:
        wname = "blocks.0.attn.qkv_attn.weight"
        ref_g = dict(ref_model.named_parameters())[wname].grad
        C, KV = 64, 32 * 2 // 2 * 2  # n_embd, kv_dim (n_head_kv*hd = 32)
        KV = 32
        for r in range(2):
            got = torch.from_numpy(results[r][3])
            exp = torch.cat([
                ref_g[r * C // 2:(r + 1) * C // 2],
                ref_g[C + r * KV // 2:C + (r + 1) * KV // 2],
                ref_g[C + KV + r * KV // 2:C + KV + (r + 1) * KV // 2]])
            torch.testing.assert_close(got, exp, rtol=2e-4, atol=2e-4)
    else:
        wname = "blocks.0.attn.q_attn.weight"
        ref_g = dict(ref_model.named_parameters())[wname].grad
        per = ref_g.shape[0] // 2
        for r in range(2):
            got = torch.from_numpy(results[r][3])
            torch.testing.assert_close(got, ref_g[r * per:(r + 1) * per],
                                       rtol=2e-4, atol=2e-4)
