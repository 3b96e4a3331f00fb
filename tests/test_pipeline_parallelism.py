"""PP correctness (reference model: tests/fsdp2_parallelization/
pipeline_parallelism/test_pp_fwd_bwd_pass.py:35-60): stage assignment,
pp=2 fwd/bwd loss equivalence vs single-process for GPipe and 1F1B,
gradient equivalence on stage-local params. CPU/gloo."""

import pytest
import torch

from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.parallel.pp import (balanced_stage_assignment,
                                        split_model_into_stages)
from tests.conftest import find_free_port
from tests.utils_dist import run_distributed

VOCAB = 128
N_MB = 4


def tiny_cfg():
    return GPT2LLMConfig(vocab_size=VOCAB, n_layer=4, n_head_q=4, n_head_kv=2,
                         n_embd=64, ffn_hidden=256, sequence_length=16,
                         seed=3, dropout=0.0)


def make_batch(seed=5, batch=8, seqlen=16):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, VOCAB, (batch, seqlen + 1), generator=g)
    return ids[:, :-1], ids[:, 1:]


def test_balanced_stage_assignment():
    assert balanced_stage_assignment(8, 2) == [4, 4]
    assert sum(balanced_stage_assignment(32, 4, 1.0, 1.0)) == 32
    counts = balanced_stage_assignment(9, 3, input_weight=2.0, output_weight=2.0)
    assert sum(counts) == 9
    assert counts[0] <= counts[1]  # first stage lighter (carries embedding)
    with pytest.raises(ValueError):
        balanced_stage_assignment(2, 3)


def test_stage_split_covers_model():
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages(model, 2)
    assert stages[0].is_first and not stages[0].is_last
    assert stages[1].is_last
    assert len(stages[0].blocks) + len(stages[1].blocks) == 4
    assert hasattr(stages[0], "wte") and not hasattr(stages[1], "wte")
    assert hasattr(stages[1], "lm_head")
    # stage-chained forward == full model forward
    x, _ = make_batch()
    with torch.no_grad():
        full = model({"input_ids": x})["logits"]
        chained = stages[1](stages[0](x))
    torch.testing.assert_close(full, chained)


def reference_loss_and_grads():
    """Single-process run with identical microbatching semantics."""
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    x, y = make_batch()
    losses = []
    for mb_x, mb_y in zip(x.chunk(N_MB), y.chunk(N_MB)):
        out = model({"input_ids": mb_x})["logits"]
        loss = torch.nn.functional.cross_entropy(
            out.reshape(-1, VOCAB).float(), mb_y.reshape(-1))
        (loss / N_MB).backward()
        losses.append(loss.item())
    grads = {n: p.grad.clone().numpy() for n, p in model.named_parameters()}
    return losses, grads


def _pp_worker(rank, world, variant):
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import get_pipeline_schedule
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages(model, world)
    stage = stages[rank]
    sched = get_pipeline_schedule(
        variant, stage=stage, stage_idx=rank, num_stages=world,
        n_microbatches=N_MB, group=dist.group.WORLD)
    x, y = make_batch()
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    losses = sched.step(x, y, loss_fn)
    grads = {n: p.grad.clone().numpy() for n, p in stage.named_parameters()
             if p.grad is not None}
    return [l.item() for l in losses], grads, rank


@pytest.mark.parametrize("variant", ["gpipe", "1f1b"])
def test_pp2_matches_single_process(variant):
    ref_losses, ref_grads = reference_loss_and_grads()
    results = run_distributed(_pp_worker, world_size=2, port=find_free_port(),
                              args=(variant,))
    # last stage observed the per-microbatch losses
    pp_losses = results[1][0]
    assert pp_losses == pytest.approx(ref_losses, rel=1e-5)
    # stage-0 params: wte + blocks 0..k map onto reference names directly
    g0 = results[0][1]
    torch.testing.assert_close(torch.from_numpy(g0["wte.weight"]),
                               torch.from_numpy(ref_grads["wte.weight"]),
                               rtol=1e-4, atol=1e-6)
    # stage-1 params: blocks are renumbered from 0; head/norm keep names
    g1 = results[1][1]
    torch.testing.assert_close(torch.from_numpy(g1["lm_head.weight"]),
                               torch.from_numpy(ref_grads["lm_head.weight"]),
                               rtol=1e-4, atol=1e-6)
    n0 = len(g0) - 1  # blocks on stage 0
    n_blocks0 = len([k for k in g0 if k.startswith("blocks.")])
    n_stage0_blocks = len({k.split(".")[1] for k in g0 if k.startswith("blocks.")})
    first_stage1_block = n_stage0_blocks
    torch.testing.assert_close(
        torch.from_numpy(g1["blocks.0.attn.q_attn.weight"]),
        torch.from_numpy(
            ref_grads[f"blocks.{first_stage1_block}.attn.q_attn.weight"]),
        rtol=1e-4, atol=1e-6)


def _pp_trainer_worker(rank, world):
    """PP=2 through the full Trainer loop (PP-schedule dispatch, loss
    broadcast so every rank logs the true loss)."""
    import torch.distributed as dist

    from modalities_amd.batch import DatasetBatch
    from modalities_amd.logging_broker.broker import (MessageBroker,
                                                      MessagePublisher)
    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import get_pipeline_schedule
    from modalities_amd.training.trainer import Trainer

    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stage = split_model_into_stages(model, world)[rank]
    sched = get_pipeline_schedule(
        "1f1b", stage=stage, stage_idx=rank, num_stages=world,
        n_microbatches=N_MB, group=dist.group.WORLD)
    opt = torch.optim.AdamW(stage.parameters(), lr=1e-3)
    broker = MessageBroker()
    pub = MessagePublisher(broker, global_rank=rank, local_rank=rank)
    trainer = Trainer(global_rank=rank, progress_publisher=pub,
                      evaluation_result_publisher=pub, gradient_acc_steps=1,
                      global_num_tokens_per_train_step=8 * 16,
                      num_seen_train_steps=0, global_num_seen_tokens=0,
                      num_target_steps=3, num_target_tokens=3 * 8 * 16,
                      pp_schedule=sched)
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    losses = []
    x, y = make_batch(100)  # same batch each step -> loss must decrease
    for i in range(3):
        batch = DatasetBatch(samples={"input_ids": x}, targets={"target_ids": y})
        done, loss, _ = trainer._train_batch(batch, stage, opt, None, loss_fn, i)
        assert done
        losses.append(float(loss))
    return losses


def test_pp_through_trainer():
    results = run_distributed(_pp_trainer_worker, world_size=2,
                              port=find_free_port())
    # both stages log identical (broadcast) losses; training decreases them
    assert results[0] == pytest.approx(results[1], rel=1e-6)
    assert results[0][-1] < results[0][0]


def _pp2_dp2_worker(rank, world):
    """PP=2 x DP=2 over a world-4 mesh: each dp replica pipelines its half
    of the global batch through its 2 stages; stage grads are then averaged
    over the dp group (plain all-reduce — stage params are replicated
    across dp). Returns dp-averaged wte/lm_head grads for comparison."""
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.mesh import DeviceMesh, ParallelismDegrees
    from modalities_amd.parallel.pp import get_pipeline_schedule

    mesh = DeviceMesh(world_size=world, rank=rank, pp=2, dp_shard=2)
    pp = mesh.dims[ParallelismDegrees.PP]
    dp = mesh.dims[ParallelismDegrees.DP_SHARD]
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages(model, pp.size)
    stage = stages[pp.rank]
    sched = get_pipeline_schedule(
        "gpipe", stage=stage, stage_idx=pp.rank, num_stages=pp.size,
        n_microbatches=2, group=pp.group)
    x, y = make_batch(batch=4)
    rows = slice(dp.rank * 2, dp.rank * 2 + 2)
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    sched.step(x[rows], y[rows], loss_fn)
    out = {}
    for n, p in stage.named_parameters():
        if p.grad is not None and n in ("wte.weight", "lm_head.weight"):
            g = p.grad.clone()
            dist.all_reduce(g, group=dp.group)
            out[n] = (g / dp.size).numpy()
    return out


def test_pp2_dp2_matches_single_process():
    torch.manual_seed(0)
    ref_model = GPT2LLM(tiny_cfg())
    x, y = make_batch(batch=4)
    for mb_x, mb_y in zip(x.chunk(2), y.chunk(2)):
        out = ref_model({"input_ids": mb_x})["logits"]
        loss = torch.nn.functional.cross_entropy(
            out.reshape(-1, VOCAB).float(), mb_y.reshape(-1))
        (loss / 2).backward()
    ref = {n: p.grad for n, p in ref_model.named_parameters()}
    results = run_distributed(_pp2_dp2_worker, world_size=4,
                              port=find_free_port())
    # dp chunks of 2 rows == reference microbatches of 2 rows, so the
    # dp-mean of per-replica grads equals the reference accumulated mean
    found = 0
    for r in range(4):
        for name, g in results[r].items():
            torch.testing.assert_close(torch.from_numpy(g), ref[name],
                                       rtol=1e-4, atol=1e-6)
            found += 1
    assert found >= 2  # wte on stage-0 ranks, lm_head on stage-1 ranks


# ---------------------------------------------------------------------------
# Interleaved 1F1B: pp=2 with 2 virtual chunks per rank
# (4 global stages), vs single-process reference.

def _pp_interleaved_worker(rank, world):
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import (ScheduleInterleaved1F1B,
                                            interleaved_stage_ids)
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    v = 2
    stages_all = split_model_into_stages(model, world * v)
    my = [stages_all[i] for i in interleaved_stage_ids(rank, world, v)]
    sched = ScheduleInterleaved1F1B(my, pp_rank=rank, pp_size=world,
                                    n_microbatches=N_MB,
                                    group=dist.group.WORLD)
    x, y = make_batch()
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    losses = sched.step(x, y, loss_fn)
    mean = sched.broadcast_mean_loss(losses)
    grads = {}
    for c, st in enumerate(my):
        for n, p in st.named_parameters():
            if p.grad is not None:
                grads[f"chunk{c}.{n}"] = p.grad.clone().numpy()
    # eval_step: forward-only losses must equal the training-step losses
    # (same params, no optimizer step in between)
    eval_losses = sched.eval_step(x, y, loss_fn)
    return ([l.item() for l in losses], grads, mean.item(),
            [l.item() for l in eval_losses])


def test_pp2_interleaved_matches_single_process():
    ref_losses, ref_grads = reference_loss_and_grads()
    results = run_distributed(_pp_interleaved_worker, world_size=2,
                              port=find_free_port())
    # rank 1 holds the last stage (chunk 1 = global stage 3)
    pp_losses = results[1][0]
    assert sorted(pp_losses) == pytest.approx(sorted(ref_losses), rel=1e-5)
    assert results[0][2] == pytest.approx(sum(ref_losses) / N_MB, rel=1e-5)
    assert results[1][3] == pytest.approx(pp_losses, rel=1e-5)  # eval_step
    # 4 blocks over 4 stages: rank0 chunk0 = {wte, block0}, chunk1 = block2;
    # rank1 chunk0 = block1, chunk1 = {block3, head}
    g0, g1 = results[0][1], results[1][1]
    torch.testing.assert_close(torch.from_numpy(g0["chunk0.wte.weight"]),
                               torch.from_numpy(ref_grads["wte.weight"]),
                               rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(
        torch.from_numpy(g0["chunk1.blocks.0.attn.q_attn.weight"]),
        torch.from_numpy(ref_grads["blocks.2.attn.q_attn.weight"]),
        rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(
        torch.from_numpy(g1["chunk0.blocks.0.attn.q_attn.weight"]),
        torch.from_numpy(ref_grads["blocks.1.attn.q_attn.weight"]),
        rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(torch.from_numpy(g1["chunk1.lm_head.weight"]),
                               torch.from_numpy(ref_grads["lm_head.weight"]),
                               rtol=1e-4, atol=1e-6)


def _pp_eval_worker(rank, world, variant):
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import get_pipeline_schedule
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages(model, world)
    sched = get_pipeline_schedule(
        variant, stage=stages[rank], stage_idx=rank, num_stages=world,
        n_microbatches=N_MB, group=dist.group.WORLD)
    x, y = make_batch()
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    losses = sched.eval_step(x, y, loss_fn)
    return sched.broadcast_mean_loss(losses).item()


@pytest.mark.parametrize("variant", ["gpipe", "1f1b"])
def test_pp2_eval_step_forward_only(variant):
    ref_losses, _ = reference_loss_and_grads()
    results = run_distributed(_pp_eval_worker, world_size=2,
                              port=find_free_port(), args=(variant,))
    for r in range(2):
        assert results[r] == pytest.approx(sum(ref_losses) / N_MB, rel=1e-5)


# ---------------------------------------------------------------------------
# TP2 x PP2 x DP2 world-8 composition (VERDICT r1 #2b: stress the engine's
# collective ordering at the full 8-GPU-node world size on CPU/gloo).

def _tp2_pp2_dp2_worker(rank, world):
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.mesh import DeviceMesh, ParallelismDegrees
    from modalities_amd.parallel.pp import get_pipeline_schedule
    from modalities_amd.parallel.tp import get_gpt2_tensor_parallelized_model

    mesh = DeviceMesh(world_size=world, rank=rank, pp=2, dp_shard=2, tp=2)
    pp = mesh.dims[ParallelismDegrees.PP]
    dp = mesh.dims[ParallelismDegrees.DP_SHARD]
    tp = mesh.dims[ParallelismDegrees.TP]
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    # TP-shard IN PLACE first (patches block forwards), then PP-split: the
    # stages reference the already-sharded blocks (reference order:
    # model_factory TP plan -> PipelineFactory)
    model = get_gpt2_tensor_parallelized_model(
        model, group=tp.group, tp_rank=tp.rank, tp_size=tp.size)
    stages = split_model_into_stages(model, pp.size)
    stage = stages[pp.rank]
    sched = get_pipeline_schedule(
        "1f1b", stage=stage, stage_idx=pp.rank, num_stages=pp.size,
        n_microbatches=2, group=pp.group)
    x, y = make_batch(batch=4)
    rows = slice(dp.rank * 2, dp.rank * 2 + 2)
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    losses = sched.step(x[rows], y[rows], loss_fn)
    mean = sched.broadcast_mean_loss(losses)
    dist.all_reduce(mean, group=dp.group)  # global mean across dp replicas
    mean /= dp.size

    out = {}
    for n, p in stage.named_parameters():
        if p.grad is None:
            continue
        if n in ("wte.weight", "lm_head.weight",
                 "blocks.0.attn.q_attn.weight"):
            g = p.grad.clone()
            dist.all_reduce(g, group=dp.group)  # dp-mean (replicated grads)
            out[n] = (g / dp.size).numpy()
    return out, mean.item(), tp.rank, pp.rank


def test_tp2_pp2_dp2_world8_matches_single_process():
    torch.manual_seed(0)
    ref_model = GPT2LLM(tiny_cfg())
    x, y = make_batch(batch=4)
    losses = []
    for mb_x, mb_y in zip(x.chunk(2), y.chunk(2)):
        out = ref_model({"input_ids": mb_x})["logits"]
        loss = torch.nn.functional.cross_entropy(
            out.reshape(-1, VOCAB).float(), mb_y.reshape(-1))
        (loss / 2).backward()
        losses.append(loss.item())
    ref = {n: p.grad for n, p in ref_model.named_parameters()}
    ref_mean = sum(losses) / 2

    results = run_distributed(_tp2_pp2_dp2_worker, world_size=8,
                              port=find_free_port(), timeout_s=300)
    checked = 0
    for r in range(8):
        grads, mean, tp_rank, pp_rank = results[r]
        assert mean == pytest.approx(ref_mean, rel=1e-5)
        for name, g in grads.items():
            got = torch.from_numpy(g)
            if name == "blocks.0.attn.q_attn.weight":
                # column-parallel: this tp rank holds rows slice tp_rank;
                # stage-1 blocks renumber, so only compare on stage 0
                if pp_rank != 0:
                    continue
                per = ref[name].shape[0] // 2
                exp = ref[name][tp_rank * per:(tp_rank + 1) * per]
            else:
                exp = ref[name]
            torch.testing.assert_close(got, exp, rtol=1e-4, atol=1e-6)
            checked += 1
    assert checked >= 8


# ---------------------------------------------------------------------------
# Generic FQN splitter (model-agnostic splitting; reference
# pipeline_parallelism.py:131-277 + stages_generator.py:15-120)

def test_fqn_split_gpt2_equals_full_model():
    from modalities_amd.parallel.pp_split import split_model_into_stages_by_fqn
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages_by_fqn(model, 2)
    assert stages[0].is_first and stages[1].is_last
    x, _ = make_batch()
    with torch.no_grad():
        full = model({"input_ids": x})["logits"]
        chained = stages[1](stages[0](x))
    torch.testing.assert_close(full, chained)
    # every segment lives on exactly one stage; balanced by param count
    all_fqns = stages[0].fqns + stages[1].fqns
    assert all_fqns[0] == "embed" and all_fqns[-1] == "head"
    assert len(set(all_fqns)) == len(all_fqns) == 4 + 2


def test_fqn_split_explicit_fqns_and_errors():
    from modalities_amd.parallel.pp_split import split_model_into_stages_by_fqn
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages_by_fqn(
        model, 2, stage_fqns=[["embed", "blocks.0"],
                              ["blocks.1", "blocks.2", "blocks.3", "head"]])
    x, _ = make_batch()
    with torch.no_grad():
        full = model({"input_ids": x})["logits"]
        torch.testing.assert_close(stages[1](stages[0](x)), full)
    with pytest.raises(KeyError, match="does_not_exist"):
        split_model_into_stages_by_fqn(model, 2,
                                       stage_fqns=[["does_not_exist"], ["head"]])


def test_fqn_split_arbitrary_sequential():
    """Any nn.Sequential splits without model-specific code."""
    from modalities_amd.parallel.pp_split import split_model_into_stages_by_fqn
    torch.manual_seed(1)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 64), torch.nn.GELU(), torch.nn.Linear(64, 64),
        torch.nn.GELU(), torch.nn.Linear(64, 64), torch.nn.GELU(),
        torch.nn.Linear(64, 64))
    stages = split_model_into_stages_by_fqn(model, 2)
    x = torch.randn(3, 64)
    with torch.no_grad():
        torch.testing.assert_close(stages[1](stages[0](x)), model(x))
    # balance by params: 2 of the 4 equal linears per stage
    w = [sum(p.numel() for p in s.parameters()) for s in stages]
    assert max(w) / max(1, min(w)) < 1.5, w


def _pp_fqn_worker(rank, world):
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import get_pipeline_schedule
    from modalities_amd.parallel.pp_split import split_model_into_stages_by_fqn
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages_by_fqn(model, world)
    sched = get_pipeline_schedule(
        "1f1b", stage=stages[rank], stage_idx=rank, num_stages=world,
        n_microbatches=N_MB, group=dist.group.WORLD)
    x, y = make_batch()
    losses = sched.step(x, y, CLMCrossEntropyLoss("target_ids", "logits"))
    return [l.item() for l in losses]


def test_pp2_fqn_stages_match_single_process():
    ref_losses, _ = reference_loss_and_grads()
    results = run_distributed(_pp_fqn_worker, world_size=2,
                              port=find_free_port())
    assert results[1] == pytest.approx(ref_losses, rel=1e-5)


def _pp_gradnorm_worker(rank, world):
    import torch.distributed as dist

    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import get_pipeline_schedule
    from modalities_amd.training.gradient_clipping import GradientClipper
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages(model, world)
    sched = get_pipeline_schedule(
        "gpipe", stage=stages[rank], stage_idx=rank, num_stages=world,
        n_microbatches=2, group=dist.group.WORLD)
    x, y = make_batch()
    sched.step(x, y, CLMCrossEntropyLoss("target_ids", "logits"))
    clipper = GradientClipper(max_norm=1.0, pp_group=dist.group.WORLD)
    return clipper(stages[rank]).item()


def test_pp_grad_norm_is_model_global():
    """The published grad norm under PP must be the MODEL-global norm
    (identical on every stage; reference fsdp_gradient_clipper.py:166-169),
    not the stage-local norm."""
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    x, y = make_batch()
    for mb_x, mb_y in zip(x.chunk(2), y.chunk(2)):
        out = model({"input_ids": mb_x})["logits"]
        loss = torch.nn.functional.cross_entropy(
            out.reshape(-1, VOCAB).float(), mb_y.reshape(-1))
        (loss / 2).backward()
    ref = torch.sqrt(sum(p.grad.float().pow(2).sum()
                         for p in model.parameters())).item()
    results = run_distributed(_pp_gradnorm_worker, world_size=2,
                              port=find_free_port())
    assert results[0] == pytest.approx(results[1], rel=1e-6)
    assert results[0] == pytest.approx(ref, rel=1e-4)


def _evaluator_pp_worker(rank, world):
    import torch.distributed as dist

    from modalities_amd.batch import DatasetBatch
    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.parallel.pp import get_pipeline_schedule
    from modalities_amd.training.evaluator import Evaluator
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    stages = split_model_into_stages(model, world)
    sched = get_pipeline_schedule(
        "1f1b", stage=stages[rank], stage_idx=rank, num_stages=world,
        n_microbatches=N_MB, group=dist.group.WORLD)
    ev = Evaluator(progress_publisher=_NullPub(), evaluation_result_publisher=_NullPub(),
                   pp_schedule=sched)
    x, y = make_batch()
    batch = DatasetBatch(samples={"input_ids": x}, targets={"target_ids": y})
    loss = ev.evaluate_batch(batch, stages[rank],
                             CLMCrossEntropyLoss("target_ids", "logits"))
    return loss.item()


class _NullPub:
    def publish_message(self, *a, **k):
        pass


def test_evaluator_dispatches_pp_schedule():
    """Evaluator.evaluate_batch runs the forward-only PP schedule and every
    rank sees the broadcast mean loss (reference evaluator.py:88-180)."""
    ref_losses, _ = reference_loss_and_grads()
    ref_mean = sum(ref_losses) / N_MB
    results = run_distributed(_evaluator_pp_worker, world_size=2,
                              port=find_free_port())
    for r in range(2):
        assert results[r] == pytest.approx(ref_mean, rel=1e-5)
