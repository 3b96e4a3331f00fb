"""Checkpoint save/load + warmstart equivalence tests (CPU).

Mirrors the reference's flagship warmstart correctness test (reference:
tests/end2end_tests/test_fsdp_warmstart.py:54-160): losses of a run resumed
from a checkpoint must exactly equal the uninterrupted run. Adds
cross-world-size resharding coverage the reference delegates to DCP.
"""

from pathlib import Path

import pytest
import torch

from modalities_amd.checkpointing import (AppState, CheckpointSaving,
                                          SaveEveryKStepsCheckpointingStrategy,
                                          SaveKMostRecentCheckpointsStrategy,
                                          ShardedCheckpointLoading,
                                          ShardedCheckpointSaving,
                                          read_last_checkpoint_info)
from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.optimizers.lr_schedulers import get_linear_warmup_cosine_annealing
from modalities_amd.optimizers.optimizer_factory import get_adam_w
from modalities_amd.parallel.fsdp import XGMIShardedModel
from modalities_amd.training.progress import TrainingProgress
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


def build(world=1, rank=0, lr=1e-3, warmup=2, total=16):
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    sharded = XGMIShardedModel.from_transformer(
        model, torch.device("cpu"), param_dtype=torch.float32,
        rank=rank, world_size=world)
    opt = get_adam_w(sharded, lr=lr, weight_decay=0.0)
    sched = get_linear_warmup_cosine_annealing(opt, warmup, total)
    return sharded, opt, sched


def train_steps(sharded, opt, sched, step_ids, rank=0, world=1):
    losses = []
    for i in step_ids:
        x, y = make_batch(100 + i)
        n = x.shape[0] // world
        xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
        out = sharded({"input_ids": xs})
        loss = torch.nn.functional.cross_entropy(
            out["logits"].view(-1, VOCAB).float(), ys.reshape(-1))
        loss.backward()
        sharded.backward_epilogue()
        opt.step()
        sched.step()
        opt.zero_grad()
        losses.append(loss.item())
    return losses


def progress_at(step):
    return TrainingProgress(num_seen_steps_current_run=step,
                            num_seen_tokens_current_run=step * 64,
                            num_target_steps=8, num_target_tokens=512)


def test_warmstart_loss_equivalence_world1(tmp_path):
    # uninterrupted 8 steps
    sharded, opt, sched = build()
    ref = train_steps(sharded, opt, sched, range(8))

    # run A: 4 steps then checkpoint
    sharded, opt, sched = build()
    a = train_steps(sharded, opt, sched, range(4))
    app = AppState(sharded, opt, sched)
    saving = ShardedCheckpointSaving(tmp_path, "exp0", global_rank=0)
    saving.save_checkpoint(app, progress_at(4))

    # run B: fresh everything, load, 4 more steps
    sharded2, opt2, sched2 = build()
    # perturb to prove the load overwrites
    sharded2.units[0].master_shard.add_(1.0)
    app2 = AppState(sharded2, opt2, sched2)
    folder = read_last_checkpoint_info(tmp_path / "exp0")
    meta = ShardedCheckpointLoading(0).load_checkpoint_(app2, folder)
    assert meta["num_seen_steps"] == 4
    b = train_steps(sharded2, opt2, sched2, range(4, 8))

    assert ref == pytest.approx(a + b, rel=1e-6), (ref, a + b)


def _warmstart_dp2(rank, world, tmp_dir):
    sharded, opt, sched = build(world, rank)
    ref = train_steps(sharded, opt, sched, range(8), rank, world)

    sharded, opt, sched = build(world, rank)
    a = train_steps(sharded, opt, sched, range(4), rank, world)
    app = AppState(sharded, opt, sched)
    ShardedCheckpointSaving(tmp_dir, "exp0", rank).save_checkpoint(
        app, progress_at(4))

    sharded2, opt2, sched2 = build(world, rank)
    app2 = AppState(sharded2, opt2, sched2)
    folder = read_last_checkpoint_info(f"{tmp_dir}/exp0")
    ShardedCheckpointLoading(rank).load_checkpoint_(app2, folder)
    b = train_steps(sharded2, opt2, sched2, range(4, 8), rank, world)
    return ref, a + b


def test_warmstart_loss_equivalence_dp2(tmp_path):
    results = run_distributed(_warmstart_dp2, world_size=2,
                              port=find_free_port(), args=(str(tmp_path),))
    for rank, (ref, resumed) in results.items():
        assert ref == pytest.approx(resumed, rel=1e-6)


def _save_dp2(rank, world, tmp_dir):
    sharded, opt, sched = build(world, rank)
    train_steps(sharded, opt, sched, range(3), rank, world)
    app = AppState(sharded, opt, sched)
    ShardedCheckpointSaving(tmp_dir, "exp0", rank).save_checkpoint(
        app, progress_at(3))
    return [l.item() if hasattr(l, "item") else l
            for l in [sharded.units[0].master_shard.sum()]]


def test_reshard_world2_to_world1(tmp_path):
    """Checkpoint written at world 2 loads into a world-1 model; the
    continued training matches a world-2 continuation's global behavior by
    comparing full gathered parameters."""
    run_distributed(_save_dp2, world_size=2, port=find_free_port(),
                    args=(str(tmp_path),))

    # world-1 reference trained the same 3 global steps
    sharded_ref, opt_ref, sched_ref = build()
    train_steps(sharded_ref, opt_ref, sched_ref, range(3))

    sharded1, opt1, sched1 = build()
    app1 = AppState(sharded1, opt1, sched1)
    folder = read_last_checkpoint_info(tmp_path / "exp0")
    ShardedCheckpointLoading(0).load_checkpoint_(app1, folder)

    ref_sd = sharded_ref.gather_full_state_dict()
    got_sd = sharded1.gather_full_state_dict()
    assert set(ref_sd) == set(got_sd)
    for k in ref_sd:
        torch.testing.assert_close(ref_sd[k], got_sd[k], rtol=1e-5, atol=1e-6,
                                   msg=lambda m: f"{k}: {m}")


def test_save_every_k_strategy():
    s = SaveEveryKStepsCheckpointingStrategy(k=3)
    decisions = [s.get_checkpoint_instruction(progress_at(i)).save_current
                 for i in range(1, 7)]
    assert decisions == [False, False, True, False, False, True]


def test_keep_k_most_recent_strategy():
    s = SaveKMostRecentCheckpointsStrategy(k=2)
    i1 = s.get_checkpoint_instruction(progress_at(1))
    i2 = s.get_checkpoint_instruction(progress_at(2))
    i3 = s.get_checkpoint_instruction(progress_at(3))
    assert i1.save_current and not i1.checkpoints_to_delete
    assert not i2.checkpoints_to_delete
    assert [p.num_seen_steps_total for p in i3.checkpoints_to_delete] == [1]


def test_checkpoint_saving_combiner_deletes(tmp_path):
    sharded, opt, sched = build()
    app = AppState(sharded, opt, sched)
    saving = CheckpointSaving(SaveKMostRecentCheckpointsStrategy(k=1),
                              ShardedCheckpointSaving(tmp_path, "e", 0))
    train_steps(sharded, opt, sched, range(1))
    saving.save_checkpoint_and_free_memory(progress_at(1), app)
    train_steps(sharded, opt, sched, range(1, 2))
    saving.save_checkpoint_and_free_memory(progress_at(2), app)
    folders = sorted(p.name for p in (tmp_path / "e").iterdir() if p.is_dir())
    assert len(folders) == 1 and "seen_steps_2" in folders[0]


def _load_w2_worker(rank, world, tmp_dir):
    """Load a world-1 checkpoint into a world-2 model (reshard up)."""
    sharded, opt, sched = build(world, rank)
    app = AppState(sharded, opt, sched)
    folder = read_last_checkpoint_info(f"{tmp_dir}/exp1")
    ShardedCheckpointLoading(rank).load_checkpoint_(app, folder)
    sd = sharded.gather_full_state_dict()
    return {k: v.numpy() for k, v in sd.items()}


def test_reshard_world1_to_world2(tmp_path):
    sharded, opt, sched = build()
    train_steps(sharded, opt, sched, range(3))
    app = AppState(sharded, opt, sched)
    ShardedCheckpointSaving(tmp_path, "exp1", 0).save_checkpoint(
        app, progress_at(3))
    ref_sd = sharded.gather_full_state_dict()

    results = run_distributed(_load_w2_worker, world_size=2,
                              port=find_free_port(), args=(str(tmp_path),))
    for rank in (0, 1):
        got = results[rank]
        assert set(got) == set(ref_sd)
        for k in ref_sd:
            torch.testing.assert_close(torch.from_numpy(got[k]), ref_sd[k],
                                       rtol=1e-6, atol=1e-7)


def test_full_state_reassembly_single_process(tmp_path):
    """Sharded checkpoint -> plain module weights (the inference/export
    path)."""
    from modalities_amd.checkpointing.loading import \
        load_full_model_state_from_checkpoint
    sharded, opt, sched = build()
    train_steps(sharded, opt, sched, range(2))
    app = AppState(sharded, opt, sched)
    ShardedCheckpointSaving(tmp_path, "expf", 0).save_checkpoint(
        app, progress_at(2))
    ref_sd = sharded.gather_full_state_dict()

    torch.manual_seed(123)  # different init; must be overwritten
    fresh = GPT2LLM(tiny_cfg())
    folder = read_last_checkpoint_info(tmp_path / "expf")
    load_full_model_state_from_checkpoint(folder, fresh)
    for name, p in fresh.named_parameters():
        torch.testing.assert_close(p.data, ref_sd[name], rtol=1e-6, atol=1e-7)


# ---------------------------------------------------------------------------
# Warmstart equivalence under PP x TP composition (reference
# tests/end2end_tests/test_fsdp2_warmstart_pp_tp.py): the partitioned
# checkpoint layout (one shard layout per pp/tp model partition).

def _pp_tp_build(rank, world):
    import torch.distributed as dist

    from modalities_amd.parallel.mesh import DeviceMesh, ParallelismDegrees
    from modalities_amd.parallel.pp import (get_pipeline_schedule,
                                            split_model_into_stages)
    from modalities_amd.parallel.tp import get_gpt2_tensor_parallelized_model

    mesh = DeviceMesh(world_size=world, rank=rank, pp=2, tp=2, dp_shard=1)
    pp = mesh.dims[ParallelismDegrees.PP]
    tp = mesh.dims[ParallelismDegrees.TP]
    dp = mesh.dims[ParallelismDegrees.DP_SHARD]
    torch.manual_seed(0)
    model = GPT2LLM(tiny_cfg())
    model = get_gpt2_tensor_parallelized_model(
        model, group=tp.group, tp_rank=tp.rank, tp_size=tp.size)
    stages = split_model_into_stages(model, pp.size)
    engine = XGMIShardedModel.from_transformer(
        stages[pp.rank], torch.device("cpu"), process_group=dp.group,
        param_dtype=torch.float32, rank=dp.rank, world_size=dp.size)
    opt = get_adam_w(engine, lr=1e-3, weight_decay=0.0)
    sched = get_pipeline_schedule(
        "1f1b", stage=stages[pp.rank], stage_idx=pp.rank, num_stages=pp.size,
        n_microbatches=2, group=pp.group, sharded_engine=engine)
    return mesh, pp, tp, engine, opt, sched


def _pp_tp_train(sched, opt, step_ids):
    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    means = []
    for i in step_ids:
        x, y = make_batch(100 + i)
        losses = sched.step(x, y, loss_fn)
        opt.step()
        opt.zero_grad()
        means.append(sched.broadcast_mean_loss(losses).item())
    return means


def _warmstart_pp_tp(rank, world, tmp_dir):
    mesh, pp, tp, engine, opt, sched = _pp_tp_build(rank, world)
    part = f"pp{pp.rank}tp{tp.rank}"

    # uninterrupted reference
    ref = _pp_tp_train(sched, opt, range(6))

    # run A: 3 steps, checkpoint (each partition saves its own shards+meta)
    mesh, pp, tp, engine, opt, sched = _pp_tp_build(rank, world)
    a = _pp_tp_train(sched, opt, range(3))
    app = AppState(engine, opt)
    saving = ShardedCheckpointSaving(Path(tmp_dir), "exp_pptp",
                                     global_rank=rank, partition=part,
                                     dp_rank=0, dp_world=1)
    saving.save_checkpoint(app, progress_at(3))

    # run B: fresh build, load this partition, continue
    mesh, pp, tp, engine2, opt2, sched2 = _pp_tp_build(rank, world)
    engine2.units[0].master_shard.add_(0.5)  # prove the load overwrites
    app2 = AppState(engine2, opt2)
    folder = read_last_checkpoint_info(Path(tmp_dir) / "exp_pptp")
    meta = ShardedCheckpointLoading(rank, partition=part).load_checkpoint_(
        app2, folder)
    assert meta["num_seen_steps"] == 3
    b = _pp_tp_train(sched2, opt2, range(3, 6))
    return ref, a + b


def test_warmstart_loss_equivalence_pp2_tp2(tmp_path):
    from tests.utils_dist import run_distributed
    results = run_distributed(_warmstart_pp_tp, world_size=4,
                              port=find_free_port(), args=(str(tmp_path),),
                              timeout_s=300)
    for r in range(4):
        ref, resumed = results[r]
        assert ref == pytest.approx(resumed, rel=1e-6), (r, ref, resumed)
