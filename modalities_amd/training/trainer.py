"""The training loop (capability parity with reference
src/modalities/trainer.py:30-392): micro-batch loop with gradient
accumulation, sharded-engine-aware backward/step, loss all-reduce on the log
interval, throughput + MFU + peak-memory publishing, GC control, steppable
profiler hook, evaluation/checkpointing callbacks."""

import gc
import time
from typing import Callable

import torch
import torch.distributed as dist

from modalities_amd.batch import DatasetBatch, EvaluationResultBatch, ResultItem
from modalities_amd.logging_broker.broker import (ExperimentStatus, MessagePublisher,
                                                  MessageTypes, ProgressUpdate)
from modalities_amd.loss_functions import Loss
from modalities_amd.models.model import model_predict_batch
from modalities_amd.parallel.fsdp import XGMIShardedModel
from modalities_amd.running_env import Reducer, is_dist
from modalities_amd.training.progress import TrainingProgress


class GarbageCollection:
    """Disable automatic gc; collect on a fixed step cadence so all ranks
    stall together (pattern also used by TorchTitan and the reference,
    trainer.py:30-46)."""

    def __init__(self, gc_freq: int = 1000):
        self.gc_freq = gc_freq
        gc.disable()
        gc.collect(1)

    def run(self, step_count: int):
        if self.gc_freq > 0 and step_count % self.gc_freq == 0:
            gc.collect(1)


class Trainer:
    def __init__(self, global_rank: int, progress_publisher: MessagePublisher,
                 evaluation_result_publisher: MessagePublisher,
                 gradient_acc_steps: int, global_num_tokens_per_train_step: int,
                 num_seen_train_steps: int, global_num_seen_tokens: int,
                 num_target_steps: int, num_target_tokens: int,
                 gradient_clipper=None, mfu_calculator=None,
                 evaluation_interval_in_steps: int = 0,
                 checkpointing_interval_in_steps: int = 0,
                 training_log_interval_in_steps: int = 1,
                 gc_freq: int = 1000, profiler=None, device=None,
                 pp_schedule=None):
        self.global_rank = global_rank
        self.progress_publisher = progress_publisher
        self.evaluation_result_publisher = evaluation_result_publisher
        self.gradient_acc_steps = gradient_acc_steps
        self.global_num_tokens_per_train_step = global_num_tokens_per_train_step
        self.training_log_interval_in_steps = training_log_interval_in_steps
        self.evaluation_interval_in_steps = evaluation_interval_in_steps
        self.checkpointing_interval_in_steps = checkpointing_interval_in_steps
        self.gradient_clipper = gradient_clipper
        self.mfu_calculator = mfu_calculator
        self.gc = GarbageCollection(gc_freq)
        self.profiler = profiler
        self.device = device or torch.device("cpu")
        self.pp_schedule = pp_schedule  # PipelineSchedule when PP is active
        self.training_progress = TrainingProgress(
            num_seen_steps_current_run=0, num_seen_tokens_current_run=0,
            num_target_steps=num_target_steps, num_target_tokens=num_target_tokens,
            num_seen_steps_previous_run=num_seen_train_steps,
            num_seen_tokens_previous_run=global_num_seen_tokens)

    # ------------------------------------------------------------------
    def _train_batch(self, batch: DatasetBatch, model, optimizer, scheduler,
                     loss_fun: Loss, micro_batch_id: int):
        """One micro-batch: forward, loss, backward; on accumulation boundary
        clip + step + zero. Returns (step_performed, loss detached, grad_norm).

        With a pipeline schedule, the schedule's internal micro-batching IS
        the accumulation: one call = one full fwd/bwd over its splits
        (reference trainer.py:162-177 pp_schedule.step dispatch)."""
        if self.pp_schedule is not None:
            inputs = next(iter(batch.samples.values()))
            targets = next(iter(batch.targets.values()))
            losses: list = []
            self.pp_schedule.step(inputs, targets, loss_fun, losses)
            loss = self.pp_schedule.broadcast_mean_loss(losses)
            grad_norm = None
            if (micro_batch_id + 1) % self.gradient_acc_steps == 0:
                if self.gradient_clipper is not None:
                    grad_norm = self.gradient_clipper(model)
                optimizer.step()
                if scheduler is not None:
                    scheduler.step()
                optimizer.zero_grad()
                return True, loss, grad_norm
            return False, loss, None
        sharded = isinstance(model, XGMIShardedModel)
        self._propagate_tp_vocab_info(model, loss_fun)
        cp_info = self._cp_info_of(model)
        if cp_info is not None:
            # CP: logits come back seq-sharded [B, T/cp, V]; slice the
            # targets identically so the loss shapes match.
            from modalities_amd.parallel.cp import slice_targets_for_cp
            group, cp_rank, cp_size = cp_info
            batch = DatasetBatch(
                samples=batch.samples,
                targets={k: slice_targets_for_cp(v, cp_rank, cp_size)
                         for k, v in batch.targets.items()})
        result_batch = model_predict_batch(model, batch)
        loss = loss_fun(result_batch)
        # With CP the local mean covers 1/cp of the tokens: backward through
        # loss/cp so the cp-SUM of grads equals the full-sequence gradient
        # (matches tests/test_context_parallelism.py loss normalization).
        loss_scale = self.gradient_acc_steps * (cp_info[2] if cp_info else 1)
        (loss / loss_scale).backward()
        if sharded:
            model.backward_epilogue()

        grad_norm = None
        step_performed = False
        if (micro_batch_id + 1) % self.gradient_acc_steps == 0:
            if cp_info is not None:
                # sum CP-partial grads once per optimizer step (linear in the
                # accumulated grads), before clipping / the optimizer.
                from modalities_amd.parallel.cp import cp_grad_allreduce_
                cp_grad_allreduce_(model, cp_info[0])
            if self.gradient_clipper is not None:
                grad_norm = self.gradient_clipper(model)
            optimizer.step()
            if scheduler is not None:
                scheduler.step()
            optimizer.zero_grad()
            step_performed = True
        return step_performed, loss.detach(), grad_norm

    # ------------------------------------------------------------------
    def train(self, model, train_loader, optimizer, scheduler, loss_fun: Loss,
              evaluation_callback: Callable[[int], None] = lambda s: None,
              checkpointing_callback: Callable[[TrainingProgress], None] = lambda p: None):
        model.train()
        # initial callbacks at step 0 (reference: trainer.py:250-259)
        evaluation_callback(self.training_progress.num_seen_steps_total)

        cumulated_losses = self._reset_loss_tracker()
        grad_norm_window: list[torch.Tensor] = []
        micro_batch_id = 0
        window_start = time.perf_counter()
        num_steps_at_window_start = self.training_progress.num_seen_steps_total
        profiler_cm = self.profiler if self.profiler is not None else _NullProfiler()

        max_micro_batches = (self.training_progress.num_target_steps
                             - self.training_progress.num_seen_steps_previous_run) \
            * self.gradient_acc_steps

        with profiler_cm:
            for batch in train_loader:
                if micro_batch_id >= max_micro_batches:
                    break
                batch = batch.to(self.device)
                step_performed, loss, grad_norm = self._train_batch(
                    batch, model, optimizer, scheduler, loss_fun, micro_batch_id)
                cumulated_losses[0] += loss.item()
                cumulated_losses[1] = loss.item()  # true last-batch loss
                cumulated_losses[-1] += 1
                if grad_norm is not None:
                    grad_norm_window.append(grad_norm)

                if step_performed:
                    self.training_progress.num_seen_steps_current_run += 1
                    self.training_progress.num_seen_tokens_current_run += \
                        self.global_num_tokens_per_train_step
                    steps_total = self.training_progress.num_seen_steps_total

                    self.progress_publisher.publish_message(
                        ProgressUpdate(steps_total, ExperimentStatus.TRAIN, "train"),
                        MessageTypes.BATCH_PROGRESS_UPDATE)

                    if steps_total % self.training_log_interval_in_steps == 0:
                        self._publish_train_metrics(
                            model, optimizer, loss_fun, cumulated_losses,
                            grad_norm_window, window_start, num_steps_at_window_start)
                        cumulated_losses = self._reset_loss_tracker()
                        grad_norm_window = []
                        window_start = time.perf_counter()
                        num_steps_at_window_start = steps_total

                    self.gc.run(steps_total)
                    if self.evaluation_interval_in_steps > 0 and \
                            steps_total % self.evaluation_interval_in_steps == 0:
                        evaluation_callback(steps_total)
                        model.train()
                    if self.checkpointing_interval_in_steps > 0 and \
                            steps_total % self.checkpointing_interval_in_steps == 0:
                        checkpointing_callback(self.training_progress)
                micro_batch_id += 1
                profiler_cm.step()
        # final checkpoint at end of training if not already checkpointed
        if self.checkpointing_interval_in_steps > 0 and (
                self.training_progress.num_seen_steps_total
                % self.checkpointing_interval_in_steps != 0):
            checkpointing_callback(self.training_progress)

    # ------------------------------------------------------------------
    @staticmethod
    def _propagate_tp_vocab_info(model, loss_fun) -> None:
        """When the model's lm_head is vocab-sharded (TP shard_vocab), the
        loss must combine across the tp group (vocab-parallel CE). The TP
        transform leaves `_tp_vocab_info` on the model; hand it to the
        loss once."""
        if getattr(loss_fun, "tp_vocab_info", "no") is None:
            for m in (model, getattr(model, "module", None)):
                info = getattr(m, "_tp_vocab_info", None) if m is not None else None
                if info is not None:
                    loss_fun.tp_vocab_info = info
                    return

    @staticmethod
    def _cp_info_of(model):
        """(group, cp_rank, cp_size) when the model (or the module inside a
        sharded wrapper) was CP-patched; None otherwise."""
        for m in (model, getattr(model, "module", None)):
            info = getattr(m, "_cp_info", None) if m is not None else None
            if info is not None and info[2] > 1:
                return info
        return None

    @staticmethod
    def _reset_loss_tracker() -> torch.Tensor:
        # [sum of last-interval losses, last loss, num batches]
        return torch.zeros(3)

    def _publish_train_metrics(self, model, optimizer, loss_fun, cumulated_losses,
                               grad_norm_window, window_start, num_steps_at_window_start):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if is_dist():
            dist.barrier()
        elapsed = time.perf_counter() - window_start
        steps_done = self.training_progress.num_seen_steps_total - num_steps_at_window_start
        samples_per_step = self.global_num_tokens_per_train_step
        # losses: all-reduce over ranks (reference: trainer.py:321-333)
        reduced = Reducer.reduce(cumulated_losses.clone())
        n = max(reduced[-1].item(), 1.0)
        avg_loss = reduced[0] / n
        world = dist.get_world_size() if is_dist() else 1

        tokens_per_second = steps_done * self.global_num_tokens_per_train_step / \
            max(elapsed, 1e-9)
        throughput = {
            "train tokens/s (global)": ResultItem(torch.tensor(tokens_per_second), 1),
            "train steps/s": ResultItem(torch.tensor(steps_done / max(elapsed, 1e-9)), 3),
        }
        if self.mfu_calculator is not None:
            samples_per_second = tokens_per_second / self.mfu_calculator.sequence_length
            throughput["train mfu (global)"] = ResultItem(
                self.mfu_calculator.compute(torch.tensor(samples_per_second)), 4)
        if torch.cuda.is_available():
            throughput["peak memory rank0 (MB)"] = ResultItem(
                torch.tensor(torch.cuda.max_memory_allocated() / 1e6), 1)
        metrics = {
            "consumed tokens": ResultItem(
                torch.tensor(self.training_progress.num_seen_tokens_total), 0),
            "lr mean": ResultItem(torch.tensor(
                sum(g["lr"] for g in optimizer.param_groups)
                / len(optimizer.param_groups)), 6),
        }
        if grad_norm_window:
            gw = torch.stack([g if isinstance(g, torch.Tensor) else torch.tensor(g)
                              for g in grad_norm_window]).float()
            metrics["grad norm avg"] = ResultItem(gw.mean(), 4)
            metrics["grad norm last"] = ResultItem(gw[-1], 4)

        result = EvaluationResultBatch(
            dataloader_tag="train",
            num_train_steps_done=self.training_progress.num_seen_steps_total,
            losses={f"{loss_fun.tag} average": ResultItem(avg_loss, 4),
                    # last-BATCH loss, rank-averaged (reference publishes the
                    # true last loss, not the window average again)
                    f"{loss_fun.tag} last": ResultItem(reduced[1] / world, 4)},
            metrics=metrics, throughput_metrics=throughput)
        self.evaluation_result_publisher.publish_message(
            result, MessageTypes.EVALUATION_RESULT)


class _NullProfiler:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False

    def step(self):
        pass
