"""Evaluation loop over multiple dataloaders (capability parity with
reference src/modalities/evaluator.py:45-180)."""

import time

import torch

from modalities_amd.batch import EvaluationResultBatch, ResultItem
from modalities_amd.logging_broker.broker import (ExperimentStatus, MessagePublisher,
                                                  MessageTypes, ProgressUpdate)
from modalities_amd.loss_functions import Loss
from modalities_amd.models.model import model_predict_batch
from modalities_amd.running_env import Reducer


class Evaluator:
    def __init__(self, progress_publisher: MessagePublisher,
                 evaluation_result_publisher: MessagePublisher, device=None,
                 pp_schedule=None):
        self.progress_publisher = progress_publisher
        self.evaluation_result_publisher = evaluation_result_publisher
        self.device = device or torch.device("cpu")
        # PipelineSchedule when PP is active: evaluation dispatches the
        # forward-only schedule (reference evaluator.py:88-180)
        self.pp_schedule = pp_schedule

    @torch.no_grad()
    def evaluate_batch(self, batch, model, loss_fun: Loss) -> torch.Tensor:
        from modalities_amd.training.trainer import Trainer
        Trainer._propagate_tp_vocab_info(model, loss_fun)
        if self.pp_schedule is not None:
            batch = batch.to(self.device)
            inputs = next(iter(batch.samples.values()))
            targets = next(iter(batch.targets.values()))
            losses: list = []
            self.pp_schedule.eval_step(inputs, targets, loss_fun, losses)
            return self.pp_schedule.broadcast_mean_loss(losses)
        batch = batch.to(self.device)
        cp_info = Trainer._cp_info_of(model)
        if cp_info is not None:
            # CP-patched model returns seq-sharded logits [B, T/cp, V]:
            # slice the targets identically (same as Trainer.train_step)
            from modalities_amd.batch import DatasetBatch
            from modalities_amd.parallel.cp import slice_targets_for_cp
            group, cp_rank, cp_size = cp_info
            batch = DatasetBatch(
                samples=batch.samples,
                targets={k: slice_targets_for_cp(v, cp_rank, cp_size)
                         for k, v in batch.targets.items()})
        result_batch = model_predict_batch(model, batch)
        return loss_fun(result_batch)

    def evaluate(self, model, data_loaders: list, loss_fun: Loss,
                 num_train_steps_done: int) -> dict[str, EvaluationResultBatch]:
        model.eval()
        results = {}
        for data_loader in data_loaders:
            tag = getattr(data_loader, "dataloader_tag", "val")
            cumulated = torch.zeros(3)
            start = time.perf_counter()
            for step, batch in enumerate(data_loader):
                loss = self.evaluate_batch(batch, model, loss_fun)
                cumulated[0] += loss.item()
                cumulated[1] = loss.item()
                cumulated[2] += 1
                self.progress_publisher.publish_message(
                    ProgressUpdate(step, ExperimentStatus.EVALUATION, tag),
                    MessageTypes.BATCH_PROGRESS_UPDATE)
            elapsed = time.perf_counter() - start
            reduced = Reducer.reduce(cumulated.clone())
            n = max(reduced[2].item(), 1.0)
            result = EvaluationResultBatch(
                dataloader_tag=tag, num_train_steps_done=num_train_steps_done,
                losses={loss_fun.tag: ResultItem(reduced[0] / n, 4)},
                throughput_metrics={"eval batches/s": ResultItem(
                    torch.tensor(cumulated[2].item() / max(elapsed, 1e-9)), 2)})
            self.evaluation_result_publisher.publish_message(
                result, MessageTypes.EVALUATION_RESULT)
            results[tag] = result
        model.train()
        return results
