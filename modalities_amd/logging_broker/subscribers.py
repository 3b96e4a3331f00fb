"""Message subscribers: rich console, JSONL file, dummy
(capability parity with reference src/modalities/logging_broker/
subscriber_impl/results_subscriber.py and progress_subscriber.py; a WandB
sink is intentionally not bundled — offline environment)."""

import json
from pathlib import Path
from typing import Optional

import torch

from modalities_amd.batch import EvaluationResultBatch
from modalities_amd.logging_broker.broker import Message, MessageSubscriberIF


class DummyResultSubscriber(MessageSubscriberIF):
    def consume_message(self, message: Message) -> None:
        pass

    def consume_dict(self, message_dict: dict) -> None:
        pass


class DummyProgressSubscriber(MessageSubscriberIF):
    def consume_message(self, message: Message) -> None:
        pass

    def consume_dict(self, message_dict: dict) -> None:
        pass


def _jsonify(value):
    if isinstance(value, torch.Tensor):
        return value.item() if value.numel() == 1 else value.tolist()
    return value


class RichResultSubscriber(MessageSubscriberIF):
    """Console sink for evaluation results (rank 0)."""

    def __init__(self, num_ranks: int = 1, global_rank: int = 0):
        self.enabled = global_rank == 0

    def consume_message(self, message: Message) -> None:
        if not self.enabled:
            return
        payload = message.payload
        if isinstance(payload, EvaluationResultBatch):
            print(str(payload), flush=True)

    def consume_dict(self, message_dict: dict) -> None:
        if self.enabled:
            print(json.dumps(message_dict, default=str), flush=True)


class RichProgressSubscriber(MessageSubscriberIF):
    """Live progress bars per dataloader tag (rank 0), built on `rich`
    (reference: logging_broker/subscriber_impl/progress_subscriber.py:22-120).
    Falls back to a no-op when rich is unavailable or stdout is not a TTY
    environment worth animating."""

    def __init__(self, num_ranks: int = 1, global_rank: int = 0,
                 train_split_lengths: Optional[dict] = None,
                 eval_split_lengths: Optional[dict] = None):
        self.enabled = global_rank == 0
        self._tasks: dict = {}
        self._progress = None
        if not self.enabled:
            return
        try:
            from rich.progress import (BarColumn, MofNCompleteColumn, Progress,
                                       TaskProgressColumn, TextColumn,
                                       TimeElapsedColumn)
            self._progress = Progress(
                TextColumn("[progress.description]{task.description}"),
                BarColumn(), MofNCompleteColumn(), TaskProgressColumn(),
                TimeElapsedColumn(), refresh_per_second=2)
            self._progress.start()
            for tag, total in (train_split_lengths or {}).items():
                self._tasks[tag] = self._progress.add_task(tag, total=total)
            for tag, total in (eval_split_lengths or {}).items():
                self._tasks[tag] = self._progress.add_task(tag, total=total)
        except Exception:
            self._progress = None

    def consume_message(self, message: Message) -> None:
        if self._progress is None:
            return
        payload = message.payload
        tag = getattr(payload, "dataloader_tag", None)
        step = getattr(payload, "num_steps_done", None)
        if tag is None or step is None:
            return
        if tag not in self._tasks:
            self._tasks[tag] = self._progress.add_task(tag, total=None)
        self._progress.update(self._tasks[tag], completed=step)

    def consume_dict(self, message_dict: dict) -> None:
        pass

    def close(self):
        if self._progress is not None:
            self._progress.stop()


class ResultsToDiscSubscriber(MessageSubscriberIF):
    """Rank-0 JSONL sink — the machine-readable benchmark/evaluation output
    consumed by sweep tooling (reference: results_subscriber.py:120-168)."""

    def __init__(self, output_file_path: Path, global_rank: int = 0):
        self.enabled = global_rank == 0
        self.path = Path(output_file_path)
        if self.enabled:
            self.path.parent.mkdir(parents=True, exist_ok=True)

    def _write(self, record: dict) -> None:
        with self.path.open("a") as f:
            f.write(json.dumps(record, default=str) + "\n")

    def consume_message(self, message: Message) -> None:
        if not self.enabled:
            return
        payload = message.payload
        if isinstance(payload, EvaluationResultBatch):
            record = {
                "dataloader_tag": payload.dataloader_tag,
                "num_train_steps_done": payload.num_train_steps_done,
                "losses": {k: _jsonify(v.value) for k, v in payload.losses.items()},
                "metrics": {k: _jsonify(v.value) for k, v in payload.metrics.items()},
                "throughput_metrics": {k: _jsonify(v.value)
                                       for k, v in payload.throughput_metrics.items()},
            }
            self._write(record)

    def consume_dict(self, message_dict: dict) -> None:
        if self.enabled:
            self._write(message_dict)


class WandBEvaluationResultSubscriber(MessageSubscriberIF):
    """WandB sink for evaluation results (reference:
    logging_broker/subscriber_impl/results_subscriber.py:61-117). wandb is
    imported lazily — this environment has no network, so construction
    raises a clear error unless wandb is installed and configured."""

    def __init__(self, project: str, experiment_id: str, mode: str = "OFFLINE",
                 directory: Optional[str] = None,
                 config_file_path: Optional[str] = None,
                 global_rank: int = 0):
        self.enabled = global_rank == 0
        if not self.enabled:
            self._run = None
            return
        try:
            import wandb
        except ImportError as e:
            raise ImportError(
                "results_subscriber/wandb requires the wandb package; use "
                "variant_key 'rich' or 'save_all' in offline environments"
            ) from e
        self._run = wandb.init(project=project, name=experiment_id,
                               mode=mode.lower(), dir=directory)
        if config_file_path is not None:
            self._run.save(str(config_file_path))

    def consume_message(self, message: Message) -> None:
        if not self.enabled or self._run is None:
            return
        payload = message.payload
        if isinstance(payload, EvaluationResultBatch):
            step = payload.num_train_steps_done
            prefix = payload.dataloader_tag

            def val(item):
                v = item.value
                return v.item() if isinstance(v, torch.Tensor) and \
                    v.numel() == 1 else v

            logs = {f"{prefix} {k}": val(v) for k, v in payload.losses.items()}
            logs.update({f"{prefix} {k}": val(v)
                         for k, v in payload.metrics.items()})
            logs.update({f"{prefix} {k}": val(v)
                         for k, v in payload.throughput_metrics.items()})
            self._run.log(data=logs, step=step)

    def consume_dict(self, message_dict: dict) -> None:
        if self.enabled and self._run is not None:
            self._run.log(data=message_dict)
