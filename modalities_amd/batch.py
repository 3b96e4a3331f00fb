"""Batch types (capability parity with reference src/modalities/batch.py)."""

from dataclasses import dataclass, field
from typing import Optional

import torch


class Batch:
    @staticmethod
    def _to_device(d: dict, device: torch.device) -> dict:
        return {k: v.to(device, non_blocking=True) for k, v in d.items()}


@dataclass
class DatasetBatch(Batch):
    """A batch of samples and targets, usually produced by a collate fn."""

    samples: dict[str, torch.Tensor]
    targets: dict[str, torch.Tensor]
    batch_dim: int = 0

    def to(self, device: torch.device) -> "DatasetBatch":
        self.samples = self._to_device(self.samples, device)
        self.targets = self._to_device(self.targets, device)
        return self

    def detach(self) -> "DatasetBatch":
        self.samples = {k: v.detach() for k, v in self.samples.items()}
        self.targets = {k: v.detach() for k, v in self.targets.items()}
        return self

    def __len__(self) -> int:
        return next(iter(self.samples.values())).shape[self.batch_dim]


@dataclass
class InferenceResultBatch(Batch):
    """Model predictions with the targets they should be scored against."""

    targets: dict[str, torch.Tensor] = field(default_factory=dict)
    predictions: dict[str, torch.Tensor] = field(default_factory=dict)
    batch_dim: int = 0

    def get_predictions(self, key: str) -> torch.Tensor:
        if key not in self.predictions:
            raise ValueError(f"Prediction key {key!r} not found ({list(self.predictions)})")
        return self.predictions[key]

    def get_targets(self, key: str) -> torch.Tensor:
        if key not in self.targets:
            raise ValueError(f"Target key {key!r} not found ({list(self.targets)})")
        return self.targets[key]

    def to(self, device: torch.device) -> "InferenceResultBatch":
        self.targets = self._to_device(self.targets, device)
        self.predictions = self._to_device(self.predictions, device)
        return self

    def detach(self) -> "InferenceResultBatch":
        self.targets = {k: v.detach() for k, v in self.targets.items()}
        self.predictions = {k: v.detach() for k, v in self.predictions.items()}
        return self


@dataclass
class ResultItem:
    value: torch.Tensor
    decimal_places: Optional[int] = None


@dataclass
class EvaluationResultBatch(Batch):
    """Aggregated metrics for a dataloader at a training step."""

    dataloader_tag: str
    num_train_steps_done: int
    losses: dict[str, ResultItem] = field(default_factory=dict)
    metrics: dict[str, ResultItem] = field(default_factory=dict)
    throughput_metrics: dict[str, ResultItem] = field(default_factory=dict)

    def __str__(self) -> str:
        def fmt(d):
            parts = []
            for k, item in d.items():
                v = item.value
                v = v.item() if isinstance(v, torch.Tensor) and v.numel() == 1 else v
                if item.decimal_places is not None and isinstance(v, float):
                    v = round(v, item.decimal_places)
                parts.append(f"{k}: {v}")
            return " ".join(parts)

        return (f"Evaluation result on dataset tag {self.dataloader_tag} after "
                f"{self.num_train_steps_done} steps: {fmt(self.losses)} "
                f"{fmt(self.metrics)} {fmt(self.throughput_metrics)}")
