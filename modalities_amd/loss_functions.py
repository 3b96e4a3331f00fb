"""Loss functions (capability parity with reference
src/modalities/loss_functions.py:10-88). CLM cross-entropy runs the fused
HIP kernel (K8) on device."""

from abc import ABC, abstractmethod

import torch

from modalities_amd.batch import InferenceResultBatch
from modalities_amd.ops import fused_cross_entropy


class Loss(ABC, torch.nn.Module):
    def __init__(self, tag: str):
        super().__init__()
        self._tag = tag

    @property
    def tag(self) -> str:
        return self._tag

    @abstractmethod
    def forward(self, forward_batch: InferenceResultBatch) -> torch.Tensor:
        """Return the batch loss (scalar tensor)."""


class CLMCrossEntropyLoss(Loss):
    def __init__(self, target_key: str, prediction_key: str, tag: str = "CLMCrossEntropyLoss"):
        super().__init__(tag)
        self.target_key = target_key
        self.prediction_key = prediction_key

    def forward(self, forward_batch) -> torch.Tensor:
        # Also directly callable as (logits, targets) for pipeline schedules
        # (reference: loss_functions.py:44-52 dual signature).
        if isinstance(forward_batch, InferenceResultBatch):
            labels = forward_batch.get_targets(self.target_key)
            logits = forward_batch.get_predictions(self.prediction_key)
        else:
            logits, labels = forward_batch
        return fused_cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.reshape(-1), ignore_index=-100)
