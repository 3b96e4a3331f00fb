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
        # set by the trainer/evaluator when the model's lm_head is
        # vocab-sharded (TP): (group, tp_rank, tp_size, vocab)
        self.tp_vocab_info = None

    def forward(self, forward_batch) -> torch.Tensor:
        # Also directly callable as (logits, targets) for pipeline schedules
        # (reference: loss_functions.py:44-52 dual signature).
        if isinstance(forward_batch, InferenceResultBatch):
            labels = forward_batch.get_targets(self.target_key)
            logits = forward_batch.get_predictions(self.prediction_key)
        else:
            logits, labels = forward_batch
        if self.tp_vocab_info is not None:
            from modalities_amd.parallel.tp import vocab_parallel_cross_entropy
            group, tp_rank, tp_size, vocab = self.tp_vocab_info
            return vocab_parallel_cross_entropy(
                logits.view(-1, logits.shape[-1]), labels.reshape(-1),
                group, tp_rank, tp_size, vocab, ignore_index=-100)
        return fused_cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.reshape(-1), ignore_index=-100)


class NCELoss(Loss):
    """Symmetric InfoNCE contrastive loss over paired embeddings
    (capability parity with reference src/modalities/loss_functions.py:125-167;
    used by CoCa)."""

    def __init__(self, prediction_key1: str, prediction_key2: str,
                 is_asymmetric: bool = False, temperature: float = 1.0,
                 tag: str = "NCELoss"):
        super().__init__(tag)
        self.prediction_key1 = prediction_key1
        self.prediction_key2 = prediction_key2
        self.is_asymmetric = is_asymmetric
        self.temperature = temperature

    def forward(self, forward_batch: InferenceResultBatch) -> torch.Tensor:
        emb1 = forward_batch.get_predictions(self.prediction_key1).float()
        emb2 = forward_batch.get_predictions(self.prediction_key2).float()
        emb1 = torch.nn.functional.normalize(emb1, dim=-1)
        emb2 = torch.nn.functional.normalize(emb2, dim=-1)
        logits = emb1 @ emb2.t() / self.temperature
        labels = torch.arange(emb1.shape[0], device=emb1.device)
        loss = torch.nn.functional.cross_entropy(logits, labels)
        if not self.is_asymmetric:
            loss = 0.5 * (loss + torch.nn.functional.cross_entropy(
                logits.t(), labels))
        return loss
