"""LLMDataLoader + collate (capability parity with reference
src/modalities/dataloader/dataloader.py:12-93 and
src/modalities/models/gpt2/collator.py:7-36)."""

from typing import Callable, Optional

import torch
from torch.utils.data import BatchSampler, DataLoader, Dataset

from modalities_amd.batch import DatasetBatch


class GPT2LLMCollateFn:
    """Stack samples; targets are inputs shifted by one token."""

    def __init__(self, sample_key: str, target_key: str):
        self.sample_key = sample_key
        self.target_key = target_key

    def __call__(self, batch: list[dict[str, torch.Tensor]]) -> DatasetBatch:
        sample_tensor = torch.stack([item[self.sample_key] for item in batch])
        samples = {self.sample_key: sample_tensor[:, :-1]}
        targets = {self.target_key: sample_tensor[:, 1:]}
        return DatasetBatch(targets=targets, samples=samples)


class LossMaskingCollateFnWrapper:
    """Mask target spans outside assistant turns for instruction tuning
    (reference: dataloader/collate_fns/collator_fn_wrapper_for_loss_masking.py).

    Masks everything except tokens strictly between b_include_to_loss_token and
    e_include_to_loss_token (vectorized cumsum span logic)."""

    def __init__(self, wrapped_collate_fn, target_keys_to_mask: list[str],
                 loss_ignore_index: int, b_mask_token_id: int, e_mask_token_id: int):
        self.wrapped = wrapped_collate_fn
        self.target_keys_to_mask = target_keys_to_mask
        self.loss_ignore_index = loss_ignore_index
        self.b_mask_token_id = b_mask_token_id
        self.e_mask_token_id = e_mask_token_id

    def __call__(self, batch) -> DatasetBatch:
        db = self.wrapped(batch)
        for key in self.target_keys_to_mask:
            target = db.targets[key]
            mask = torch.zeros_like(target)
            mask += (target == self.b_mask_token_id).int()
            mask -= (target == self.e_mask_token_id).int()
            inside = mask.cumsum(-1)
            # shift by one so the begin token itself is not trained on
            inside = torch.cat([torch.zeros_like(inside[:, :1]), inside[:, :-1]], dim=-1)
            new_target = torch.where(inside > 0, target,
                                     torch.full_like(target, self.loss_ignore_index))
            # never train on the mask tokens themselves
            new_target = torch.where((target == self.b_mask_token_id)
                                     | (target == self.e_mask_token_id),
                                     torch.full_like(target, self.loss_ignore_index),
                                     new_target)
            db.targets[key] = new_target
        return db


class LLMDataLoader(DataLoader):
    """Thin torch DataLoader: batch_sampler-driven, shuffle handled by the
    sampler; carries a dataloader_tag for logging."""

    def __init__(self, dataloader_tag: str, dataset: Dataset, batch_sampler: BatchSampler,
                 collate_fn: Optional[Callable] = None, num_workers: int = 0,
                 pin_memory: bool = False):
        super().__init__(dataset=dataset, batch_sampler=batch_sampler,
                         collate_fn=collate_fn, num_workers=num_workers,
                         pin_memory=pin_memory, shuffle=False)
        self._dataloader_tag = dataloader_tag

    @property
    def dataloader_tag(self) -> str:
        return self._dataloader_tag
