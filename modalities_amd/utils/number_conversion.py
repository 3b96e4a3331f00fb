"""Config-time arithmetic components (capability parity with reference
src/modalities/utils/number_conversion.py:74-372): steps <-> tokens <->
samples <-> batches conversions, plus parsing seen/target steps/tokens out
of a checkpoint path (our checkpoint folders use the same
``seen_steps_N-seen_tokens_N-target_steps_N-target_tokens_N`` schema,
modalities_amd/checkpointing/saving.py)."""

import pickle
import re
from pathlib import Path


class NumberConversion:
    @staticmethod
    def _checkpoint_value(pattern: str, string: str) -> int:
        matches = re.findall(pattern, string)
        if len(matches) > 1:
            raise ValueError(f"Multiple matches for {pattern} in {string}")
        if not matches:
            raise ValueError(f"No match for {pattern} in {string}")
        return int(matches[0])

    # ---- pure arithmetic -------------------------------------------------

    @staticmethod
    def get_local_num_batches_from_num_samples(num_ranks: int,
                                               global_num_samples: int,
                                               local_micro_batch_size: int) -> int:
        return global_num_samples // (num_ranks * local_micro_batch_size)

    @staticmethod
    def get_num_samples_from_num_tokens(num_tokens: int, sequence_length: int) -> int:
        return num_tokens // sequence_length

    @staticmethod
    def get_local_num_batches_from_num_tokens(num_ranks: int, global_num_tokens: int,
                                              sequence_length: int,
                                              local_micro_batch_size: int) -> int:
        global_num_samples = global_num_tokens // sequence_length
        return global_num_samples // (num_ranks * local_micro_batch_size)

    @staticmethod
    def get_num_steps_from_num_samples(num_ranks: int, local_micro_batch_size: int,
                                       global_num_samples: int,
                                       gradient_accumulation_steps: int) -> int:
        return global_num_samples // (num_ranks * local_micro_batch_size
                                      * gradient_accumulation_steps)

    @staticmethod
    def get_num_steps_from_num_tokens(dp_degree: int, local_micro_batch_size: int,
                                      global_num_tokens: int, sequence_length: int,
                                      gradient_accumulation_steps: int) -> int:
        global_num_samples = global_num_tokens // sequence_length
        return global_num_samples // (dp_degree * local_micro_batch_size
                                      * gradient_accumulation_steps)

    @staticmethod
    def get_num_tokens_from_num_steps(num_steps: int, dp_degree: int,
                                      local_micro_batch_size: int,
                                      sequence_length: int,
                                      gradient_accumulation_steps: int) -> int:
        return (num_steps * dp_degree * local_micro_batch_size * sequence_length
                * gradient_accumulation_steps)

    @staticmethod
    def get_num_samples_from_num_steps(num_steps: int, dp_degree: int,
                                       local_micro_batch_size: int,
                                       gradient_accumulation_steps: int) -> int:
        return (num_steps * dp_degree * local_micro_batch_size
                * gradient_accumulation_steps)

    # ---- checkpoint-path parsing ----------------------------------------

    @staticmethod
    def get_last_step_from_checkpoint_path(checkpoint_path: Path) -> int:
        return NumberConversion._checkpoint_value(r"seen_steps_(\d+)",
                                                  str(checkpoint_path)) - 1

    @staticmethod
    def get_num_seen_steps_from_checkpoint_path(checkpoint_path: Path) -> int:
        return NumberConversion._checkpoint_value(r"seen_steps_(\d+)",
                                                  str(checkpoint_path))

    @staticmethod
    def get_global_num_seen_tokens_from_checkpoint_path(checkpoint_path: Path) -> int:
        return NumberConversion._checkpoint_value(r"seen_tokens_(\d+)",
                                                  str(checkpoint_path))

    @staticmethod
    def get_global_num_target_tokens_from_checkpoint_path(checkpoint_path: Path) -> int:
        return NumberConversion._checkpoint_value(r"target_tokens_(\d+)",
                                                  str(checkpoint_path))

    @staticmethod
    def get_num_target_steps_from_checkpoint_path(checkpoint_path: Path) -> int:
        return NumberConversion._checkpoint_value(r"target_steps_(\d+)",
                                                  str(checkpoint_path))

    # ---- dataset-derived -------------------------------------------------

    @staticmethod
    def get_num_tokens_from_packed_mem_map_dataset_continuous(
            dataset_path: Path, sequence_length: int, num_ranks: int,
            local_micro_batch_size: int, gradient_accumulation_steps: int) -> int:
        """Number of tokens the training will actually consume from a .pbin
        (floored to whole steps; reference number_conversion.py:288-341)."""
        from modalities_amd.dataloader.packed_data import EmbeddedStreamData
        data = EmbeddedStreamData(Path(dataset_path))
        num_dataset_tokens = data.data_len // data.token_size_in_bytes
        num_samples = (num_dataset_tokens - 1) // sequence_length
        num_steps = NumberConversion.get_num_steps_from_num_samples(
            num_ranks, local_micro_batch_size, num_samples,
            gradient_accumulation_steps)
        return NumberConversion.get_num_tokens_from_num_steps(
            num_steps, num_ranks, local_micro_batch_size, sequence_length,
            gradient_accumulation_steps)

    @staticmethod
    def get_num_steps_from_raw_dataset_index(raw_index_path: Path, num_ranks: int,
                                             local_micro_batch_size: int,
                                             gradient_accumulation_steps: int) -> int:
        with open(raw_index_path, "rb") as f:
            index = pickle.load(f)
        return NumberConversion.get_num_steps_from_num_samples(
            num_ranks, local_micro_batch_size, len(index),
            gradient_accumulation_steps)
