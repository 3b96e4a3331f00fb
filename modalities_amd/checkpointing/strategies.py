"""Checkpoint save/delete decision strategies (capability parity with
reference src/modalities/checkpointing/checkpoint_saving_strategies.py:36-120).
"""

from dataclasses import dataclass, field

from modalities_amd.training.progress import TrainingProgress


@dataclass
class CheckpointingInstruction:
    save_current: bool = False
    checkpoints_to_delete: list[TrainingProgress] = field(default_factory=list)


class CheckpointSavingStrategy:
    def get_checkpoint_instruction(self, training_progress: TrainingProgress,
                                   ) -> CheckpointingInstruction:
        raise NotImplementedError


class SaveEveryKStepsCheckpointingStrategy(CheckpointSavingStrategy):
    def __init__(self, k: int):
        self.k = k

    def get_checkpoint_instruction(self, training_progress: TrainingProgress):
        save = training_progress.num_seen_steps_total % self.k == 0 \
            and training_progress.num_seen_steps_total > 0
        return CheckpointingInstruction(save_current=save)


class SaveKMostRecentCheckpointsStrategy(CheckpointSavingStrategy):
    """Always save; keep only the k most recent (k=-1: keep all; k=0: keep
    none after the run — matches reference semantics)."""

    def __init__(self, k: int = -1):
        self.k = k
        self.saved_instances: list[TrainingProgress] = []

    def get_checkpoint_instruction(self, training_progress: TrainingProgress):
        import copy
        self.saved_instances.append(copy.deepcopy(training_progress))
        to_delete: list[TrainingProgress] = []
        if self.k >= 0 and len(self.saved_instances) > self.k:
            while len(self.saved_instances) > max(self.k, 1):
                to_delete.append(self.saved_instances.pop(0))
            if self.k == 0:
                to_delete.append(self.saved_instances.pop(0))
        return CheckpointingInstruction(save_current=True,
                                        checkpoints_to_delete=to_delete)
