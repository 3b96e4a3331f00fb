from modalities_amd.checkpointing.app_state import AppState
from modalities_amd.checkpointing.loading import (ShardedCheckpointLoading,
                                                  read_checkpoint_meta,
                                                  read_last_checkpoint_info)
from modalities_amd.checkpointing.saving import (CheckpointSaving,
                                                 ShardedCheckpointSaving,
                                                 checkpoint_folder_name)
from modalities_amd.checkpointing.strategies import (
    CheckpointingInstruction, SaveEveryKStepsCheckpointingStrategy,
    SaveKMostRecentCheckpointsStrategy)

__all__ = [
    "AppState", "CheckpointSaving", "ShardedCheckpointSaving",
    "ShardedCheckpointLoading", "CheckpointingInstruction",
    "SaveEveryKStepsCheckpointingStrategy", "SaveKMostRecentCheckpointsStrategy",
    "checkpoint_folder_name", "read_checkpoint_meta", "read_last_checkpoint_info",
]
