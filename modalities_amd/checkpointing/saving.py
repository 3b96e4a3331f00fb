"""Sharded checkpoint saving (DCP-equivalent for the MI355X-native engine).

Capability parity with the reference's strategy/execution split
(reference: src/modalities/checkpointing/checkpoint_saving.py:8-55,
checkpointing/fsdp/fsdp_checkpoint_saving.py:179-282): each rank writes its
flat fp32 shard bundle into a checkpoint folder whose name encodes
experiment id + seen/target steps/tokens; rank 0 writes ``meta.json`` (shard
layout for cross-world-size resume) and ``last_checkpoint_info.json`` (the
resume pointer, same schema as the reference:
``{"checkpoint_folder_path": <abs path>}``).

On-disk layout of one checkpoint:
    eid_{id}-seen_steps_{s}-seen_tokens_{t}-target_steps_{S}-target_tokens_{T}/
        meta.json                  # world_size, shard layout, progress, replicated state
        shards_rank_{r}.pt         # {key: flat fp32 tensor} for rank r
"""

import json
import shutil
from pathlib import Path

import torch
import torch.distributed as dist

from modalities_amd.checkpointing.app_state import AppState
from modalities_amd.checkpointing.strategies import CheckpointSavingStrategy
from modalities_amd.running_env import is_dist
from modalities_amd.training.progress import TrainingProgress

CHECKPOINT_FOLDER_FMT = ("eid_{experiment_id}-seen_steps_{seen_steps}"
                         "-seen_tokens_{seen_tokens}-target_steps_{target_steps}"
                         "-target_tokens_{target_tokens}")


def checkpoint_folder_name(experiment_id: str, progress: TrainingProgress) -> str:
    return CHECKPOINT_FOLDER_FMT.format(
        experiment_id=experiment_id,
        seen_steps=progress.num_seen_steps_total,
        seen_tokens=progress.num_seen_tokens_total,
        target_steps=progress.num_target_steps,
        target_tokens=progress.num_target_tokens)


class ShardedCheckpointSaving:
    """Execution half: writes/deletes checkpoint folders.

    ``partition`` namespaces a MODEL partition under PP/TP composition
    (e.g. "pp0_tp1"): the engine's DP-shard resharding applies within one
    partition; each partition carries its own shard layout and meta file
    (reference analog: DCP's global FQN space handles this implicitly;
    tests/end2end_tests/test_fsdp2_warmstart_pp_tp.py). The default ""
    keeps the flat single-partition layout (meta.json / shards_rank_N.pt).

    ``dp_rank``/``dp_world`` override the file-naming rank/world when the
    engine's DP group is a subgroup of WORLD (PP/TP composition); the
    default uses the global rank/world (pure-DP runs).
    """

    def __init__(self, checkpoint_path: Path, experiment_id: str,
                 global_rank: int, partition: str = "",
                 dp_rank: int = None, dp_world: int = None,
                 write_enabled: bool = True):
        self.checkpoint_path = Path(checkpoint_path)
        self.experiment_id = experiment_id
        self.global_rank = global_rank
        self.partition = partition
        self.dp_rank = dp_rank
        self.dp_world = dp_world
        # replicate/CP peers hold identical parameters: only their rank-0
        # peer writes shards/meta (all ranks still hit the closing barrier)
        self.write_enabled = write_enabled

    def _folder(self, progress: TrainingProgress) -> Path:
        return (self.checkpoint_path / self.experiment_id
                / checkpoint_folder_name(self.experiment_id, progress))

    def _suffix(self) -> str:
        return f"_{self.partition}" if self.partition else ""

    @torch.no_grad()
    def save_checkpoint(self, app_state: AppState, progress: TrainingProgress):
        folder = self._folder(progress)
        folder.mkdir(parents=True, exist_ok=True)
        world = self.dp_world if self.dp_world is not None else             (dist.get_world_size() if is_dist() else 1)
        rank = self.dp_rank if self.dp_rank is not None else self.global_rank

        if self.write_enabled:
            shards = {k: v.detach().to("cpu")
                      for k, v in app_state.shard_state().items()}
            torch.save(shards, folder / f"shards{self._suffix()}_rank_{rank}.pt")

        if rank == 0 and self.write_enabled:
            meta = {
                "world_size": world,
                "shard_layout": app_state.shard_layout(),
                "replicated_state": _to_jsonable(app_state.replicated_state()),
                "training_progress": {
                    "num_seen_steps": progress.num_seen_steps_total,
                    "num_seen_tokens": progress.num_seen_tokens_total,
                    "num_target_steps": progress.num_target_steps,
                    "num_target_tokens": progress.num_target_tokens,
                },
            }
            with open(folder / f"meta{self._suffix()}.json", "w",
                      encoding="utf-8") as f:
                json.dump(meta, f, indent=1)
        if self.global_rank == 0:
            info = {"checkpoint_folder_path": str(folder.absolute())}
            with open(folder.parent / "last_checkpoint_info.json", "w",
                      encoding="utf-8") as f:
                json.dump(info, f)
        if is_dist():
            # all ranks leave together so trainer throughput windows stay honest
            dist.barrier()

    def delete_checkpoint(self, progress: TrainingProgress):
        if self.global_rank != 0:
            return
        folder = self._folder(progress)
        if folder.exists():
            shutil.rmtree(folder)


def _to_jsonable(obj):
    if isinstance(obj, dict):
        return {k: _to_jsonable(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_to_jsonable(v) for v in obj]
    if isinstance(obj, torch.Tensor):
        return obj.tolist()
    return obj


class CheckpointSaving:
    """Strategy + execution combiner (reference:
    checkpointing/checkpoint_saving.py:8-55)."""

    def __init__(self, checkpoint_saving_strategy: CheckpointSavingStrategy,
                 checkpoint_saving_execution: ShardedCheckpointSaving):
        self.strategy = checkpoint_saving_strategy
        self.execution = checkpoint_saving_execution

    def save_checkpoint_and_free_memory(self, training_progress: TrainingProgress,
                                        app_state: AppState):
        instruction = self.strategy.get_checkpoint_instruction(training_progress)
        if instruction.save_current:
            self.execution.save_checkpoint(app_state, training_progress)
        for tp in instruction.checkpoints_to_delete:
            self.execution.delete_checkpoint(tp)
