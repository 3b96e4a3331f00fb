"""Sharded checkpoint loading with cross-world-size resharding.

Capability parity with the reference's DCP load path (reference:
src/modalities/checkpointing/fsdp/fsdp_checkpoint_loading.py:104-133), native
to the flat-shard engine: because every unit's parameters live in ONE
deterministic flat fp32 vector, resharding to a different world size is pure
index arithmetic — each rank streams only the saved rank-files that overlap
its new shard range (one saved file in memory at a time), no global gather.

Padding note: a unit's flat length is padded to ``world*64``, so the padded
total differs between world sizes; all slicing here happens in LOGICAL
coordinates [0, logical_numel) recorded in ``meta.json``.
"""

import json
from pathlib import Path

import torch

from modalities_amd.checkpointing.app_state import AppState


def read_last_checkpoint_info(experiment_checkpoint_root: Path) -> Path:
    with open(Path(experiment_checkpoint_root) / "last_checkpoint_info.json",
              encoding="utf-8") as f:
        return Path(json.load(f)["checkpoint_folder_path"])


def read_checkpoint_meta(folder: Path, partition: str = "") -> dict:
    sfx = f"_{partition}" if partition else ""
    with open(Path(folder) / f"meta{sfx}.json", encoding="utf-8") as f:
        return json.load(f)


class ShardedCheckpointLoading:
    def __init__(self, global_rank: int, partition: str = ""):
        self.global_rank = global_rank
        self.partition = partition  # PP/TP model partition (see saving.py)

    @torch.no_grad()
    def load_checkpoint_(self, app_state: AppState, folder: Path) -> dict:
        """Load (in-place) a checkpoint folder into ``app_state``; reshard if
        the saved world size differs. Returns the saved training-progress
        metadata dict."""
        folder = Path(folder)
        meta = read_checkpoint_meta(folder, self.partition)
        saved_world = meta["world_size"]
        saved_layout = meta["shard_layout"]

        model = app_state.model
        new_world, new_rank = model.world, model.rank

        # model/checkpoint compatibility: every engine unit must exist in
        # the saved layout with the same LOGICAL length (resharding changes
        # only the per-rank split; a different model is a hard error).
        for u in model.units:
            if u.name not in saved_layout:
                raise ValueError(
                    f"Checkpoint {folder} does not match the model: unit "
                    f"{u.name!r} missing from the saved shard layout "
                    f"(saved units: {sorted(saved_layout)})")
            logical_here = (u.offsets[-1] + u.numels[-1]) if u.numels else 0
            if saved_layout[u.name]["logical_numel"] != logical_here:
                raise ValueError(
                    f"Checkpoint {folder} does not match the model: unit "
                    f"{u.name!r} has {saved_layout[u.name]['logical_numel']} "
                    f"saved parameters vs {logical_here} in the model "
                    f"(different architecture or parallel layout)")

        # target CPU staging buffers for this rank's new shards
        targets: dict[str, torch.Tensor] = {}
        for u in model.units:
            targets[f"model.{u.name}.master_shard"] = torch.zeros(
                u.shard_numel, dtype=torch.float32)
            targets[f"optim.{u.name}.exp_avg"] = torch.zeros(
                u.shard_numel, dtype=torch.float32)
            targets[f"optim.{u.name}.exp_avg_sq"] = torch.zeros(
                u.shard_numel, dtype=torch.float32)

        for saved_rank in range(saved_world):
            if not self._rank_file_overlaps(model, saved_layout, saved_rank,
                                            saved_world, new_rank):
                continue
            sfx = f"_{self.partition}" if self.partition else ""
            shards = torch.load(folder / f"shards{sfx}_rank_{saved_rank}.pt",
                                map_location="cpu", weights_only=True)
            for u in model.units:
                lay = saved_layout[u.name]
                logical = lay["logical_numel"]
                ss = lay["shard_numel"]
                s_lo, s_hi = saved_rank * ss, min((saved_rank + 1) * ss, logical)
                n_lo = new_rank * u.shard_numel
                n_hi = min((new_rank + 1) * u.shard_numel, logical)
                lo, hi = max(s_lo, n_lo), min(s_hi, n_hi)
                if lo >= hi:
                    continue
                for prefix in ("model", "optim"):
                    for suffix in (("master_shard",) if prefix == "model"
                                   else ("exp_avg", "exp_avg_sq")):
                        key = f"{prefix}.{u.name}.{suffix}"
                        if key not in shards:
                            continue
                        src = shards[key][lo - s_lo:hi - s_lo]
                        targets[key][lo - n_lo:hi - n_lo].copy_(src)
            del shards

        optim_steps = meta["replicated_state"].get("optim_steps", {})
        app_state.load_shard_state(
            {k: v for k, v in targets.items()}, optim_steps)
        app_state.load_replicated_state(meta["replicated_state"])
        return meta["training_progress"]

    @staticmethod
    def _rank_file_overlaps(model, saved_layout, saved_rank, saved_world,
                            new_rank) -> bool:
        for u in model.units:
            lay = saved_layout[u.name]
            ss = lay["shard_numel"]
            logical = lay["logical_numel"]
            s_lo, s_hi = saved_rank * ss, min((saved_rank + 1) * ss, logical)
            n_lo = new_rank * u.shard_numel
            n_hi = min((new_rank + 1) * u.shard_numel, logical)
            if max(s_lo, n_lo) < min(s_hi, n_hi):
                return True
        return False


@torch.no_grad()
def load_full_model_state_from_checkpoint(folder: Path, model) -> None:
    """Single-process path: reassemble the flat fp32 master weights of a
    sharded checkpoint and load them into a PLAIN (unsharded) module in
    place. Used by text generation / HF export (reference analog: the
    "torch" checkpoint_loading variant).

    The flat layout is reconstructed exactly as XGMIShardedModel built it
    (same unit grouping): we wrap a throwaway sharded view of `model` at
    world size 1 to recover offsets, then fill its master shards from the
    saved rank files and publish back into the module parameters."""
    from modalities_amd.checkpointing.app_state import AppState
    from modalities_amd.parallel.fsdp import XGMIShardedModel

    folder = Path(folder)
    meta = read_checkpoint_meta(folder)
    device = next(model.parameters()).device
    sharded = XGMIShardedModel.from_transformer(
        model, device, rank=0, world_size=1,
        param_dtype=next(model.parameters()).dtype
        if next(model.parameters()).dtype in (torch.float32, torch.bfloat16)
        else torch.float32)
    app = AppState(sharded, _NullOptimizer(sharded))
    ShardedCheckpointLoading(0).load_checkpoint_(app, folder)
    # materialize full params back onto the module
    sd = sharded.gather_full_state_dict()
    for name, p in model.named_parameters():
        p.data = sd[name].to(p.dtype)
    return None


class _NullOptimizer:
    """Minimal stand-in so AppState can load model shards without a real
    optimizer (moments in the checkpoint are skipped)."""

    def __init__(self, sharded):
        self.state = {u.master_shard: {} for u in sharded.units}
        self.param_groups = []
