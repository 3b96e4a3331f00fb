"""Application state (model + optimizer + lr scheduler) for checkpointing.

Capability parity with the reference's ``AppState`` Stateful wrapper
(reference: src/modalities/checkpointing/stateful/app_state.py:27-119) built
for the MI355X-native sharding engine instead of torch DCP/DTensor: the
model/optimizer state is already rank-local flat fp32 shards
(`XGMIShardedModel` master shards, `ShardedAdamW` exp_avg/exp_avg_sq shards),
so a checkpoint is simply each rank's shard bundle plus small replicated
metadata — no gather, no DTensor resharding machinery at save time.
"""

from typing import Any, Optional

import torch

from modalities_amd.parallel.fsdp import XGMIShardedModel


class AppState:
    """Holds the three stateful training entities. ``shard_state`` /
    ``load_shard_state`` deal with rank-local sharded tensors;
    ``replicated_state`` / ``load_replicated_state`` with small state that is
    identical on every rank (scheduler, step counters)."""

    def __init__(self, model, optimizer, lr_scheduler: Optional[Any] = None):
        self.model = model
        self.optimizer = optimizer
        self.lr_scheduler = lr_scheduler
        self._is_loaded = False

    # -- sharded ---------------------------------------------------------

    def shard_state(self) -> dict[str, torch.Tensor]:
        """Rank-local flat shards: model master weights + optimizer moments."""
        if not isinstance(self.model, XGMIShardedModel):
            raise TypeError("AppState.shard_state requires an XGMIShardedModel")
        out: dict[str, torch.Tensor] = {}
        for u in self.model.units:
            out[f"model.{u.name}.master_shard"] = u.master_shard
            st = self.optimizer.state.get(u.master_shard, {})
            if "exp_avg" in st:
                out[f"optim.{u.name}.exp_avg"] = st["exp_avg"]
                out[f"optim.{u.name}.exp_avg_sq"] = st["exp_avg_sq"]
        return out

    def optim_steps(self) -> dict[str, int]:
        steps = {}
        for u in self.model.units:
            st = self.optimizer.state.get(u.master_shard, {})
            steps[u.name] = int(st.get("step", 0))
        return steps

    @torch.no_grad()
    def load_shard_state(self, shards: dict[str, torch.Tensor],
                         optim_steps: dict[str, int]):
        for u in self.model.units:
            u.master_shard.copy_(shards[f"model.{u.name}.master_shard"])
            st = self.optimizer.state[u.master_shard]
            key = f"optim.{u.name}.exp_avg"
            if key in shards and "exp_avg" in st:
                st["exp_avg"].copy_(shards[key])
                st["exp_avg_sq"].copy_(shards[f"optim.{u.name}.exp_avg_sq"])
            st["step"] = optim_steps.get(u.name, 0)
            u.publish_master()
        self._is_loaded = True

    # -- replicated -------------------------------------------------------

    def replicated_state(self) -> dict:
        out = {"optim_steps": self.optim_steps(),
               "param_group_lrs": [g["lr"] for g in self.optimizer.param_groups]}
        if self.lr_scheduler is not None:
            out["lr_scheduler"] = self.lr_scheduler.state_dict()
        return out

    def load_replicated_state(self, state: dict):
        if self.lr_scheduler is not None and "lr_scheduler" in state:
            self.lr_scheduler.load_state_dict(state["lr_scheduler"])
        # torch schedulers' load_state_dict does NOT push the restored lr back
        # into the optimizer; without this the first resumed step would run at
        # the freshly-built lr (warmstart equivalence test catches it).
        for g, lr in zip(self.optimizer.param_groups,
                         state.get("param_group_lrs", [])):
            g["lr"] = lr

    # -- layout metadata (for cross-world-size resharding) ----------------

    def shard_layout(self) -> dict:
        layout = {}
        for u in self.model.units:
            logical = (u.offsets[-1] + u.numels[-1]) if u.numels else 0
            layout[u.name] = {"logical_numel": logical,
                              "total_numel": u.total_numel,
                              "shard_numel": u.shard_numel}
        return layout
