"""Model wrapping factories (capability parity with reference
src/modalities/models/model_factory.py: FSDP2 wrap -> get_sharded_model on
the XGMI engine; weight init; activation checkpointing; torch.compile;
meta-device init + materialization)."""

from typing import Optional

import torch
import torch.nn as nn

from modalities_amd.nn.model_initialization import ModelInitializationIF
from modalities_amd.parallel.fsdp import XGMIShardedModel
from modalities_amd.parallel.mesh import DeviceMesh, ParallelismDegrees
from modalities_amd.running_env import device_for_rank
from modalities_amd.training.activation_checkpointing import (
    ActivationCheckpointingVariant, apply_activation_checkpointing_)


class ModelFactory:
    @staticmethod
    def get_weight_initialized_model(model: nn.Module,
                                     model_initializer: ModelInitializationIF
                                     ) -> nn.Module:
        """Materialize (if meta) then run the initializer (reference
        model_factory.py:248-281 to_empty -> reset_parameters -> init)."""
        if any(p.is_meta for p in model.parameters()):
            model = model.to_empty(device=device_for_rank())
            for m in model.modules():
                if hasattr(m, "reset_parameters"):
                    m.reset_parameters()
        model_initializer.initialize_in_place(model)
        return model

    @staticmethod
    def get_activation_checkpointed_model_(model: nn.Module,
                                           activation_checkpointing_variant:
                                           ActivationCheckpointingVariant,
                                           layers_fqn: str = "blocks",
                                           every_k_layers: int = 1) -> nn.Module:
        v = activation_checkpointing_variant
        if isinstance(v, str):  # YAML gives enum NAME or value; both accepted
            try:
                v = ActivationCheckpointingVariant(v.lower())
            except ValueError:
                v = ActivationCheckpointingVariant[v.upper()]
        apply_activation_checkpointing_(model, v, layers_fqn, every_k_layers)
        return model

    @staticmethod
    def get_compiled_model(model: nn.Module, block_names: Optional[list[str]] = None,
                           fullgraph: bool = False, debug: bool = False) -> nn.Module:
        """Per-block torch.compile (reference model_factory.py:353-408).
        Optional on ROCm — the HIP ops are the primary speed path."""
        if not hasattr(torch, "compile"):
            return model
        if block_names:
            for name, module in model.named_modules():
                if type(module).__name__ in block_names:
                    module.compile(fullgraph=fullgraph)
        else:
            model.compile(fullgraph=fullgraph)
        return model

    @staticmethod
    def get_sharded_model(model: nn.Module,
                          device_mesh: Optional[DeviceMesh] = None,
                          blocks_per_unit: int = 1,
                          reshard_after_forward: bool = False,
                          param_dtype: str = "bf16",
                          device: Optional[torch.device] = None
                          ) -> XGMIShardedModel:
        """The FSDP2-equivalent wrap: flat-shard units over the dp_shard
        group, optional replicate group for HSDP (reference
        model_factory.py:168-246)."""
        device = device or device_for_rank()
        dtype = {"bf16": torch.bfloat16, "fp32": torch.float32,
                 "fp16": torch.float16}[param_dtype]
        if device.type != "cuda" and dtype != torch.float32:
            dtype = torch.float32  # CPU tests run fp32
        shard_group = replicate_group = None
        rank = world = None
        if device_mesh is not None:
            shard_dim = device_mesh.dims[ParallelismDegrees.DP_SHARD]
            shard_group = shard_dim.group
            rank, world = shard_dim.rank, shard_dim.size
            rep_dim = device_mesh.dims[ParallelismDegrees.DP_REPLICATE]
            if rep_dim.size > 1:
                replicate_group = rep_dim.group
        return XGMIShardedModel.from_transformer(
            model, device, process_group=shard_group, rank=rank,
            world_size=world, blocks_per_unit=blocks_per_unit,
            reshard_after_forward=reshard_after_forward, param_dtype=dtype,
            replicate_group=replicate_group)
