"""The component table: every (component_key, variant_key) the framework
exposes to YAML configs (capability parity with reference
src/modalities/registry/components.py:187-531 — same two-level namespace so
reference-shaped configs carry over; variants that were CUDA-dependency-
specific map to the MI355X-native equivalents)."""

from pathlib import Path
from typing import Optional

import torch
from torch.utils.data import BatchSampler

from modalities_amd.checkpointing import (AppState,
                                          CheckpointSaving,
                                          SaveEveryKStepsCheckpointingStrategy,
                                          SaveKMostRecentCheckpointsStrategy,
                                          ShardedCheckpointLoading,
                                          ShardedCheckpointSaving)
from modalities_amd.dataloader.dataloader import (GPT2LLMCollateFn, LLMDataLoader,
                                                  LossMaskingCollateFnWrapper)
from modalities_amd.dataloader.dataset import (CombinedDataset, DummyDataset,
                                               MemMapDataset,
                                               PackedMemMapDatasetContinuous,
                                               PackedMemMapDatasetMegatron,
                                               SyntheticLMDataset)
from modalities_amd.dataloader.samplers import ResumableDistributedSampler
from modalities_amd.loss_functions import CLMCrossEntropyLoss, NCELoss
from modalities_amd.logging_broker.subscribers import (DummyProgressSubscriber,
                                                       DummyResultSubscriber,
                                                       ResultsToDiscSubscriber,
                                                       RichProgressSubscriber,
                                                       RichResultSubscriber, WandBEvaluationResultSubscriber)
from modalities_amd.models.coca import CoCa
from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.models.vision_transformer import VisionTransformer
from modalities_amd.models.model_factory import ModelFactory
from modalities_amd.nn.model_initialization import (Llama3LikeInitialization,
                                                    get_composed_model_initializer)
from modalities_amd.optimizers.lr_schedulers import (DummyLRScheduler,
                                                     get_constant_lr,
                                                     get_cosine_annealing,
                                                     get_linear_lr,
                                                     get_linear_warmup_cosine_annealing,
                                                     get_onecycle_lr, get_step_lr)
from modalities_amd.optimizers.optimizer_factory import get_adam_w
from modalities_amd.parallel.mesh import get_device_mesh
from modalities_amd.parallel.cp import get_gpt2_context_parallel_model
from modalities_amd.parallel.pp import (
    get_staged_pipeline_schedule as _pp_get_staged_pipeline_schedule,
    get_stage_from_schedule as _pp_get_stage_from_schedule)
from modalities_amd.parallel.tp import get_gpt2_tensor_parallelized_model
from modalities_amd.registry.registry import ComponentEntity, Registry
from modalities_amd.tokenization.tokenizer_wrapper import (CharTokenizer,
                                                           PreTrainedHFTokenizer,
                                                           PreTrainedSPTokenizer)
from modalities_amd.training.gradient_clipping import (DummyGradientClipper,
                                                       GradientClipper)
from modalities_amd.utils.debug_components import get_debugging_enriched_model
from modalities_amd.utils.mfu import GPT2MFUCalculator, detect_device_peak_flops
from modalities_amd.utils.profilers import get_profiler
from modalities_amd.utils.number_conversion import NumberConversion


# ---- adapter factories ------------------------------------------------------

def get_gpt2_model(**kwargs) -> GPT2LLM:
    return GPT2LLM(GPT2LLMConfig(**kwargs))


def get_checkpointed_model(checkpoint_path: Path, model: torch.nn.Module):
    """Load a full (gathered, fp32) state dict saved via
    `gather_full_state_dict` / torch.save into a freshly built model
    (text-generation / conversion path)."""
    sd = torch.load(Path(checkpoint_path), map_location="cpu", weights_only=True)
    model.load_state_dict(sd)
    return model


def get_raw_app_state(model, optimizer, lr_scheduler=None) -> AppState:
    return AppState(model, optimizer, lr_scheduler)


def _mesh_partition(device_mesh):
    """(partition, shard_rank, shard_world, write_enabled) for saving under
    parallelism composition. Each pp/tp coordinate is its own model
    partition with its own shard layout; the persisted layout inside a
    partition is the DP-SHARD group's (replicate/CP peers hold identical
    parameters, so only their rank-0 peer writes)."""
    if device_mesh is None:
        return "", None, None, True
    from modalities_amd.parallel.mesh import ParallelismDegrees
    pp = device_mesh.dims[ParallelismDegrees.PP]
    tp = device_mesh.dims[ParallelismDegrees.TP]
    rep = device_mesh.dims[ParallelismDegrees.DP_REPLICATE]
    cp = device_mesh.dims[ParallelismDegrees.CP]
    shard = device_mesh.dims[ParallelismDegrees.DP_SHARD]
    write = rep.rank == 0 and cp.rank == 0
    if pp.size == 1 and tp.size == 1 and rep.size == 1 and cp.size == 1:
        return "", None, None, True
    part = ((f"pp{pp.rank}" if pp.size > 1 else "")
            + (f"tp{tp.rank}" if tp.size > 1 else ""))
    return part, shard.rank, shard.size, write


def get_warmstart_app_state(model, optimizer, checkpoint_folder_path: Path,
                            lr_scheduler=None, device_mesh=None) -> AppState:
    import torch.distributed as dist
    app_state = AppState(model, optimizer, lr_scheduler)
    rank = dist.get_rank() if dist.is_initialized() else 0
    part, _, _, _ = _mesh_partition(device_mesh)
    ShardedCheckpointLoading(rank, partition=part).load_checkpoint_(
        app_state, Path(checkpoint_folder_path))
    return app_state


def get_checkpoint_saving(checkpoint_saving_strategy,
                          checkpoint_saving_execution) -> CheckpointSaving:
    return CheckpointSaving(checkpoint_saving_strategy, checkpoint_saving_execution)


def get_sharded_checkpoint_saving_execution(checkpoint_path: Path,
                                            experiment_id: str,
                                            device_mesh=None
                                            ) -> ShardedCheckpointSaving:
    import torch.distributed as dist
    rank = dist.get_rank() if dist.is_initialized() else 0
    part, dp_rank, dp_world, write = _mesh_partition(device_mesh)
    return ShardedCheckpointSaving(Path(checkpoint_path), experiment_id, rank,
                                   partition=part, dp_rank=dp_rank,
                                   dp_world=dp_world, write_enabled=write)


def get_resumable_sampler(dataset, epoch: int = 0, shuffle: bool = False,
                          seed: int = 0, drop_last: bool = False,
                          skip_num_global_samples: int = 0,
                          rank: Optional[int] = None,
                          num_replicas: Optional[int] = None,
                          device_mesh=None) -> ResumableDistributedSampler:
    """Rank/degree default to the device mesh's DP coordinates (reference
    sampler_factory.py:28-53)."""
    if device_mesh is not None:
        rank = device_mesh.dp_rank if rank is None else rank
        num_replicas = device_mesh.dp_degree if num_replicas is None else num_replicas
    import torch.distributed as dist
    if num_replicas is None:
        num_replicas = dist.get_world_size() if dist.is_initialized() else 1
    if rank is None:
        rank = dist.get_rank() if dist.is_initialized() else 0
    return ResumableDistributedSampler(
        dataset=dataset, rank=rank, num_replicas=num_replicas, epoch=epoch,
        shuffle=shuffle, seed=seed, drop_last=drop_last,
        skip_num_global_samples=skip_num_global_samples)


def get_gpt2_mfu_calculator(n_layer: int, sequence_length: int, n_embd: int,
                            world_size: int, num_params: int,
                            peak_flops_per_gpu: Optional[float] = None
                            ) -> GPT2MFUCalculator:
    if peak_flops_per_gpu is None:
        peak_flops_per_gpu = detect_device_peak_flops()
    return GPT2MFUCalculator(n_layer, sequence_length, n_embd, world_size,
                             num_params, peak_flops_per_gpu)


# ---- the table --------------------------------------------------------------

COMPONENTS: list[ComponentEntity] = [
    # models
    ComponentEntity("model", "gpt2", get_gpt2_model, GPT2LLMConfig),
    ComponentEntity("model", "checkpointed", get_checkpointed_model, None),
    ComponentEntity("model", "coca", CoCa, None),
    ComponentEntity("model", "vision_transformer", VisionTransformer, None),
    ComponentEntity("sharded_model", "xgmi_fsdp", ModelFactory.get_sharded_model, None),
    ComponentEntity("wrapped_model", "fsdp2", ModelFactory.get_sharded_model, None),
    ComponentEntity("initialized_model", "default",
                    ModelFactory.get_weight_initialized_model, None),
    ComponentEntity("activation_checkpointed_model", "default",
                    ModelFactory.get_activation_checkpointed_model_, None),
    ComponentEntity("compiled_model", "default", ModelFactory.get_compiled_model, None),
    # init
    ComponentEntity("model_initialization", "composed",
                    get_composed_model_initializer, None),
    ComponentEntity("model_initialization", "llama3",
                    Llama3LikeInitialization, None),
    # optimizers / schedulers
    ComponentEntity("optimizer", "adam_w", get_adam_w, None),
    ComponentEntity("scheduler", "dummy_lr", DummyLRScheduler, None),
    ComponentEntity("scheduler", "linear_warmup_cosine_annealing",
                    get_linear_warmup_cosine_annealing, None),
    ComponentEntity("scheduler", "cosine_annealing_lr", get_cosine_annealing, None),
    ComponentEntity("scheduler", "constant_lr", get_constant_lr, None),
    ComponentEntity("scheduler", "step_lr", get_step_lr, None),
    ComponentEntity("scheduler", "linear_lr", get_linear_lr, None),
    ComponentEntity("scheduler", "onecycle_lr", get_onecycle_lr, None),
    # loss
    ComponentEntity("loss", "clm_cross_entropy_loss", CLMCrossEntropyLoss, None),
    ComponentEntity("loss", "nce_loss", NCELoss, None),
    # data
    ComponentEntity("dataset", "packed_mem_map_dataset_continuous",
                    PackedMemMapDatasetContinuous, None),
    ComponentEntity("dataset", "packed_mem_map_dataset_megatron",
                    PackedMemMapDatasetMegatron, None),
    ComponentEntity("dataset", "mem_map_dataset", MemMapDataset, None),
    ComponentEntity("dataset", "dummy_dataset", DummyDataset, None),
    ComponentEntity("dataset", "combined", CombinedDataset, None),
    ComponentEntity("dataset", "synthetic_lm", SyntheticLMDataset, None),
    ComponentEntity("sampler", "resumable_distributed_sampler",
                    get_resumable_sampler, None),
    ComponentEntity("batch_sampler", "default", BatchSampler, None),
    ComponentEntity("collate_fn", "gpt_2_llm_collator", GPT2LLMCollateFn, None),
    ComponentEntity("collate_fn", "mask_loss_collator_wrapper",
                    LossMaskingCollateFnWrapper, None),
    ComponentEntity("data_loader", "default", LLMDataLoader, None),
    # tokenizers
    ComponentEntity("tokenizer", "pretrained_hf_tokenizer", PreTrainedHFTokenizer, None),
    ComponentEntity("tokenizer", "pretrained_sp_tokenizer", PreTrainedSPTokenizer, None),
    ComponentEntity("tokenizer", "char", CharTokenizer, None),
    # checkpointing
    ComponentEntity("checkpoint_saving", "default", get_checkpoint_saving, None),
    ComponentEntity("checkpoint_saving_strategy", "save_every_k_steps_checkpointing_strategy",
                    SaveEveryKStepsCheckpointingStrategy, None),
    ComponentEntity("checkpoint_saving_strategy", "save_k_most_recent_checkpoints_strategy",
                    SaveKMostRecentCheckpointsStrategy, None),
    ComponentEntity("checkpoint_saving_execution", "sharded",
                    get_sharded_checkpoint_saving_execution, None),
    ComponentEntity("checkpoint_saving_execution", "dcp",
                    get_sharded_checkpoint_saving_execution, None),
    ComponentEntity("app_state", "raw", get_raw_app_state, None),
    ComponentEntity("app_state", "dcp", get_warmstart_app_state, None),
    ComponentEntity("app_state", "sharded_warmstart", get_warmstart_app_state, None),
    # parallelism
    ComponentEntity("device_mesh", "default", get_device_mesh, None),
    ComponentEntity("tensor_parallelized_model", "gpt2_tp",
                    get_gpt2_tensor_parallelized_model, None),
    ComponentEntity("context_parallelized_model", "gpt2_cp",
                    get_gpt2_context_parallel_model, None),
    ComponentEntity("pp_schedule", "staged",
                    _pp_get_staged_pipeline_schedule, None),
    ComponentEntity("pipelined_model", "selector",
                    _pp_get_stage_from_schedule, None),
    # training aux
    ComponentEntity("gradient_clipper", "fsdp2", GradientClipper, None),
    ComponentEntity("gradient_clipper", "default", GradientClipper, None),
    ComponentEntity("gradient_clipper", "dummy", DummyGradientClipper, None),
    ComponentEntity("mfu_calculator", "gpt2", get_gpt2_mfu_calculator, None),
    # observability / debugging / profiling
    ComponentEntity("profiler", "default", get_profiler, None),
    ComponentEntity("debugging_enriched_model", "default",
                    get_debugging_enriched_model, None),
    ComponentEntity("progress_subscriber", "dummy", DummyProgressSubscriber, None),
    ComponentEntity("progress_subscriber", "rich", RichProgressSubscriber, None),
    ComponentEntity("results_subscriber", "dummy", DummyResultSubscriber, None),
    ComponentEntity("results_subscriber", "rich", RichResultSubscriber, None),
    ComponentEntity("results_subscriber", "save_to_disc", ResultsToDiscSubscriber, None),
    ComponentEntity("results_subscriber", "wandb", WandBEvaluationResultSubscriber, None),
    # number conversion (config-time arithmetic components)
    ComponentEntity("number_conversion", "local_num_batches_from_num_samples",
                    NumberConversion.get_local_num_batches_from_num_samples, None),
    ComponentEntity("number_conversion", "local_num_batches_from_num_tokens",
                    NumberConversion.get_local_num_batches_from_num_tokens, None),
    ComponentEntity("number_conversion", "num_samples_from_num_tokens",
                    NumberConversion.get_num_samples_from_num_tokens, None),
    ComponentEntity("number_conversion", "num_steps_from_num_samples",
                    NumberConversion.get_num_steps_from_num_samples, None),
    ComponentEntity("number_conversion", "num_samples_from_num_steps",
                    NumberConversion.get_num_samples_from_num_steps, None),
    ComponentEntity("number_conversion", "num_steps_from_num_tokens",
                    NumberConversion.get_num_steps_from_num_tokens, None),
    ComponentEntity("number_conversion", "num_tokens_from_num_steps",
                    NumberConversion.get_num_tokens_from_num_steps, None),
    ComponentEntity("number_conversion", "last_step_from_checkpoint_path",
                    NumberConversion.get_last_step_from_checkpoint_path, None),
    ComponentEntity("number_conversion", "num_seen_steps_from_checkpoint_path",
                    NumberConversion.get_num_seen_steps_from_checkpoint_path, None),
    ComponentEntity("number_conversion", "global_num_seen_tokens_from_checkpoint_path",
                    NumberConversion.get_global_num_seen_tokens_from_checkpoint_path, None),
    ComponentEntity("number_conversion", "global_num_target_tokens_from_checkpoint_path",
                    NumberConversion.get_global_num_target_tokens_from_checkpoint_path,
                    None),
    ComponentEntity("number_conversion", "num_target_steps_from_checkpoint_path",
                    NumberConversion.get_num_target_steps_from_checkpoint_path, None),
    ComponentEntity("number_conversion", "num_tokens_from_packed_mem_map_dataset_continuous",
                    NumberConversion.get_num_tokens_from_packed_mem_map_dataset_continuous,
                    None),
    ComponentEntity("number_conversion", "num_steps_from_raw_dataset_index",
                    NumberConversion.get_num_steps_from_raw_dataset_index, None),
]


def get_default_registry() -> Registry:
    return Registry(COMPONENTS)
