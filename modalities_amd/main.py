"""The orchestrator (capability parity with reference
src/modalities/main.py:36-274): load config, build the component DAG, wire
publishers, persist the resolved config into the experiment folder, build
Trainer/Evaluator/Gym, print the training report, run."""

import json
import shutil
from pathlib import Path
from typing import Optional, Type

import torch.distributed as dist
import yaml

from modalities_amd.config.component_factory import ComponentFactory
from modalities_amd.config.instantiation_models import (
    TrainingComponentsInstantiationModel, TrainingReportGenerator)
from modalities_amd.config.yaml_loader import load_app_config_dict
from modalities_amd.logging_broker.broker import (MessageBroker, MessagePublisher,
                                                  MessageTypes)
from modalities_amd.registry.components import get_default_registry
from modalities_amd.registry.registry import ComponentEntity
from modalities_amd.running_env import global_rank, is_dist
from modalities_amd.training.evaluator import Evaluator
from modalities_amd.training.gym import Gym
from modalities_amd.training.trainer import Trainer
from modalities_amd.utils.experiment_id import get_synced_experiment_id_of_run


def get_logging_publishers(progress_subscriber, results_subscriber):
    """Wire the pub/sub broker (reference main.py:234-274)."""
    broker = MessageBroker()
    if results_subscriber is not None:
        broker.add_subscriber(MessageTypes.EVALUATION_RESULT, results_subscriber)
    if progress_subscriber is not None:
        broker.add_subscriber(MessageTypes.BATCH_PROGRESS_UPDATE, progress_subscriber)
    progress_publisher = MessagePublisher(
        broker, global_rank=global_rank(), local_rank=global_rank())
    results_publisher = MessagePublisher(
        broker, global_rank=global_rank(), local_rank=global_rank())
    return progress_publisher, results_publisher, broker


class Main:
    def __init__(self, config_path: Path,
                 additional_resolver_funs: Optional[dict] = None,
                 experiment_id: Optional[str] = None):
        self.config_path = Path(config_path)
        if experiment_id is None:
            experiment_id = get_synced_experiment_id_of_run(self.config_path)
        self.experiment_id = experiment_id
        self.config_dict = load_app_config_dict(
            self.config_path, experiment_id=experiment_id,
            additional_resolver_funs=additional_resolver_funs)
        self.registry = get_default_registry()
        self.component_factory = ComponentFactory(self.registry)

    def add_custom_component(self, component_key: str, variant_key: str,
                             custom_component: Type, custom_config: Optional[Type] = None
                             ) -> None:
        """Library use case: register user components before build
        (reference main.py:61-81)."""
        self.registry.add_entity(ComponentEntity(component_key, variant_key,
                                                 custom_component, custom_config))

    def build_components(self, components_model_type: Type = TrainingComponentsInstantiationModel):
        return self.component_factory.build_components(self.config_dict,
                                                       components_model_type)

    def run(self, components: TrainingComponentsInstantiationModel) -> None:
        settings = components.settings
        rank = settings.cuda_env.global_rank

        # experiment folder + resolved-config copy (reference main.py:117-143)
        if settings.paths.checkpoint_saving_path is not None:
            exp_dir = Path(settings.paths.checkpoint_saving_path) / self.experiment_id
            if rank == 0:
                exp_dir.mkdir(parents=True, exist_ok=True)
                shutil.copy2(self.config_path, exp_dir / self.config_path.name)
                with open(exp_dir / (self.config_path.name + ".resolved"), "w") as f:
                    yaml.safe_dump(_jsonable(self.config_dict), f, sort_keys=False)
            if is_dist():
                dist.barrier()

        progress_publisher, results_publisher, _ = get_logging_publishers(
            components.progress_subscriber, components.evaluation_subscriber)

        sp = settings.step_profile
        # tokens/step scale with the DATA-parallel degree, not world_size
        # (reference main.py:168-173): tp/cp/pp ranks see the same tokens.
        dp_degree = sp.dp_degree
        if dp_degree is None and components.device_mesh is not None:
            dp_degree = components.device_mesh.dp_degree
        if dp_degree is None:
            dp_degree = settings.cuda_env.world_size
        global_num_tokens_per_train_step = (
            sp.gradient_accumulation_steps * sp.local_train_micro_batch_size
            * sp.sequence_length * dp_degree)

        import torch
        device = (torch.device("cuda", settings.cuda_env.local_rank)
                  if torch.cuda.is_available() else torch.device("cpu"))

        trainer = Trainer(
            global_rank=rank,
            progress_publisher=progress_publisher,
            evaluation_result_publisher=results_publisher,
            gradient_acc_steps=sp.gradient_accumulation_steps,
            global_num_tokens_per_train_step=global_num_tokens_per_train_step,
            num_seen_train_steps=settings.training_progress.num_seen_steps,
            global_num_seen_tokens=settings.training_progress.global_num_seen_tokens,
            num_target_steps=settings.training_target.num_target_steps,
            num_target_tokens=settings.training_target.num_target_tokens,
            gradient_clipper=components.gradient_clipper,
            mfu_calculator=components.mfu_calculator,
            evaluation_interval_in_steps=settings.intervals.evaluation_interval_in_steps,
            checkpointing_interval_in_steps=settings.intervals.checkpointing_interval_in_steps,
            training_log_interval_in_steps=settings.intervals.training_log_interval_in_steps,
            device=device,
            pp_schedule=components.pp_schedule)
        evaluator = Evaluator(progress_publisher, results_publisher, device=device,
                              pp_schedule=components.pp_schedule)
        gym = Gym(trainer, evaluator, components.loss_fn,
                  num_ranks=settings.cuda_env.world_size)

        num_params = sum(p.numel() for p in components.wrapped_model.parameters())
        if hasattr(components.wrapped_model, "units"):
            num_params = sum(u.total_numel for u in components.wrapped_model.units)
        report = TrainingReportGenerator(settings, num_params).get_report()
        if rank == 0:
            print(report, flush=True)
        if components.evaluation_subscriber is not None:
            components.evaluation_subscriber.consume_dict(
                {"num_parameters": num_params})

        gym.run(model=components.wrapped_model,
                optimizer=components.optimizer,
                scheduler=components.scheduler,
                train_data_loader=components.train_dataloader,
                evaluation_data_loaders=components.eval_dataloaders,
                checkpoint_saving=components.checkpoint_saving,
                app_state=components.app_state)


def _jsonable(obj):
    if isinstance(obj, dict):
        return {k: _jsonable(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_jsonable(v) for v in obj]
    if isinstance(obj, Path):
        return str(obj)
    return obj
