"""Top-level instantiation models (capability parity with reference
src/modalities/config/instantiation_models.py:73-347): the pydantic shells
that the ComponentFactory fills with live components, plus consistency
validation (tokens-per-step vs dataset size, interval alignment) and the
startup training report."""

from pathlib import Path
from typing import Any, Optional

from pydantic import BaseModel, ConfigDict, Field, model_validator


class WarnErrorIgnoreEnum(str):
    pass


class IntervalSettings(BaseModel):
    training_log_interval_in_steps: int = 1
    checkpointing_interval_in_steps: int = 0
    evaluation_interval_in_steps: int = 0


class TrainingTarget(BaseModel):
    num_target_tokens: int
    num_target_steps: int


class TrainingProgressSettings(BaseModel):
    global_num_seen_tokens: int = 0
    num_seen_steps: int = 0
    num_seen_samples: int = 0
    last_step: int = -1


class CudaEnvSettings(BaseModel):
    local_rank: int = 0
    global_rank: int = 0
    world_size: int = 1


class PathSettings(BaseModel):
    model_config = ConfigDict(extra="allow")
    checkpoint_saving_path: Optional[Path] = None
    train_dataset_path: Optional[Path] = None


class StepProfile(BaseModel):
    gradient_accumulation_steps: int = 1
    local_train_micro_batch_size: int = 1
    sequence_length: int = 1
    # Data-parallel degree (dp_replicate * dp_shard). Tokens per train step
    # scale with the number of DATA-parallel replicas, not world_size: a
    # composed tp/cp/pp config has world_size > dp_degree and using
    # world_size would inflate token accounting by the tp*cp*pp factor
    # (reference main.py:168-173 multiplies by dp_degree). None = derive
    # from the device mesh at runtime, falling back to world_size.
    dp_degree: Optional[int] = None


class ConsistencyEnforcement(BaseModel):
    enforce_tokens_per_step_consistency: bool = True
    enforce_last_step_logged: bool = False
    enforce_last_step_evaluated: bool = False
    enforce_last_step_checkpointed: bool = False


class TrainingSettings(BaseModel):
    model_config = ConfigDict(extra="allow")
    experiment_id: str = "exp"
    config_file_path: Optional[Path] = None
    referencing_keys: dict[str, str] = Field(
        default_factory=lambda: {"sample_key": "input_ids",
                                 "target_key": "target_ids",
                                 "prediction_key": "logits"})
    cuda_env: CudaEnvSettings = CudaEnvSettings()
    paths: PathSettings = PathSettings()
    intervals: IntervalSettings = IntervalSettings()
    consistency_enforcement: ConsistencyEnforcement = ConsistencyEnforcement()
    step_profile: StepProfile = StepProfile()
    training_target: TrainingTarget = TrainingTarget(num_target_tokens=0,
                                                     num_target_steps=0)
    training_progress: TrainingProgressSettings = TrainingProgressSettings()
    seed: int = 42

    @model_validator(mode="after")
    def _validate_tokens_per_step(self):
        """tokens/step implied by the step profile must divide the target
        tokens into the target steps (reference
        instantiation_models.py:110-179)."""
        sp = self.step_profile
        dp_degree = sp.dp_degree if sp.dp_degree is not None \
            else self.cuda_env.world_size
        tokens_per_step = (sp.gradient_accumulation_steps
                           * sp.local_train_micro_batch_size * sp.sequence_length
                           * dp_degree)
        tt = self.training_target
        if tt.num_target_steps > 0 and tt.num_target_tokens > 0:
            expected = tokens_per_step * tt.num_target_steps
            if expected != tt.num_target_tokens and \
                    self.consistency_enforcement.enforce_tokens_per_step_consistency:
                raise ValueError(
                    f"Inconsistent training target: {tt.num_target_steps} steps x "
                    f"{tokens_per_step} tokens/step = {expected} != "
                    f"{tt.num_target_tokens} target tokens")
        return self


class TrainingComponentsInstantiationModel(BaseModel):
    model_config = ConfigDict(arbitrary_types_allowed=True)

    settings: TrainingSettings
    wrapped_model: Any
    optimizer: Any
    scheduler: Any = None
    loss_fn: Any
    train_dataloader: Any
    eval_dataloaders: list = Field(default_factory=list)
    checkpoint_saving: Any = None
    gradient_clipper: Any = None
    progress_subscriber: Any = None
    evaluation_subscriber: Any = None
    device_mesh: Any = None
    mfu_calculator: Any = None
    app_state: Any = None
    pp_schedule: Any = None  # PipelineSchedule when PP is active


class TextGenerationSettings(BaseModel):
    model_config = ConfigDict(extra="allow")
    referencing_keys: dict[str, str] = Field(
        default_factory=lambda: {"sample_key": "input_ids",
                                 "prediction_key": "logits"})
    device: str = "cpu"
    sequence_length: int = 1024


class TextGenerationInstantiationModel(BaseModel):
    model_config = ConfigDict(arbitrary_types_allowed=True)
    settings: TextGenerationSettings
    model: Any
    tokenizer: Any


class TrainingReportGenerator:
    """Startup report (reference instantiation_models.py:245-347)."""

    def __init__(self, settings: TrainingSettings, num_params: int,
                 extra: Optional[dict] = None):
        self.settings = settings
        self.num_params = num_params
        self.extra = extra or {}

    def get_report(self) -> str:
        s = self.settings
        sp = s.step_profile
        dp_degree = sp.dp_degree if sp.dp_degree is not None \
            else s.cuda_env.world_size
        tokens_per_step = (sp.gradient_accumulation_steps
                           * sp.local_train_micro_batch_size * sp.sequence_length
                           * dp_degree)
        lines = [
            "==== Training Report ====",
            f"experiment_id:        {s.experiment_id}",
            f"world_size:           {s.cuda_env.world_size}",
            f"num parameters:       {self.num_params:,}",
            f"sequence_length:      {sp.sequence_length}",
            f"micro_batch_size:     {sp.local_train_micro_batch_size}",
            f"grad_accumulation:    {sp.gradient_accumulation_steps}",
            f"tokens per step:      {tokens_per_step:,}",
            f"target steps:         {s.training_target.num_target_steps:,}",
            f"target tokens:        {s.training_target.num_target_tokens:,}",
        ]
        for k, v in self.extra.items():
            lines.append(f"{k}: {v}")
        return "\n".join(lines)
