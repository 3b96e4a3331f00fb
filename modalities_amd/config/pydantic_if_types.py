"""Typed interface aliases for component references in pydantic configs
(capability parity with reference src/modalities/config/pydantic_if_types.py):
annotated types that validate a built component is an instance of the
expected interface."""

from typing import Annotated, Any

from pydantic import BeforeValidator

from modalities_amd.checkpointing.app_state import AppState
from modalities_amd.checkpointing.saving import CheckpointSaving
from modalities_amd.dataloader.dataloader import LLMDataLoader
from modalities_amd.logging_broker.broker import MessageSubscriberIF
from modalities_amd.loss_functions import Loss
from modalities_amd.nn.model_initialization import ModelInitializationIF
from modalities_amd.parallel.mesh import DeviceMesh
from modalities_amd.tokenization.tokenizer_wrapper import TokenizerWrapper


def _instance_of(t):
    def check(v: Any):
        if not isinstance(v, t):
            raise TypeError(f"expected an instance of {t.__name__}, "
                            f"got {type(v).__name__}")
        return v
    return BeforeValidator(check)


import torch  # noqa: E402

PydanticNNModuleIFType = Annotated[Any, _instance_of(torch.nn.Module)]
PydanticOptimizerIFType = Annotated[Any, _instance_of(torch.optim.Optimizer)]
PydanticLossIFType = Annotated[Any, _instance_of(Loss)]
PydanticLLMDataLoaderIFType = Annotated[Any, _instance_of(LLMDataLoader)]
PydanticTokenizerIFType = Annotated[Any, _instance_of(TokenizerWrapper)]
PydanticAppStateType = Annotated[Any, _instance_of(AppState)]
PydanticCheckpointSavingIFType = Annotated[Any, _instance_of(CheckpointSaving)]
PydanticMessageSubscriberIFType = Annotated[Any, _instance_of(MessageSubscriberIF)]
PydanticDeviceMeshIFType = Annotated[Any, _instance_of(DeviceMesh)]
PydanticModelInitializationIFType = Annotated[Any,
                                              _instance_of(ModelInitializationIF)]
PydanticDatasetIFType = Annotated[Any,
                                  _instance_of(torch.utils.data.Dataset)]
PydanticSamplerIFType = Annotated[Any,
                                  _instance_of(torch.utils.data.Sampler)]
