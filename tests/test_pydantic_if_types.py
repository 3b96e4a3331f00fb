"""Typed interface aliases validate component instances."""

import pytest
import torch
from pydantic import BaseModel, ConfigDict

from modalities_amd.config.pydantic_if_types import (PydanticLossIFType,
                                                     PydanticNNModuleIFType)
from modalities_amd.loss_functions import CLMCrossEntropyLoss


class _M(BaseModel):
    model_config = ConfigDict(arbitrary_types_allowed=True)
    model: PydanticNNModuleIFType
    loss: PydanticLossIFType


def test_interface_aliases_accept_and_reject():
    m = _M(model=torch.nn.Linear(2, 2),
           loss=CLMCrossEntropyLoss("t", "p"))
    assert isinstance(m.model, torch.nn.Module)
    with pytest.raises(Exception):
        _M(model="not a module", loss=CLMCrossEntropyLoss("t", "p"))
