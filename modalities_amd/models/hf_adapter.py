"""HuggingFace adapter (capability parity with reference
src/modalities/models/huggingface_adapters/hf_adapter.py:14-160): wrap a
modalities_amd model as a transformers PreTrainedModel so HF pipelines /
generate() work, plus a wrapper to USE a HF pretrained model inside this
framework (reference models/huggingface/huggingface_model.py)."""

from typing import Optional

import torch

from modalities_amd.models.model import NNModel


def _require_transformers():
    try:
        import transformers  # noqa: F401
    except ImportError as e:
        raise ImportError("transformers is required for the HF adapter") from e


class HFAdapterConfig:
    """Lazily constructed to avoid a hard transformers import at module load."""

    def __new__(cls, *args, **kwargs):
        _require_transformers()
        from transformers import PretrainedConfig

        class _Cfg(PretrainedConfig):
            model_type = "modalities_amd"

            def __init__(self, config: Optional[dict] = None, **kw):
                super().__init__(**kw)
                self.config = config or {}

        return _Cfg(*args, **kwargs)


def get_hf_model_adapter(model: NNModel, sample_key: str = "input_ids",
                         prediction_key: str = "logits"):
    """Wrap `model` as a PreTrainedModel exposing CausalLM forward/generate."""
    _require_transformers()
    from transformers import GenerationMixin, PretrainedConfig, PreTrainedModel
    from transformers.modeling_outputs import CausalLMOutput

    class _Cfg(PretrainedConfig):
        model_type = "modalities_amd"

    class HFModelAdapter(PreTrainedModel, GenerationMixin):
        config_class = _Cfg
        _no_split_modules = ["GPT2Block"]

        def __init__(self, config, inner):
            super().__init__(config)
            self.inner = inner

        def forward(self, input_ids=None, attention_mask=None, labels=None,
                    return_dict=True, **kwargs):
            out = self.inner({sample_key: input_ids})
            logits = out[prediction_key]
            loss = None
            if labels is not None:
                loss = torch.nn.functional.cross_entropy(
                    logits[:, :-1].reshape(-1, logits.shape[-1]).float(),
                    labels[:, 1:].reshape(-1))
            if not return_dict:
                return (loss, logits) if loss is not None else (logits,)
            return CausalLMOutput(loss=loss, logits=logits)

        def prepare_inputs_for_generation(self, input_ids, **kwargs):
            return {"input_ids": input_ids}

    cfg = _Cfg()
    mc = getattr(model, "config", None)
    if mc is not None:
        cfg.vocab_size = getattr(mc, "vocab_size", None)
        cfg.num_hidden_layers = getattr(mc, "n_layer", 1)
        cfg.hidden_size = getattr(mc, "n_embd", None)
        cfg.num_attention_heads = getattr(mc, "n_head_q", None)
        cfg.max_position_embeddings = getattr(mc, "sequence_length", None)
    return HFModelAdapter(cfg, model)


class HuggingFacePretrainedModel(NNModel):
    """Use a HF pretrained causal LM as a modalities_amd model component
    (reference models/huggingface/huggingface_model.py)."""

    def __init__(self, model_name: str, sample_key: str = "input_ids",
                 prediction_key: str = "logits"):
        _require_transformers()
        super().__init__()
        from transformers import AutoModelForCausalLM
        self.sample_key = sample_key
        self.prediction_key = prediction_key
        self.huggingface_model = AutoModelForCausalLM.from_pretrained(model_name)

    def forward(self, inputs: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        out = self.huggingface_model(input_ids=inputs[self.sample_key])
        return {self.prediction_key: out.logits}
