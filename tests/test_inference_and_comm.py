"""Functional coverage for text generation (reference
inference/text/inference_component.py behavior) and the communication
self-test (utils/communication_test.py) on gloo world 2."""

import pytest
import torch

from modalities_amd.inference.text_generation import TextInferenceComponent
from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.tokenization.tokenizer_wrapper import CharTokenizer

from tests.utils_dist import run_distributed


def _tiny_model(vocab=260, seq=48):
    torch.manual_seed(7)
    return GPT2LLM(GPT2LLMConfig(
        vocab_size=vocab, n_layer=2, n_head_q=4, n_head_kv=4, n_embd=64,
        ffn_hidden=128, sequence_length=seq, seed=5, dropout=0.0))


class CountingModel:
    def __init__(self, inner):
        self.inner = inner
        self.calls = 0

    def eval(self):
        self.inner.eval()

    def __call__(self, inputs):
        self.calls += 1
        return self.inner(inputs)


def test_greedy_generation_is_deterministic_and_bounded():
    tok = CharTokenizer()
    counting = CountingModel(_tiny_model())
    gen = TextInferenceComponent(counting, tok, prompt_template="{text}",
                                 sequence_length=24, temperature=0.0)
    out1 = gen.generate_tokens("hello")
    calls1 = counting.calls
    out2 = gen.generate_tokens("hello")
    assert out1 == out2  # greedy = deterministic
    # prompt is 5 tokens; at most seq_len - prompt forward steps per call
    assert calls1 <= 24 - 5


def test_generation_stops_at_eod():
    tok = CharTokenizer()
    model = _tiny_model()

    class EodAfterTwo:
        """Wrap the model to force the eod token at the 3rd step."""

        def __init__(self, inner):
            self.inner = inner
            self.calls = 0

        def eval(self):
            self.inner.eval()

        def __call__(self, inputs):
            self.calls += 1
            out = self.inner(inputs)
            if self.calls >= 3:
                out["logits"] = out["logits"].clone()
                out["logits"][:, -1, :] = -1e9
                out["logits"][:, -1, tok.get_token_id("<eod>")] = 1e9
            return out

    gen = TextInferenceComponent(EodAfterTwo(model), tok,
                                 prompt_template="{text}",
                                 sequence_length=40, temperature=0.0)
    out = gen.generate_tokens("ab")
    assert gen.model.calls == 3  # stopped right at the forced eod step
    assert "<eod>" not in out


def test_temperature_sampling_runs():
    tok = CharTokenizer()
    model = _tiny_model()
    torch.manual_seed(11)
    gen = TextInferenceComponent(model, tok, prompt_template="{text}",
                                 sequence_length=16, temperature=0.8)
    out = gen.generate_tokens("xy")
    assert isinstance(out, str)


def _comm_worker(rank, world):
    from modalities_amd.utils.communication_test import run_communication_test
    run_communication_test(device=torch.device("cpu"))
    return "ok"


def test_communication_self_test_world2():
    results = run_distributed(_comm_worker, world_size=2, port=29461)
    assert results == {0: "ok", 1: "ok"}


def test_generate_text_config_path(tmp_path):
    """The `generate_text` CLI's component path: config -> model+tokenizer
    -> TextInferenceComponent (non-interactive; we call generate_tokens
    instead of the stdin loop)."""
    cfg_text = """\
settings:
  referencing_keys:
    sample_key: input_ids
    prediction_key: logits
  device: cpu
  sequence_length: 24

model:
  component_key: model
  variant_key: gpt2
  config:
    sample_key: input_ids
    prediction_key: logits
    vocab_size: 260
    n_layer: 2
    n_head_q: 4
    n_head_kv: 4
    n_embd: 64
    ffn_hidden: 128
    sequence_length: 24
    seed: 3

tokenizer:
  component_key: tokenizer
  variant_key: char
  config: {}
"""
    cfg = tmp_path / "gen.yaml"
    cfg.write_text(cfg_text)

    import torch

    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.config.instantiation_models import \
        TextGenerationInstantiationModel
    from modalities_amd.config.yaml_loader import load_app_config_dict
    from modalities_amd.registry.components import get_default_registry

    config_dict = load_app_config_dict(cfg)
    factory = ComponentFactory(get_default_registry())
    components = factory.build_components(config_dict,
                                          TextGenerationInstantiationModel)
    comp = TextInferenceComponent(
        components.model, components.tokenizer, prompt_template="{text}",
        sequence_length=components.settings.sequence_length, temperature=0.0,
        device=torch.device(components.settings.device))
    out = comp.generate_tokens("hi")
    assert isinstance(out, str)


@pytest.mark.parametrize("kw", [
    dict(),
    dict(fused_qkv=True),
    dict(n_head_kv=2),
    dict(use_qk_norm=True),
    dict(poe_type="ABSOLUTE"),
])
def test_kv_cache_decode_matches_full_reforward(kw):
    """forward_cached (prefill + per-token decode) must reproduce the full
    re-forward logits at every step."""
    torch.manual_seed(0)
    cfg = dict(vocab_size=97, n_layer=2, n_head_q=4, n_head_kv=4, n_embd=64,
               ffn_hidden=128, sequence_length=32, seed=11, dropout=0.0)
    cfg.update(kw)
    model = GPT2LLM(GPT2LLMConfig(**cfg))
    model.eval()

    g = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 97, (2, 7), generator=g)

    with torch.no_grad():
        cache = model.new_kv_cache(2, max_len=16)
        out_c = model.forward_cached({"input_ids": prompt}, cache)["logits"]
        ref = model({"input_ids": prompt})["logits"]
        torch.testing.assert_close(out_c, ref, rtol=1e-4, atol=1e-4)

        ids = prompt
        for _ in range(6):
            nxt = ref[:, -1, :].argmax(-1, keepdim=True)
            ids = torch.cat([ids, nxt], dim=1)
            ref = model({"input_ids": ids})["logits"]
            out_c = model.forward_cached({"input_ids": nxt}, cache)["logits"]
            torch.testing.assert_close(out_c[:, -1], ref[:, -1],
                                       rtol=1e-4, atol=1e-4)
        assert cache.pos == ids.shape[1]



def test_top_k_and_top_p_sampling():
    """top_k=1 must equal greedy; top_p keeps exactly the nucleus."""
    tok = CharTokenizer()
    model = _tiny_model()

    greedy = TextInferenceComponent(model, tok, "{text}", 16, temperature=0.0)
    topk1 = TextInferenceComponent(model, tok, "{text}", 16, temperature=0.7,
                                   top_k=1)
    assert greedy.generate_tokens("ab") == topk1.generate_tokens("ab")

    # nucleus filter keeps the minimal prefix covering top_p
    gen = TextInferenceComponent(model, tok, "{text}", 16, temperature=1.0,
                                 top_p=0.5)
    logits = torch.log(torch.tensor([[0.4, 0.3, 0.2, 0.1]]))
    torch.manual_seed(0)
    seen = {gen._sample(logits).item() for _ in range(200)}
    assert seen == {0, 1}  # 0.4 then 0.4+0.3 > 0.5 -> nucleus = {0, 1}

    # top_k=2 restricts to the two best
    gen2 = TextInferenceComponent(model, tok, "{text}", 16, temperature=1.0,
                                  top_k=2)
    seen2 = {gen2._sample(logits).item() for _ in range(200)}
    assert seen2 == {0, 1}


def test_kv_cache_overflow_raises():
    model = _tiny_model()
    model.eval()
    cache = model.new_kv_cache(1, max_len=8)
    with torch.no_grad():
        model.forward_cached({"input_ids": torch.randint(0, 97, (1, 6))}, cache)
        with pytest.raises(ValueError, match="overflow"):
            model.forward_cached({"input_ids": torch.randint(0, 97, (1, 4))},
                                 cache)
    cache.reset()
    assert cache.pos == 0
