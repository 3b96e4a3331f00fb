"""HTTP inference server: /health and /generate over the KV-cache decoder
(starlette TestClient, no network)."""

import pytest

fastapi = pytest.importorskip("fastapi")

from fastapi.testclient import TestClient  # noqa: E402

from modalities_amd.inference.server import build_app  # noqa: E402
from modalities_amd.inference.text_generation import \
    TextInferenceComponent  # noqa: E402
from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig  # noqa: E402
from modalities_amd.tokenization.tokenizer_wrapper import \
    CharTokenizer  # noqa: E402


@pytest.fixture(scope="module")
def client():
    import torch
    torch.manual_seed(0)
    model = GPT2LLM(GPT2LLMConfig(
        vocab_size=260, n_layer=2, n_head_q=4, n_head_kv=4, n_embd=64,
        ffn_hidden=128, sequence_length=48, seed=5, dropout=0.0))
    comp = TextInferenceComponent(model, CharTokenizer(),
                                  prompt_template="{text}",
                                  sequence_length=48, temperature=0.0)
    return TestClient(build_app(comp))


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json() == {"status": "ok"}


def test_generate_greedy_deterministic(client):
    req = {"prompt": "hello", "max_new_tokens": 8, "temperature": 0.0}
    r1 = client.post("/generate", json=req)
    r2 = client.post("/generate", json=req)
    assert r1.status_code == 200
    b1, b2 = r1.json(), r2.json()
    assert b1["text"] == b2["text"]
    assert b1["prompt_tokens"] == 5
    assert b1["latency_ms"] > 0


def test_generate_respects_max_new_tokens(client):
    r = client.post("/generate", json={"prompt": "abc", "max_new_tokens": 3,
                                       "temperature": 0.0})
    assert r.status_code == 200
    # byte-level tokenizer: generated text decodes from <= 3 token ids
    assert r.json()["generated_tokens"] <= 3 + 2  # utf-8 re-encode slack


def test_generate_validation_error(client):
    r = client.post("/generate", json={"prompt": "x", "max_new_tokens": 0})
    assert r.status_code == 422


def test_serve_builds_from_config(tmp_path):
    """The serve CLI's build path: config -> TextInferenceComponent -> app."""
    cfg_text = """\
settings:
  referencing_keys: {sample_key: input_ids, prediction_key: logits}
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

text_inference:
  prompt_template: "{text}"
  temperature: 0.0
"""
    cfg = tmp_path / "gen.yaml"
    cfg.write_text(cfg_text)

    from modalities_amd.api import _build_text_inference_component
    comp = _build_text_inference_component(cfg)
    c = TestClient(build_app(comp))
    r = c.post("/generate", json={"prompt": "hey", "max_new_tokens": 4})
    assert r.status_code == 200 and isinstance(r.json()["text"], str)
