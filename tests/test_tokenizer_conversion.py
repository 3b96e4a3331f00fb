"""SentencePiece → HF tokenizer conversion (reference
conversion/gpt2/conversion_tokenizer.py behavior): the saved HF tokenizer
must encode identically to the raw SentencePiece processor."""

import pytest

sentencepiece = pytest.importorskip("sentencepiece")
transformers = pytest.importorskip("transformers")


@pytest.fixture(scope="module")
def sp_model(tmp_path_factory):
    root = tmp_path_factory.mktemp("sp")
    corpus = root / "corpus.txt"
    words = ["alpha", "beta", "gamma", "delta", "omega", "train", "model", "token"]
    import random
    rng = random.Random(3)
    corpus.write_text("\n".join(" ".join(rng.choices(words, k=12))
                                for _ in range(400)))
    sentencepiece.SentencePieceTrainer.train(
        input=str(corpus), model_prefix=str(root / "tiny"),
        vocab_size=64, model_type="bpe", minloglevel=2)
    return root / "tiny.model"


def test_sp_to_hf_conversion_roundtrip(sp_model, tmp_path):
    from modalities_amd.conversion.convert_tokenizer import convert_tokenizer
    from modalities_amd.tokenization.tokenizer_wrapper import PreTrainedSPTokenizer

    bos, eos, pad, unk = convert_tokenizer(str(sp_model), str(tmp_path / "hf_tok"))
    assert bos == 1 and eos == 2 and pad == -1 and unk == 0  # SP BPE defaults

    hf = transformers.LlamaTokenizer.from_pretrained(str(tmp_path / "hf_tok"))
    sp = PreTrainedSPTokenizer(str(sp_model))
    for text in ["alpha beta gamma train model",
                 "token omega delta", "alphabeta gamma"]:
        assert hf.encode(text) == sp.tokenize(text), text
        assert hf.decode(hf.encode(text)) == sp.decode(sp.tokenize(text))
    # no HF-side bos/eos insertion
    assert 1 not in hf.encode("alpha") and 2 not in hf.encode("alpha")


def test_convert_gpt2_to_hf_ships_sp_tokenizer(sp_model, tmp_path):
    """convert_gpt2_to_hf writes the converted tokenizer alongside the model
    export when the config names an SP tokenizer."""
    cfg_text = f"""\
model:
  component_key: model
  variant_key: gpt2
  config:
    sample_key: input_ids
    prediction_key: logits
    vocab_size: 64
    n_layer: 1
    n_head_q: 2
    n_head_kv: 2
    n_embd: 32
    ffn_hidden: 64
    sequence_length: 16
    seed: 2

tokenizer:
  component_key: tokenizer
  variant_key: pretrained_sp_tokenizer
  config:
    tokenizer_model_file: {sp_model}
"""
    cfg = tmp_path / "conv.yaml"
    cfg.write_text(cfg_text)

    from modalities_amd.conversion.convert_gpt2 import convert_gpt2_to_hf
    out = tmp_path / "export"
    convert_gpt2_to_hf(cfg, out, verify=True)
    # model export present
    assert (out / "config.json").exists()
    # tokenizer shipped and loadable
    hf_tok = transformers.LlamaTokenizer.from_pretrained(str(out))
    from modalities_amd.tokenization.tokenizer_wrapper import PreTrainedSPTokenizer
    sp = PreTrainedSPTokenizer(str(sp_model))
    assert hf_tok.encode("alpha beta") == sp.tokenize("alpha beta")
