"""SentencePiece → HuggingFace tokenizer conversion (capability parity with
reference src/modalities/conversion/gpt2/conversion_tokenizer.py:11-45):
wrap a raw SentencePiece model file as a saved HF tokenizer directory so an
exported checkpoint (convert_gpt2.py) ships with a loadable tokenizer.

Design: we hand the .model file to ``transformers.LlamaTokenizer`` in
legacy mode with special-token handling disabled, so the inner
SentencePiece processor keeps full authority over ids (no HF-side bos/eos
insertion or splitting). The true special-token ids live in the SP proto
and are returned to the caller for the model config.
"""

import shutil
import tempfile
from pathlib import Path

from modalities_amd.tokenization.tokenizer_wrapper import PreTrainedSPTokenizer


def convert_tokenizer(tokenizer_model_path: str, output_dir: str) -> tuple[int, int, int, int]:
    """Save an HF-loadable tokenizer built from a SentencePiece model file.

    Returns (bos_id, eos_id, pad_id, unk_id) as known to the SentencePiece
    model (-1 where undefined); these are NOT written into the HF tokenizer
    config — the wrapped SP processor handles them.
    """
    from transformers import LlamaTokenizer

    sp = PreTrainedSPTokenizer(tokenizer_model_path)

    # LlamaTokenizer.from_pretrained wants a directory with tokenizer.model
    with tempfile.TemporaryDirectory() as tmp:
        shutil.copy2(tokenizer_model_path, Path(tmp) / "tokenizer.model")
        hf_tok = LlamaTokenizer.from_pretrained(
            tmp,
            # neutralize HF-side special-token logic: the SP model's own
            # pieces are authoritative
            bos_token=None, eos_token=None, pad_token=None, unk_token=None,
            add_bos_token=False, add_eos_token=False,
            split_special_tokens=_splits_special_tokens(sp),
            legacy=True,
        )
    hf_tok.add_bos_token = False
    hf_tok.add_eos_token = False
    hf_tok.legacy = True
    Path(output_dir).mkdir(parents=True, exist_ok=True)
    hf_tok.save_pretrained(output_dir)
    t = sp.tokenizer
    return t.bos_id(), t.eos_id(), t.pad_id(), t.unk_id()


def _splits_special_tokens(sp: PreTrainedSPTokenizer) -> bool:
    """Whether the SP model tokenizes special-token text as plain text
    (i.e. does NOT map the piece string back to its single id)."""
    t = sp.tokenizer
    probe = next((tid for tid in (t.bos_id(), t.eos_id(), t.unk_id()) if tid >= 0), None)
    if probe is None:
        return False
    piece = t.IdToPiece(probe)
    encoded = t.encode(piece)
    return encoded != [probe]
