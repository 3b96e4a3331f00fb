"""Tokenizer wrappers (capability parity with reference
src/modalities/tokenization/tokenizer_wrapper.py:9-285): a uniform interface
over HuggingFace fast tokenizers and SentencePiece, with special-token
lookup and padding/truncation controls."""

from abc import ABC, abstractmethod
from typing import Optional


class TokenizerWrapper(ABC):
    @abstractmethod
    def tokenize(self, text: str) -> list[int]: ...

    @abstractmethod
    def decode(self, token_ids: list[int]) -> str: ...

    @property
    @abstractmethod
    def vocab_size(self) -> int: ...

    def get_token_id(self, token: str) -> int:
        raise NotImplementedError

    @property
    def special_tokens(self) -> dict[str, int]:
        return {}


class PreTrainedHFTokenizer(TokenizerWrapper):
    def __init__(self, pretrained_model_name_or_path: str,
                 truncation: Optional[bool] = False, padding: bool | str = False,
                 max_length: Optional[int] = None,
                 special_tokens: Optional[dict[str, str]] = None):
        from transformers import AutoTokenizer
        self.tokenizer = AutoTokenizer.from_pretrained(pretrained_model_name_or_path)
        if special_tokens is not None:
            # NOTE (parity with reference tokenizer_wrapper.py:64-76): adding
            # tokens beyond the vocab requires resizing model embeddings.
            self.tokenizer.add_special_tokens({
                k: v for k, v in special_tokens.items()})
        self.truncation = truncation
        self.padding = padding
        self.max_length = max_length

    @property
    def vocab_size(self) -> int:
        return self.tokenizer.vocab_size

    @property
    def special_tokens(self) -> dict[str, int]:
        return {t: self.tokenizer.convert_tokens_to_ids(t)
                for t in self.tokenizer.all_special_tokens}

    def tokenize(self, text: str) -> list[int]:
        return self.tokenizer(
            text, max_length=self.max_length, padding=self.padding,
            truncation=self.truncation)["input_ids"]

    def decode(self, token_ids: list[int]) -> str:
        return self.tokenizer.decode(token_ids)

    def get_token_id(self, token: str) -> int:
        tid = self.tokenizer.convert_tokens_to_ids(token)
        if not isinstance(tid, int):
            raise ValueError("Token is not represented by a single id")
        if tid == getattr(self.tokenizer, "unk_token_id", None) and token != \
                getattr(self.tokenizer, "unk_token", None):
            raise ValueError(f"Token {token!r} not in vocabulary")
        return tid


class PreTrainedSPTokenizer(TokenizerWrapper):
    def __init__(self, tokenizer_model_file: str):
        import sentencepiece
        self.tokenizer = sentencepiece.SentencePieceProcessor()
        self.tokenizer.Load(tokenizer_model_file)

    @property
    def vocab_size(self) -> int:
        return self.tokenizer.vocab_size()

    def tokenize(self, text: str) -> list[int]:
        return self.tokenizer.encode(text)

    def decode(self, token_ids: list[int]) -> str:
        return self.tokenizer.decode(token_ids)

    def get_token_id(self, token: str) -> int:
        tid = self.tokenizer.PieceToId(token)
        if tid == self.tokenizer.unk_id() and token != self.tokenizer.IdToPiece(
                self.tokenizer.unk_id()):
            raise ValueError(f"Token {token!r} not in vocabulary")
        return tid


class CharTokenizer(TokenizerWrapper):
    """Self-contained byte-level tokenizer (no external vocab files): token
    id = byte value; ids 256..259 reserved for special tokens. Useful for
    offline tests and smoke datasets."""

    SPECIALS = {"<|endoftext|>": 256, "<eod>": 257, "^": 258, "$": 259}

    def __init__(self):
        pass

    @property
    def vocab_size(self) -> int:
        return 260

    @property
    def special_tokens(self) -> dict[str, int]:
        return dict(self.SPECIALS)

    def tokenize(self, text: str) -> list[int]:
        out: list[int] = []
        i = 0
        while i < len(text):
            matched = False
            for tok, tid in self.SPECIALS.items():
                if text.startswith(tok, i):
                    out.append(tid)
                    i += len(tok)
                    matched = True
                    break
            if not matched:
                out.extend(text[i].encode("utf-8", errors="replace"))
                i += 1
        return out

    def decode(self, token_ids: list[int]) -> str:
        inv = {v: k for k, v in self.SPECIALS.items()}
        parts: list[bytes] = []
        for tid in token_ids:
            if tid in inv:
                parts.append(inv[tid].encode())
            elif tid < 256:
                parts.append(bytes([tid]))
        return b"".join(parts).decode("utf-8", errors="replace")

    def get_token_id(self, token: str) -> int:
        if token in self.SPECIALS:
            return self.SPECIALS[token]
        ids = token.encode("utf-8")
        if len(ids) == 1:
            return ids[0]
        raise ValueError(f"Token {token!r} is not a single id")
