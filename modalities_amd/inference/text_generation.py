"""Text generation (capability parity with reference
src/modalities/inference/text/inference_component.py:11-105): prompt
templating, temperature/argmax sampling, incremental decode printing.

MI355X-first improvement over the reference: a KV-cache-free full-context
re-forward per token is kept as the simple default (matching the reference's
behavior) but generation batches the forward on device in bf16."""

import sys
from typing import Optional

import torch

from modalities_amd.tokenization.tokenizer_wrapper import TokenizerWrapper


class TextInferenceComponent:
    def __init__(self, model, tokenizer: TokenizerWrapper, prompt_template: str,
                 sequence_length: int, temperature: float = 1.0,
                 eod_token: str = "<eod>", device: Optional[torch.device] = None,
                 sample_key: str = "input_ids", prediction_key: str = "logits",
                 top_k: Optional[int] = None, top_p: Optional[float] = None):
        self.model = model
        self.tokenizer = tokenizer
        self.prompt_template = prompt_template
        self.sequence_length = sequence_length
        self.temperature = temperature
        self.eod_token = eod_token
        self.device = device or torch.device("cpu")
        self.sample_key = sample_key
        self.prediction_key = prediction_key
        self.top_k = top_k
        self.top_p = top_p

    def _sample(self, logits: torch.Tensor) -> torch.Tensor:
        """Greedy (temperature==0) or temperature sampling with optional
        top-k / nucleus (top-p) filtering."""
        if self.temperature <= 0:
            return logits.argmax(dim=-1, keepdim=True)
        logits = logits / self.temperature
        if self.top_k is not None and self.top_k > 0:
            kth = torch.topk(logits, min(self.top_k, logits.shape[-1]),
                             dim=-1).values[..., -1:]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        if self.top_p is not None and 0.0 < self.top_p < 1.0:
            sorted_logits, sorted_idx = torch.sort(logits, descending=True,
                                                   dim=-1)
            cum = torch.softmax(sorted_logits, dim=-1).cumsum(-1)
            # drop tokens whose EXCLUSIVE cumulative prob already exceeds
            # top_p (the highest-prob token always stays)
            drop_sorted = torch.zeros_like(cum, dtype=torch.bool)
            drop_sorted[..., 1:] = cum[..., :-1] > self.top_p
            drop = torch.zeros_like(drop_sorted).scatter(
                -1, sorted_idx, drop_sorted)
            logits = logits.masked_fill(drop, float("-inf"))
        probs = torch.softmax(logits, dim=-1)
        return torch.multinomial(probs, num_samples=1)

    @torch.no_grad()
    def generate_tokens(self, context: str, echo: bool = False) -> str:
        """Greedy (temperature==0) or temperature sampling until eod or
        sequence_length tokens."""
        ids = self.tokenizer.tokenize(context)
        input_ids = torch.tensor(ids, dtype=torch.long,
                                 device=self.device).unsqueeze(0)
        try:
            eod_id = self.tokenizer.get_token_id(self.eod_token)
        except Exception:
            eod_id = None
        generated: list[int] = []
        max_new = max(0, self.sequence_length - input_ids.shape[1])
        self.model.eval()

        # KV-cache incremental decode when the model supports it (GPT2LLM);
        # fall back to full-context re-forward otherwise.
        cache = None
        if hasattr(self.model, "forward_cached") and                 hasattr(self.model, "new_kv_cache"):
            cache = self.model.new_kv_cache(input_ids.shape[0],
                                            max_len=self.sequence_length)
            out = self.model.forward_cached({self.sample_key: input_ids}, cache)
        else:
            out = self.model({self.sample_key: input_ids})

        for i in range(max_new):
            logits = out[self.prediction_key][:, -1, :].float()
            next_id = self._sample(logits)
            token = next_id.item()
            if eod_id is not None and token == eod_id:
                break
            generated.append(token)
            if echo:
                sys.stdout.write(self.tokenizer.decode([token]))
                sys.stdout.flush()
            if i + 1 >= max_new:
                break
            if cache is not None:
                if cache.pos >= self.sequence_length:
                    break
                out = self.model.forward_cached(
                    {self.sample_key: next_id}, cache)
            else:
                input_ids = torch.cat([input_ids, next_id], dim=1)
                out = self.model({self.sample_key: input_ids})
        if echo:
            sys.stdout.write("\n")
        return self.tokenizer.decode(generated)

    def run(self) -> None:
        """Interactive prompt loop (reference inference_component.py:90-105)."""
        while True:
            try:
                prompt = input("enter prompt> ")
            except (EOFError, KeyboardInterrupt):
                break
            if not prompt.strip():
                continue
            text = self.prompt_template.format(text=prompt) \
                if "{text}" in self.prompt_template else prompt
            self.generate_tokens(text, echo=True)
