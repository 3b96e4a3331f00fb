"""Minimal HTTP inference server: POST /generate over the KV-cache decoder
(MI355X-first serving addition; the reference ships only interactive CLI
generation). Built lazily on FastAPI so the dependency stays optional.

    python -m modalities_amd serve --config_file_path gen.yaml --port 8000

Request:  {"prompt": "...", "max_new_tokens": 64, "temperature": 0.8,
           "top_k": 50, "top_p": 0.95}
Response: {"text": "...", "prompt_tokens": N, "generated_tokens": M,
           "latency_ms": T}
"""

import threading
import time
from typing import Optional

from modalities_amd.inference.text_generation import TextInferenceComponent


def build_app(component: TextInferenceComponent):
    from fastapi import FastAPI
    from pydantic import BaseModel, Field

    class GenerateRequest(BaseModel):
        prompt: str
        max_new_tokens: int = Field(default=64, ge=1)
        temperature: Optional[float] = None
        top_k: Optional[int] = None
        top_p: Optional[float] = None

    app = FastAPI(title="modalities-amd inference")

    @app.get("/health")
    def health():
        return {"status": "ok"}

    # generation mutates component state (sampling overrides, KV cache):
    # serialize requests — FastAPI runs sync handlers in a thread pool
    gen_lock = threading.Lock()

    @app.post("/generate")
    def generate(req: GenerateRequest):
        text = component.prompt_template.format(text=req.prompt) \
            if "{text}" in component.prompt_template else req.prompt
        n_prompt = len(component.tokenizer.tokenize(text))
        # per-request sampling overrides; sequence_length bounds decoding
        gen_lock.acquire()
        saved = (component.temperature, component.top_k, component.top_p,
                 component.sequence_length)
        try:
            if req.temperature is not None:
                component.temperature = req.temperature
            if req.top_k is not None:
                component.top_k = req.top_k
            if req.top_p is not None:
                component.top_p = req.top_p
            component.sequence_length = min(
                saved[3], n_prompt + req.max_new_tokens)
            t0 = time.perf_counter()
            out = component.generate_tokens(text)
            latency = (time.perf_counter() - t0) * 1000
        finally:
            (component.temperature, component.top_k, component.top_p,
             component.sequence_length) = saved
            gen_lock.release()
        return {"text": out,
                "prompt_tokens": n_prompt,
                "generated_tokens": len(component.tokenizer.tokenize(out)),
                "latency_ms": round(latency, 2)}

    return app


def serve(component: TextInferenceComponent, host: str = "127.0.0.1",
          port: int = 8000):
    import uvicorn
    uvicorn.run(build_app(component), host=host, port=port)
