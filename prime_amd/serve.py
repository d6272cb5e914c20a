"""Local OpenAI-style inference server — the engine-side counterpart of
the reference's hosted inference surface (prime_cli api/inference.py:
list models + chat completion against api.pinference.ai; here the
models run on the local MI355X through the hipGraph decode path).

Endpoints: GET /health, GET /v1/models, POST /v1/completions,
POST /v1/chat/completions. Prompts are either text (requires a local
tokenizer.json) or raw token-id lists. Requests are served one at a
time under a lock (the decode session caches its hipGraph per
(batch, length-bucket); concurrent capture would race).
"""
from __future__ import annotations

import threading
import time
import uuid
from typing import Optional, Union

import torch
from pydantic import BaseModel


class CompletionRequest(BaseModel):
    model: Optional[str] = None
    prompt: Union[str, list[int]]
    max_tokens: int = 64
    temperature: float = 0.0
    top_k: int = 0
    seed: Optional[int] = None


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatRequest(BaseModel):
    model: Optional[str] = None
    messages: list[ChatMessage]
    max_tokens: int = 64
    temperature: float = 0.0
    top_k: int = 0
    seed: Optional[int] = None


def create_app(model, model_name: str, tokenizer=None):
    from fastapi import FastAPI, HTTPException

    from .models.generate import generate

    app = FastAPI(title="prime-amd inference")
    lock = threading.Lock()
    dev = next(model.parameters()).device

    def _encode(prompt: Union[str, list[int]]) -> list[int]:
        if isinstance(prompt, str):
            if tokenizer is None:
                raise HTTPException(
                    400, "text prompts need --tokenizer; send token ids")
            from .utils.tokenizer import encode

            return encode(tokenizer, prompt)
        return [int(t) for t in prompt]

    def _decode(ids: list[int]) -> Union[str, list[int]]:
        if tokenizer is None:
            return ids
        from .utils.tokenizer import decode

        return decode(tokenizer, ids)

    def _run(ids: list[int], req) -> dict:
        if not ids:
            raise HTTPException(400, "empty prompt")
        if not 1 <= req.max_tokens <= 8192:
            raise HTTPException(400, "max_tokens must be in [1, 8192]")
        toks = torch.tensor([ids], device=dev)
        with lock:
            t0 = time.perf_counter()
            out = generate(model, toks, req.max_tokens,
                           temperature=req.temperature, top_k=req.top_k,
                           seed=req.seed)
            dt = time.perf_counter() - t0
        new = [int(t) for t in out[0][len(ids):]]
        return {
            "text": _decode(new),
            "prompt_tokens": len(ids),
            "completion_tokens": len(new),
            "elapsed_s": dt,
        }

    @app.get("/health")
    def health():
        return {"status": "ok", "model": model_name, "device": str(dev)}

    @app.get("/v1/models")
    def models():
        from .models import CONFIGS

        return {"object": "list", "data": [
            {"id": model_name, "object": "model", "owned_by": "prime-amd",
             "loaded": True},
            *({"id": n, "object": "model", "owned_by": "prime-amd",
               "loaded": False} for n in CONFIGS if n != model_name),
        ]}

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        r = _run(_encode(req.prompt), req)
        return {
            "id": f"cmpl-{uuid.uuid4().hex[:12]}",
            "object": "text_completion",
            "created": int(time.time()),
            "model": model_name,
            "choices": [{"index": 0, "text": r["text"],
                         "finish_reason": "length"}],
            "usage": {"prompt_tokens": r["prompt_tokens"],
                      "completion_tokens": r["completion_tokens"],
                      "total_tokens": r["prompt_tokens"] + r["completion_tokens"]},
        }

    @app.post("/v1/chat/completions")
    def chat(req: ChatRequest):
        if tokenizer is None:
            raise HTTPException(400, "chat needs --tokenizer (text prompts)")
        prompt = "".join(f"<|{m.role}|>\n{m.content}\n" for m in req.messages)
        prompt += "<|assistant|>\n"
        creq = CompletionRequest(prompt=prompt, max_tokens=req.max_tokens,
                                 temperature=req.temperature, top_k=req.top_k,
                                 seed=req.seed)
        r = _run(_encode(prompt), creq)
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:12]}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": model_name,
            "choices": [{"index": 0, "finish_reason": "length",
                         "message": {"role": "assistant", "content": r["text"]}}],
            "usage": {"prompt_tokens": r["prompt_tokens"],
                      "completion_tokens": r["completion_tokens"],
                      "total_tokens": r["prompt_tokens"] + r["completion_tokens"]},
        }

    return app


def load_model_for_serving(model_name: str, checkpoint: str | None = None,
                           device: str | None = None):
    """Build (and optionally warm-start) a model for the server."""
    from .models import build_model

    dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
    m = build_model(model_name)
    if dev == "cuda":
        m = m.to(dev, dtype=torch.bfloat16)
    m.reset_rope(torch.device(dev))
    if checkpoint:
        from .ckpt import CheckpointManager
        from .parallel.flat import FlatParamSpace

        flat = FlatParamSpace(m)
        payload = CheckpointManager(checkpoint).load(map_location=dev)
        if payload is None:
            raise FileNotFoundError(f"no checkpoint under {checkpoint}")
        flat.load_flat_(payload["tensors"]["master32"].to(dev))
    m.eval()
    return m
