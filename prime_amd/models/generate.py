"""KV-cache generation (serving path): causal-flash prefill + the
memory-bound decode kernel per new token.

Prompts are right-padded to a multiple of 64 for the prefill kernel
(pad keys sit at positions AFTER every real query, so causal masking
keeps them out of real rows); the caches are then truncated to the true
prompt length before decoding.
"""
from __future__ import annotations

import torch

from .llama import Llama


def _pad64(n: int) -> int:
    return (n + 63) // 64 * 64


@torch.no_grad()
def generate(
    model: Llama,
    tokens: torch.Tensor,
    max_new_tokens: int,
    temperature: float = 0.0,
    top_k: int = 0,
    seed: int | None = None,
) -> torch.Tensor:
    """tokens [B, L0] -> [B, L0 + max_new_tokens] (greedy when
    temperature == 0)."""
    model.eval()
    cfg = model.cfg
    dev = tokens.device
    dtype = next(model.parameters()).dtype
    B, L0 = tokens.shape
    total = L0 + max_new_tokens
    if total > cfg.max_seq:
        raise ValueError(f"{total} tokens exceeds max_seq {cfg.max_seq}")
    smax = _pad64(total)
    caches = [
        (
            torch.zeros(B, smax, cfg.n_kv_heads, cfg.head_dim, device=dev, dtype=dtype),
            torch.zeros(B, smax, cfg.n_kv_heads, cfg.head_dim, device=dev, dtype=dtype),
        )
        for _ in range(cfg.n_layers)
    ]
    gen = torch.Generator(device="cpu")
    if seed is not None:
        gen.manual_seed(seed)

    # ---- prefill (padded to 64; pad keys are causally invisible)
    Lp = _pad64(L0)
    padded = torch.zeros(B, Lp, dtype=tokens.dtype, device=dev)
    padded[:, :L0] = tokens
    h = model(padded, caches=caches, pos=0)  # caches filled for [0, Lp)
    logits = model.lm_head(h[:, L0 - 1])

    out = [tokens]
    cur = _sample(logits, temperature, top_k, gen)
    out.append(cur)
    pos = L0
    for _ in range(max_new_tokens - 1):
        h = model(cur, caches=caches, pos=pos)
        logits = model.lm_head(h[:, 0])
        cur = _sample(logits, temperature, top_k, gen)
        out.append(cur)
        pos += 1
    return torch.cat(out, dim=1)


def _sample(logits: torch.Tensor, temperature: float, top_k: int, gen) -> torch.Tensor:
    if temperature <= 0:
        return logits.argmax(-1, keepdim=True)
    logits = logits.float() / temperature
    if top_k > 0:
        kth = logits.topk(top_k, dim=-1).values[..., -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    probs = logits.softmax(-1)
    idx = torch.multinomial(probs.cpu(), 1, generator=gen).to(logits.device)
    return idx
