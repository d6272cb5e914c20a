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
    use_graph: bool = True,
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
    if use_graph and tokens.is_cuda and max_new_tokens > 8:
        rest = _decode_graphed(model, caches, cur, pos, max_new_tokens - 1,
                               temperature, top_k, gen)
        out.extend(rest)
    else:
        for _ in range(max_new_tokens - 1):
            h = model(cur, caches=caches, pos=pos)
            logits = model.lm_head(h[:, 0])
            cur = _sample(logits, temperature, top_k, gen)
            out.append(cur)
            pos += 1
    return torch.cat(out, dim=1)


def _decode_graphed(model, caches, first_tok, pos0, n_tokens, temperature,
                    top_k, gen):
    """hipGraph-captured decode: the whole per-token pass (embed -> 42x
    {norms, qkv GEMM, rope, cache write, decode attention, o/mlp GEMMs} ->
    lm_head) replays as ONE graph launch — the eager loop is launch-bound
    (~6 kernels x n_layers x ~20 us per token). Dynamic state (position,
    cache length) lives in device scalars the kernels read at run time and
    the graph itself increments."""
    dev = first_tok.device
    pos_dev = {
        "pos32": torch.tensor(pos0, dtype=torch.int32, device=dev),
        "pos64": torch.tensor([pos0], dtype=torch.int64, device=dev),
        "len32": torch.tensor(pos0 + 1, dtype=torch.int32, device=dev),
    }
    static_tok = first_tok.clone()

    def step():
        h = model(static_tok, caches=caches, pos=0, pos_dev=pos_dev)
        logits = model.lm_head(h[:, 0])
        pos_dev["pos32"].add_(1)
        pos_dev["pos64"].add_(1)
        pos_dev["len32"].add_(1)
        return logits

    # warmup on a side stream (per torch CUDA-graphs contract), then reset
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):
            step()
            pos_dev["pos32"].fill_(pos0)
            pos_dev["pos64"].fill_(pos0)
            pos_dev["len32"].fill_(pos0 + 1)
    torch.cuda.current_stream().wait_stream(s)

    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        static_logits = step()
    pos_dev["pos32"].fill_(pos0)
    pos_dev["pos64"].fill_(pos0)
    pos_dev["len32"].fill_(pos0 + 1)

    out = []
    for _ in range(n_tokens):
        graph.replay()
        if temperature <= 0:
            nxt = static_logits.argmax(-1, keepdim=True)  # stays on device
        else:
            nxt = _sample(static_logits, temperature, top_k, gen)
        out.append(nxt)
        static_tok.copy_(nxt)
    return out


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
