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
    temperature == 0). With use_graph on CUDA, the per-token decode pass
    runs as a cached hipGraph replay (capture amortizes across calls of
    the same (batch, length-bucket))."""
    model.eval()
    cfg = model.cfg
    dev = tokens.device
    dtype = next(model.parameters()).dtype
    B, L0 = tokens.shape
    total = L0 + max_new_tokens
    if total > cfg.max_seq:
        raise ValueError(f"{total} tokens exceeds max_seq {cfg.max_seq}")
    graphing = use_graph and tokens.is_cuda and max_new_tokens > 2
    # bucket the cache length so repeat calls reuse the captured graph
    smax = min(_pad64(cfg.max_seq), _pad64(max(total, 1024))) if graphing \
        else _pad64(total)
    sessions = getattr(model, "_decode_sessions", None)
    if sessions is None:
        sessions = model._decode_sessions = {}
    key = (B, smax, dev.index)
    sess = None
    if graphing and key in sessions:
        sess = sessions[key]
        caches = sess["caches"]
    else:
        caches = [
            (
                torch.zeros(B, smax, cfg.n_kv_heads, cfg.head_dim, device=dev, dtype=dtype),
                torch.zeros(B, smax, cfg.n_kv_heads, cfg.head_dim, device=dev, dtype=dtype),
            )
            for _ in range(cfg.n_layers)
        ]
        if graphing:
            # capture BEFORE prefill: the warmup/capture passes write junk
            # rows into the cache, which the prefill then overwrites
            sess = sessions[key] = _build_session(model, caches, B, dev)
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
    if sess is not None and max_new_tokens > 1:
        rest = _decode_graphed(sess, cur, pos, max_new_tokens - 1,
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


def _build_session(model, caches, B, dev):
    """Capture the per-token decode pass (embed -> n_layers x {norms, qkv
    GEMM, rope, cache write, decode attention, o/mlp GEMMs} -> lm_head) as
    ONE replayable hipGraph. Dynamic state (position, cache length) lives
    in device scalars the kernels read at run time; the graph itself
    increments them. Capture cost amortizes across generate() calls."""
    pos_dev = {
        "pos32": torch.zeros((), dtype=torch.int32, device=dev),
        "pos64": torch.zeros(1, dtype=torch.int64, device=dev),
        "len32": torch.ones((), dtype=torch.int32, device=dev),
    }
    static_tok = torch.zeros(B, 1, dtype=torch.int64, device=dev)

    def step():
        h = model(static_tok, caches=caches, pos=0, pos_dev=pos_dev)
        logits = model.lm_head(h[:, 0])
        pos_dev["pos32"].add_(1)
        pos_dev["pos64"].add_(1)
        pos_dev["len32"].add_(1)
        return logits

    # warmup on a side stream (per torch CUDA-graphs contract)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):
            step()
    torch.cuda.current_stream().wait_stream(s)
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        static_logits = step()
    return {"caches": caches, "graph": graph, "pos_dev": pos_dev,
            "static_tok": static_tok, "static_logits": static_logits}


def _decode_graphed(sess, first_tok, pos0, n_tokens, temperature, top_k, gen):
    pos_dev = sess["pos_dev"]
    pos_dev["pos32"].fill_(pos0)
    pos_dev["pos64"].fill_(pos0)
    pos_dev["len32"].fill_(pos0 + 1)
    sess["static_tok"].copy_(first_tok)
    graph, static_logits = sess["graph"], sess["static_logits"]
    out = []
    for _ in range(n_tokens):
        graph.replay()
        if temperature <= 0:
            nxt = static_logits.argmax(-1, keepdim=True)  # stays on device
        else:
            nxt = _sample(static_logits, temperature, top_k, gen)
        out.append(nxt)
        sess["static_tok"].copy_(nxt)
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
