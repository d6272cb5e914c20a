"""Sequence (context) parallelism — Ulysses-style head-scatter attention.

Long-context training splits the sequence across the worker's local ranks
(each rank holds S/W tokens of the SAME batch). Everything token-local
(norms, MLP, embeddings, loss) runs unchanged on the local slice; only
attention needs the full sequence, so around it we do two all-to-alls:

  [B, S/W, H, D]  --a2a-->  [B, S, H/W, D]  --flash-->  --a2a back-->

Each rank then runs the ordinary flash kernel on the FULL sequence for
its H/W head group (GQA: K/V use Hkv/W). On MI355X the a2a rides RCCL
over xGMI point-to-point links — an all-to-all is the per-link-optimal
collective on that fabric (every GPU pair has a direct link).
Parameters stay replicated; gradients average like DP.

gloo (CPU plumbing) has no all_to_all — the fallback runs the same
exchange over batched isend/irecv.
"""
from __future__ import annotations

import torch
import torch.distributed as dist


def _a2a_exchange(chunks: list[torch.Tensor], group) -> list[torch.Tensor]:
    """all_to_all of equal-sized chunks; chunk[j] goes to rank j, the
    returned list holds one chunk from every rank."""
    W = dist.get_world_size(group)
    rank = dist.get_rank(group)
    out = [torch.empty_like(c) for c in chunks]
    if dist.get_backend(group) == "nccl":
        dist.all_to_all(out, [c.contiguous() for c in chunks], group=group)
        return out
    # gloo fallback: batched P2P (self-chunk copied locally)
    out[rank].copy_(chunks[rank])
    p2p = []
    for j in range(W):
        if j == rank:
            continue
        g = dist.get_global_rank(group, j) if group is not None else j
        p2p.append(dist.P2POp(dist.isend, chunks[j].contiguous(), g, group))
        p2p.append(dist.P2POp(dist.irecv, out[j], g, group))
    if p2p:
        for w in dist.batch_isend_irecv(p2p):
            w.wait()
    return out


def _scatter_heads(x: torch.Tensor, group) -> torch.Tensor:
    """[B, S/W, H, D] (local tokens, all heads) -> [B, S, H/W, D]
    (all tokens, local head group)."""
    W = dist.get_world_size(group)
    rank = dist.get_rank(group)
    B, Sl, H, D = x.shape
    assert H % W == 0, f"heads {H} not divisible by sp world {W}"
    hg = H // W
    send = [x[:, :, j * hg : (j + 1) * hg].contiguous() for j in range(W)]
    recv = _a2a_exchange(send, group)
    # recv[j] = rank j's token slice of MY head group; tokens are ordered
    # by rank (rank j holds tokens [j*Sl, (j+1)*Sl))
    return torch.cat(recv, dim=1)


def _gather_heads(x: torch.Tensor, group) -> torch.Tensor:
    """[B, S, H/W, D] -> [B, S/W, H, D] (inverse of _scatter_heads)."""
    W = dist.get_world_size(group)
    B, S, hg, D = x.shape
    Sl = S // W
    send = [x[:, j * Sl : (j + 1) * Sl].contiguous() for j in range(W)]
    recv = _a2a_exchange(send, group)
    return torch.cat(recv, dim=2)


class _ScatterHeads(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _scatter_heads(x, group)

    @staticmethod
    def backward(ctx, g):
        return _gather_heads(g.contiguous(), ctx.group), None


class _GatherHeads(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _gather_heads(x, group)

    @staticmethod
    def backward(ctx, g):
        return _scatter_heads(g.contiguous(), ctx.group), None


def ulysses_attention(q, k, v, causal: bool, group) -> torch.Tensor:
    """Sequence-parallel flash attention. q: [B, S/W, H, D]; k, v:
    [B, S/W, Hkv, D] (GQA: Hkv % W == 0 required). Returns the local
    token slice [B, S/W, H, D]."""
    from .. import ops

    qg = _ScatterHeads.apply(q, group)
    kg = _ScatterHeads.apply(k, group)
    vg = _ScatterHeads.apply(v, group)
    o = ops.flash_attention(qg, kg, vg, causal=causal)
    return _GatherHeads.apply(o, group)
