"""int8-quantized ring all-reduce for DiLoCo pseudo-gradients.

Algorithm: blockwise-int8 quantize each hop's payload,
reduce-scatter ring (W-1 hops, dequant+accumulate in fp32 at every hop),
then all-gather ring of the reduced partitions (quantized once). Comm
volume ≈ 2·N·(1 + 1/256) bytes vs 8·N for fp32 — a 4× cut sized for the
7×153 GB/s xGMI links (and for WAN TCP in the elastic path).

Works over any torch.distributed ProcessGroup: RCCL (GPU, int8 P2P over
xGMI) or gloo (CPU plumbing / elastic cross-worker).
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from .. import ops
from ..ops import QBLK


def _quant(x32: torch.Tensor):
    if x32.is_cuda:
        return ops.quant_int8(x32)
    return ops.reference.quant_int8_blockwise(x32, QBLK)


def _dequant_add(q, scales, dst, accumulate: bool):
    if dst.is_cuda:
        ops.dequant_int8(q, scales, dst, accumulate)
    else:
        v = ops.reference.dequant_int8_blockwise(q, scales, QBLK)
        if accumulate:
            dst.add_(v)
        else:
            dst.copy_(v)


def ring_allreduce_int8(
    delta: torch.Tensor,
    group: dist.ProcessGroup | None = None,
    average: bool = True,
) -> None:
    """In-place sum (or average) of fp32 `delta` across `group` with int8
    compression. Requires delta.numel() % (W * QBLK) == 0 — callers pad
    (FlatParamSpace buffers are padded by DilocoOptimizer)."""
    W = dist.get_world_size(group)
    if W == 1:
        return
    rank = dist.get_rank(group)
    n = delta.numel()
    assert n % (W * QBLK) == 0, f"delta size {n} not divisible by W*QBLK={W * QBLK}"
    part = n // W
    parts = [delta[i * part : (i + 1) * part] for i in range(W)]
    nxt = (rank + 1) % W
    prv = (rank - 1) % W
    # global ranks for P2P
    nxt_g = dist.get_global_rank(group, nxt) if group is not None else nxt
    prv_g = dist.get_global_rank(group, prv) if group is not None else prv

    recv_q = torch.empty(part, dtype=torch.int8, device=delta.device)
    recv_s = torch.empty(part // QBLK, dtype=torch.float32, device=delta.device)

    # ---- reduce-scatter ring
    for step in range(W - 1):
        send_idx = (rank - step) % W
        recv_idx = (rank - step - 1) % W
        q, s = _quant(parts[send_idx])
        q = q.contiguous()
        s = s.contiguous()
        p2p = [
            dist.P2POp(dist.isend, q, nxt_g, group),
            dist.P2POp(dist.isend, s, nxt_g, group),
            dist.P2POp(dist.irecv, recv_q, prv_g, group),
            dist.P2POp(dist.irecv, recv_s, prv_g, group),
        ]
        for w in dist.batch_isend_irecv(p2p):
            w.wait()
        _dequant_add(recv_q, recv_s, parts[recv_idx], accumulate=True)
    # rank now owns the fully-reduced partition (rank+1) % W
    own = (rank + 1) % W
    if average:
        parts[own].div_(W)
    # ---- all-gather ring (quantize the owned partition once; forward hops
    # re-send the received payload so every rank applies identical values)
    send_q, send_s = _quant(parts[own])
    send_q = send_q.contiguous()
    send_s = send_s.contiguous()
    _dequant_add(send_q, send_s, parts[own], accumulate=False)  # self-consistency
    for step in range(W - 1):
        recv_idx = (rank - step) % W
        p2p = [
            dist.P2POp(dist.isend, send_q, nxt_g, group),
            dist.P2POp(dist.isend, send_s, nxt_g, group),
            dist.P2POp(dist.irecv, recv_q, prv_g, group),
            dist.P2POp(dist.irecv, recv_s, prv_g, group),
        ]
        for w in dist.batch_isend_irecv(p2p):
            w.wait()
        _dequant_add(recv_q, recv_s, parts[recv_idx], accumulate=False)
        send_q, recv_q = recv_q.clone(), send_q
        send_s, recv_s = recv_s.clone(), send_s


def allreduce_fp32(delta: torch.Tensor, group=None, average: bool = True) -> None:
    dist.all_reduce(delta, group=group)
    if average:
        delta.div_(dist.get_world_size(group))
