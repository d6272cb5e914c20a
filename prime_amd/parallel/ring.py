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


def _pad_run(delta: torch.Tensor, align: int, fn) -> bool:
    """Run `fn(buf)` on a zero-padded copy of `delta` when its size is not
    a multiple of `align`, copying the result back. Keeps the ring usable
    for ANY live world size (an elastic fleet can land on 3/5/6/7 workers
    exactly when fault tolerance matters), not just divisors of the
    construction-time alignment. Returns True when it ran padded."""
    n = delta.numel()
    if n % align == 0:
        return False
    pad_n = (n + align - 1) // align * align
    buf = torch.zeros(pad_n, dtype=delta.dtype, device=delta.device)
    buf[:n].copy_(delta)
    fn(buf)
    delta.copy_(buf[:n])
    return True


def ring_allreduce_int8(
    delta: torch.Tensor,
    group: dist.ProcessGroup | None = None,
    average: bool = True,
) -> None:
    """In-place sum (or average) of fp32 `delta` across `group` with int8
    compression. Sizes not divisible by W*QBLK are zero-padded per call."""
    W = dist.get_world_size(group)
    if W == 1:
        return
    rank = dist.get_rank(group)
    n = delta.numel()
    if _pad_run(delta, W * QBLK, lambda b: ring_allreduce_int8(b, group, average)):
        return
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


def ring_allreduce_int8_multi(
    delta: torch.Tensor,
    group: dist.ProcessGroup | None = None,
    average: bool = True,
    n_rings: int | None = None,
) -> None:
    """Multi-ring int8 all-reduce: the buffer is split into `n_rings`
    sub-buffers, each reduced over a DIFFERENT ring orientation (ring r
    steps by r+1 ranks), so every rank drives up to n_rings xGMI
    point-to-point links concurrently instead of serializing on one
    (xGMI is 7 p2p links/GPU at ~153 GB/s each — a single ring is
    per-link bound). All rings' sends/recvs for a hop are
    issued in ONE batch_isend_irecv so RCCL can overlap them.

    Ring r's step offset is o = r+1; that ring's "next" is (rank+o) % W.
    gcd(o, W) != 1 would split the ring into cycles, so only coprime
    offsets are used (W=8: o in {1,3,5,7} -> up to 4 concurrent rings).
    """
    W = dist.get_world_size(group)
    if W == 1:
        return
    rank = dist.get_rank(group)
    offsets = [o for o in range(1, W) if _gcd(o, W) == 1]
    if n_rings is not None:
        offsets = offsets[:n_rings]
    R = len(offsets)
    if R <= 1:
        return ring_allreduce_int8(delta, group, average)
    n = delta.numel()
    if _pad_run(delta, R * W * QBLK,
                lambda b: ring_allreduce_int8_multi(b, group, average, n_rings)):
        return
    sub = n // R
    subs = [delta[r * sub : (r + 1) * sub] for r in range(R)]
    part = sub // W
    g = lambda r: dist.get_global_rank(group, r) if group is not None else r

    recv_q = [torch.empty(part, dtype=torch.int8, device=delta.device) for _ in range(R)]
    recv_s = [torch.empty(part // QBLK, dtype=torch.float32, device=delta.device) for _ in range(R)]

    # ---- reduce-scatter: W-1 hops; each hop issues all rings together
    for step in range(W - 1):
        p2p, metas = [], []
        for ri, o in enumerate(offsets):
            # ring ri in the basis of offset o: logical position of this
            # rank is rank * inv(o) ... simpler: walk indices directly:
            # at step s, send partition owned (start - s) in ring order.
            nxt, prv = (rank + o) % W, (rank - o) % W
            send_idx = _ring_pos(rank, o, W, -step)
            recv_idx = _ring_pos(rank, o, W, -step - 1)
            q, sc = _quant(subs[ri][send_idx * part : (send_idx + 1) * part])
            q, sc = q.contiguous(), sc.contiguous()
            p2p += [
                dist.P2POp(dist.isend, q, g(nxt), group),
                dist.P2POp(dist.isend, sc, g(nxt), group),
                dist.P2POp(dist.irecv, recv_q[ri], g(prv), group),
                dist.P2POp(dist.irecv, recv_s[ri], g(prv), group),
            ]
            metas.append((ri, recv_idx))
        for w in dist.batch_isend_irecv(p2p):
            w.wait()
        for ri, recv_idx in metas:
            _dequant_add(recv_q[ri], recv_s[ri],
                         subs[ri][recv_idx * part : (recv_idx + 1) * part],
                         accumulate=True)
    # ---- all-gather: W-1 hops
    send_q, send_s = [], []
    for ri, o in enumerate(offsets):
        own = _ring_pos(rank, o, W, 1)
        seg = subs[ri][own * part : (own + 1) * part]
        if average:
            seg.div_(W)
        q, sc = _quant(seg)
        q, sc = q.contiguous(), sc.contiguous()
        _dequant_add(q, sc, seg, accumulate=False)
        send_q.append(q)
        send_s.append(sc)
    for step in range(W - 1):
        p2p, metas = [], []
        for ri, o in enumerate(offsets):
            nxt, prv = (rank + o) % W, (rank - o) % W
            recv_idx = _ring_pos(rank, o, W, -step)
            p2p += [
                dist.P2POp(dist.isend, send_q[ri], g(nxt), group),
                dist.P2POp(dist.isend, send_s[ri], g(nxt), group),
                dist.P2POp(dist.irecv, recv_q[ri], g(prv), group),
                dist.P2POp(dist.irecv, recv_s[ri], g(prv), group),
            ]
            metas.append((ri, recv_idx))
        for w in dist.batch_isend_irecv(p2p):
            w.wait()
        for ri, recv_idx in metas:
            _dequant_add(recv_q[ri], recv_s[ri],
                         subs[ri][recv_idx * part : (recv_idx + 1) * part],
                         accumulate=False)
            send_q[ri], recv_q[ri] = recv_q[ri].clone(), send_q[ri]
            send_s[ri], recv_s[ri] = recv_s[ri].clone(), send_s[ri]


def _gcd(a: int, b: int) -> int:
    while b:
        a, b = b, a % b
    return a


def _ring_pos(rank: int, offset: int, W: int, hop: int) -> int:
    """Partition index held by `rank` after `hop` logical ring steps in the
    ring with step `offset`. In ring coordinates u = position of rank
    (rank = u*offset mod W), the standard schedule assigns partition
    ((u + hop) mod W) mapped back to partition ids = rank-space ids."""
    # position u of this rank in ring order: u * offset ≡ rank (mod W)
    u = (rank * _modinv(offset, W)) % W
    v = (u + hop) % W
    return (v * offset) % W


def _modinv(a: int, m: int) -> int:
    # a coprime to m
    x0, x1 = 0, 1
    mm, aa = m, a
    while aa > 1:
        q = aa // mm
        aa, mm = mm, aa - q * mm
        x1, x0 = x0, x1 - q * x0
    return x1 % m
