"""Pure-PyTorch fp32 reference implementations of every fused op.

These are (a) the CPU execution path (the gloo plumbing config runs without
GPUs), and (b) the numerics oracle the HIP kernels are tested against
(tests/test_ops_gpu.py compares each kernel to these at fp32).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd * w.float()).to(x.dtype)


def rope_tables(
    dim: int, max_seq: int, theta: float = 10000.0, device=None
) -> tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [max_seq, dim/2] fp32 (host-precomputed: guide App. B)."""
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device).float() / dim))
    t = torch.arange(max_seq, device=device).float()
    freqs = torch.outer(t, inv)  # [S, dim/2]
    return freqs.cos(), freqs.sin()


def apply_rope(
    x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, pos_offset: int = 0
) -> torch.Tensor:
    """NeoX half-split rotation. x: [B, S, H, D]."""
    B, S, H, D = x.shape
    half = D // 2
    c = cos[pos_offset : pos_offset + S].view(1, S, 1, half)
    s = sin[pos_offset : pos_offset + S].view(1, S, 1, half)
    x1, x2 = x[..., :half].float(), x[..., half:].float()
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


def swiglu(gu: torch.Tensor) -> torch.Tensor:
    g, u = gu.chunk(2, dim=-1)
    return (F.silu(g.float()) * u.float()).to(gu.dtype)


def attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, causal: bool = True
) -> torch.Tensor:
    """q: [B,S,H,D], k/v: [B,S,Hkv,D] -> [B,S,H,D]. fp32 math."""
    B, S, H, D = q.shape
    Hkv = k.shape[2]
    if Hkv != H:
        k = k.repeat_interleave(H // Hkv, dim=2)
        v = v.repeat_interleave(H // Hkv, dim=2)
    qf = q.float().transpose(1, 2)  # [B,H,S,D]
    kf = k.float().transpose(1, 2)
    vf = v.float().transpose(1, 2)
    scores = qf @ kf.transpose(-1, -2) / (D**0.5)
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    p = scores.softmax(-1)
    return (p @ vf).transpose(1, 2).to(q.dtype)


def cross_entropy(
    logits: torch.Tensor, targets: torch.Tensor, ignore_index: int = -100
) -> torch.Tensor:
    return F.cross_entropy(logits.float(), targets, ignore_index=ignore_index)


def adamw_step(
    p32: torch.Tensor,
    g: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    wd: float,
    step: int,
) -> None:
    """In-place fused-AdamW reference on fp32 master (decoupled decay)."""
    gf = g.float()
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    c1 = 1.0 / (1.0 - beta1**step)
    c2 = 1.0 / (1.0 - beta2**step)
    upd = (m * c1) / ((v * c2).sqrt() + eps) + wd * p32
    p32.add_(upd, alpha=-lr)


def quant_int8_blockwise(x: torch.Tensor, qblk: int = 1024):
    """Blockwise symmetric int8: returns (q int8, scales fp32 per block)."""
    n = x.numel()
    nblk = (n + qblk - 1) // qblk
    pad = nblk * qblk - n
    xf = x.float().flatten()
    if pad:
        xf = torch.cat([xf, xf.new_zeros(pad)])
    xb = xf.view(nblk, qblk)
    scales = xb.abs().amax(dim=1) / 127.0
    inv = torch.where(scales > 0, 1.0 / scales, torch.zeros_like(scales))
    q = (xb * inv.unsqueeze(1)).round().clamp(-127, 127).to(torch.int8)
    return q.flatten()[:n], scales


def dequant_int8_blockwise(
    q: torch.Tensor, scales: torch.Tensor, qblk: int = 1024
) -> torch.Tensor:
    n = q.numel()
    idx = torch.arange(n, device=q.device) // qblk
    return q.float() * scales[idx]
