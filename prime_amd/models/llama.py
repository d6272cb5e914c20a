"""Llama-family transformer assembled from the gfx950 fused ops.

MI355X-first design decisions:
  - qkv and gate/up projections are each ONE hipBLASLt GEMM (fused weights);
    everything between GEMMs (RMSNorm, RoPE, flash attention, SwiGLU,
    cross-entropy) is a hand-written HIP kernel (prime_amd.ops).
  - bf16 parameters/activations, fp32 only where numerics need it
    (norm stats, softmax, loss).
  - per-layer activation checkpointing (non-reentrant) for the big configs:
    288 GB HBM3E prefers recompute over host offload.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint

from .. import ops
from .configs import LlamaConfig, get_config


class PrimeLinear(nn.Linear):
    """nn.Linear (no bias) routed through layout-tuned hipBLASLt calls
    (ops.tuned_linear): NT forward, NT dX against a per-step cached W^T,
    NN dW via an LDS-tiled dY transpose. State-dict compatible with
    nn.Linear."""

    def forward(self, x):
        return ops.tuned_linear(x, self.weight)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return ops.rmsnorm(x, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        hd = cfg.head_dim
        self.wqkv = PrimeLinear(cfg.dim, (cfg.n_heads + 2 * cfg.n_kv_heads) * hd, bias=False)
        self.wo = PrimeLinear(cfg.n_heads * hd, cfg.dim, bias=False)

    # set by the Trainer for sequence-parallel training: attention runs
    # over the full context via two all-to-alls (parallel/seqpar.py)
    sp_group = None

    def forward(self, x, cos, sin, cache=None, pos: int = 0, pos_dev=None):
        B, S, _ = x.shape
        cfg = self.cfg
        hd = cfg.head_dim
        qkv = self.wqkv(x)
        q, k, v = qkv.split(
            [cfg.n_heads * hd, cfg.n_kv_heads * hd, cfg.n_kv_heads * hd], dim=-1
        )
        q = q.view(B, S, cfg.n_heads, hd)
        k = k.view(B, S, cfg.n_kv_heads, hd)
        v = v.view(B, S, cfg.n_kv_heads, hd)
        p32 = pos_dev["pos32"] if pos_dev is not None else None
        q = ops.apply_rope(q, cos, sin, pos_offset=pos, pos_dev=p32)
        k = ops.apply_rope(k, cos, sin, pos_offset=pos, pos_dev=p32)
        if cache is not None:
            kc, vc = cache  # [B, Smax, Hkv, hd]
            if pos_dev is not None:  # graph-capturable dynamic position
                kc.index_copy_(1, pos_dev["pos64"], k)
                vc.index_copy_(1, pos_dev["pos64"], v)
                o = ops.attn_decode(q, kc, vc, length=1,
                                    len_dev=pos_dev["len32"])
                return self.wo(o.reshape(B, 1, cfg.n_heads * hd))
            kc[:, pos : pos + S] = k
            vc[:, pos : pos + S] = v
            if S == 1:  # decode: memory-bound cache-streaming kernel
                o = ops.attn_decode(q, kc, vc, length=pos + 1)
                return self.wo(o.reshape(B, 1, cfg.n_heads * hd))
            # prefill (pos == 0): causal flash over the prompt
            o = ops.flash_attention(q, k, v, causal=True)
            return self.wo(o.reshape(B, S, cfg.n_heads * hd))
        if self.sp_group is not None:
            from ..parallel.seqpar import ulysses_attention

            o = ulysses_attention(q, k, v, causal=True, group=self.sp_group)
        else:
            o = ops.flash_attention(q, k, v, causal=True)
        return self.wo(o.reshape(B, S, cfg.n_heads * hd))


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.w_gateup = PrimeLinear(cfg.dim, 2 * cfg.intermediate, bias=False)
        self.w_down = PrimeLinear(cfg.intermediate, cfg.dim, bias=False)

    def forward(self, x):
        return self.w_down(ops.swiglu(self.w_gateup(x)))


class Block(nn.Module):
    """Pre-norm block threaded through the fused add+RMSNorm op: carries
    (delta, residual) instead of a single stream, so every residual add is
    fused into the following norm's pass (one fewer hidden-stream
    read+write per block side)."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.attn = Attention(cfg)
        self.mlp_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.mlp = MLP(cfg)

    def forward(self, x, res, cos, sin, cache=None, pos: int = 0, pos_dev=None):
        y, s = ops.fused_add_rmsnorm(x, res, self.attn_norm.weight,
                                     self.attn_norm.eps)
        a = self.attn(y, cos, sin, cache=cache, pos=pos, pos_dev=pos_dev)
        y2, s2 = ops.fused_add_rmsnorm(a, s, self.mlp_norm.weight,
                                       self.mlp_norm.eps)
        return self.mlp(y2), s2


class Llama(nn.Module):
    def __init__(self, cfg: LlamaConfig, activation_checkpointing: bool = False):
        super().__init__()
        self.cfg = cfg
        self.activation_checkpointing = activation_checkpointing
        self.tok_embeddings = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.layers = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.lm_head = PrimeLinear(cfg.dim, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.tok_embeddings.weight
        cos, sin = ops.reference.rope_tables(cfg.head_dim, cfg.max_seq, cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.sp_pos_offset = 0  # sequence-parallel rank's token offset
        self.apply(self._init)
        # scaled init for residual-out projections (GPT-2 style)
        std = 0.02 / math.sqrt(2 * cfg.n_layers)
        for blk in self.layers:
            nn.init.normal_(blk.attn.wo.weight, std=std)
            nn.init.normal_(blk.mlp.w_down.weight, std=std)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)

    def reset_rope(self, device=None) -> None:
        """Recompute fp32 RoPE tables (call after .to(dtype=bf16), which
        would otherwise truncate the tables to bf16)."""
        cfg = self.cfg
        cos, sin = ops.reference.rope_tables(
            cfg.head_dim, cfg.max_seq, cfg.rope_theta,
            device=device or self.rope_cos.device,
        )
        self.rope_cos = cos
        self.rope_sin = sin

    def forward(self, tokens: torch.Tensor, caches=None, pos: int = 0,
                pos_dev=None) -> torch.Tensor:
        """tokens [B,S] -> hidden states [B,S,dim] (pre-lm_head).
        caches: optional per-layer (k,v) KV caches for inference; pos_dev:
        device-scalar position dict for hipGraph-captured decode."""
        x = self.tok_embeddings(tokens)
        res = None
        if pos == 0 and self.sp_pos_offset:
            pos = self.sp_pos_offset  # SP rank's slice starts mid-context
        cos, sin = self.rope_cos, self.rope_sin
        for i, blk in enumerate(self.layers):
            if self.activation_checkpointing and self.training:
                x, res = checkpoint(blk, x, res, cos, sin, use_reentrant=False)
            else:
                x, res = blk(x, res, cos, sin,
                             cache=caches[i] if caches else None, pos=pos,
                             pos_dev=pos_dev)
        y, _ = ops.fused_add_rmsnorm(x, res, self.norm.weight, self.norm.eps)
        return y

    def loss(self, tokens: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
        """Fused lm_head + cross-entropy. tokens/targets: [B,S]."""
        h = self.forward(tokens)
        logits = self.lm_head(h).flatten(0, 1)
        return ops.cross_entropy(logits, targets.flatten(), ignore_index=-100)


def build_model(name: str, activation_checkpointing: bool = False, **overrides) -> Llama:
    return Llama(get_config(name, **overrides), activation_checkpointing)
