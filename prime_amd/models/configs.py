"""Llama-family model configs (150M / 1B / 8B / INTELLECT-1
10B / 70B). The INTELLECT-1 config is the headline benchmark model
(BASELINE.json: 10B DiLoCo H=100)."""
from __future__ import annotations

from dataclasses import dataclass, asdict


@dataclass
class LlamaConfig:
    name: str = "llama"
    dim: int = 768
    n_layers: int = 12
    n_heads: int = 12
    n_kv_heads: int = 12
    intermediate: int = 2048
    vocab_size: int = 32000
    max_seq: int = 4096
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    tie_embeddings: bool = False

    @property
    def head_dim(self) -> int:
        return self.dim // self.n_heads

    def n_params(self) -> int:
        emb = self.vocab_size * self.dim
        per_layer = (
            self.dim * (self.n_heads + 2 * self.n_kv_heads) * self.head_dim  # qkv
            + self.dim * self.dim  # o
            + 3 * self.dim * self.intermediate  # gate/up/down
            + 2 * self.dim  # norms
        )
        total = emb + self.n_layers * per_layer + self.dim
        if not self.tie_embeddings:
            total += self.vocab_size * self.dim
        return total

    def flops_per_token(self) -> float:
        """Training FLOPs/token (fwd+bwd ≈ 6*N_matmul + attention term)."""
        n_mat = self.n_params() - (1 if self.tie_embeddings else 2) * self.vocab_size * self.dim
        # attention: fwd 4*S*hd*H per token -> filled in by caller w/ seq len
        return 6.0 * n_mat

    def attn_flops_per_token(self, seq_len: int, causal: bool = True) -> float:
        """MODEL attention FLOPs/token: fwd QK^T+PV = 4*S*hd*H, x3 for
        fwd+bwd (the standard 1+2 convention), /2 causal.

        Convention note: the flash backward IMPLEMENTATION performs ~2.5x
        the forward's matmuls (dq/dkv recompute S and dP), so the MFU
        numerator deliberately under-counts what the kernels execute —
        recomputation is not useful model work. Combined with the 2.5 PF
        DENSE peak denominator (AMD's 5 PF figure is 2:1-sparse), every
        MFU printed by this repo is a conservative bound."""
        f = 4.0 * seq_len * self.head_dim * self.n_heads * 3.0
        return f / 2.0 if causal else f

    def to_dict(self) -> dict:
        return asdict(self)


CONFIGS: dict[str, LlamaConfig] = {
    "llama_150m": LlamaConfig(
        name="llama_150m", dim=768, n_layers=12, n_heads=12, n_kv_heads=12,
        intermediate=2048, vocab_size=32000, max_seq=2048, rope_theta=10000.0,
    ),
    "llama_1b": LlamaConfig(
        name="llama_1b", dim=2048, n_layers=16, n_heads=16, n_kv_heads=8,
        intermediate=8192, vocab_size=128256, max_seq=8192,
    ),
    "llama_8b": LlamaConfig(
        name="llama_8b", dim=4096, n_layers=32, n_heads=32, n_kv_heads=8,
        intermediate=14336, vocab_size=128256, max_seq=8192,
    ),
    # INTELLECT-1: Llama-3-8B architecture deepened to 42 layers (~10B params)
    "intellect_10b": LlamaConfig(
        name="intellect_10b", dim=4096, n_layers=42, n_heads=32, n_kv_heads=8,
        intermediate=14336, vocab_size=128256, max_seq=8192,
    ),
    "llama_70b": LlamaConfig(
        name="llama_70b", dim=8192, n_layers=80, n_heads=64, n_kv_heads=8,
        intermediate=28672, vocab_size=128256, max_seq=8192,
    ),
    # tiny config for unit tests
    "llama_test": LlamaConfig(
        name="llama_test", dim=64, n_layers=2, n_heads=4, n_kv_heads=2,
        intermediate=128, vocab_size=256, max_seq=256, rope_theta=10000.0,
    ),
}


def get_config(name: str, **overrides) -> LlamaConfig:
    if name not in CONFIGS:
        raise KeyError(f"unknown model config '{name}' (have: {sorted(CONFIGS)})")
    cfg = CONFIGS[name]
    if overrides:
        d = cfg.to_dict()
        d.update(overrides)
        cfg = LlamaConfig(**d)
    return cfg
