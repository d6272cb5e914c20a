"""Run metrics: tokens/sec, MFU, loss — written as JSONL per run so the CLI
`metrics` verb can render them (reference: prime train metrics view)."""
from __future__ import annotations

import json
import time
from pathlib import Path

# MI355X peak DENSE bf16 matrix throughput per GPU. AMD's ~5 PF marketing
# number includes 2:1 structured sparsity; the dense figure (what training
# can use) is ~2.5 PF and was measured at 2495 TF by MFMA microbenchmarks
# (/opt/skills/guides/MI355X_MICROARCH.md). MFU convention per BASELINE.md.
MI355X_PEAK_BF16_DENSE = 2.5e15


def model_flops_per_token(cfg, seq_len: int) -> float:
    """Training (fwd+bwd) FLOPs per token: 6*N_matmul + causal attention."""
    return cfg.flops_per_token() + cfg.attn_flops_per_token(seq_len) * cfg.n_layers


def mfu(tokens_per_sec_per_gpu: float, flops_per_token: float,
        peak: float = MI355X_PEAK_BF16_DENSE) -> float:
    return tokens_per_sec_per_gpu * flops_per_token / peak


class MetricsWriter:
    def __init__(self, path: str | Path | None):
        self._fh = open(path, "a") if path else None

    def write(self, step: int, **metrics) -> None:
        if not self._fh:
            return
        rec = {"step": step, "time": time.time(), **metrics}
        self._fh.write(json.dumps(rec) + "\n")
        self._fh.flush()

    def close(self) -> None:
        if self._fh:
            self._fh.close()
            self._fh = None


def read_metrics(path: str | Path) -> list[dict]:
    out = []
    p = Path(path)
    if not p.exists():
        return out
    for line in p.read_text().splitlines():
        try:
            out.append(json.loads(line))
        except json.JSONDecodeError:
            continue
    return out


class WandbWriter:
    """Optional Weights & Biases sink (reference forwards W&B config to its
    trainer; here it is a local opt-in). No-op unless wandb is importable
    AND enabled in the run config."""

    def __init__(self, enabled: bool, run_name: str, config: dict | None = None):
        self._run = None
        if not enabled:
            return
        try:
            import wandb  # noqa: F401

            self._run = wandb.init(project="prime-amd", name=run_name,
                                   config=config or {})
        except Exception:  # noqa: BLE001 — wandb absent/offline: stay silent
            self._run = None

    def write(self, step: int, **metrics) -> None:
        if self._run is not None:
            self._run.log(metrics, step=step)

    def close(self) -> None:
        if self._run is not None:
            self._run.finish()
