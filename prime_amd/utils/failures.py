"""Failure classification for runs (reference: RLRun.failure_analysis —
automated failure categorization rendered by the CLI)."""
from __future__ import annotations


class NonFiniteLossError(RuntimeError):
    """Loss became NaN/Inf during training."""


def classify_failure(exc: BaseException) -> dict:
    text = f"{type(exc).__name__}: {exc}"
    low = text.lower()
    if isinstance(exc, NonFiniteLossError) or "nan" in low or "non-finite" in low:
        cat, hint = "NON_FINITE_LOSS", (
            "loss diverged: lower optim.lr, raise optim.warmup_steps, or "
            "check data; resume from the last checkpoint"
        )
    elif "out of memory" in low or "outofmemory" in low or "hip error" in low and "memory" in low:
        cat, hint = "OOM", (
            "reduce data.micro_batch_size or model.seq_len, enable "
            "model.activation_checkpointing, or set diloco.outer_device='host'"
        )
    elif any(k in low for k in ("nccl", "rccl", "gloo", "connection", "timed out", "timeout", "rendezvous")):
        cat, hint = "COMM", (
            "a peer died or the network stalled; elastic runs evict the dead "
            "worker at the next outer boundary — restart the worker to rejoin"
        )
    elif "invalid config" in low or "configerror" in low:
        cat, hint = "CONFIG", "fix the TOML config (see the error detail)"
    elif isinstance(exc, KeyboardInterrupt):
        cat, hint = "INTERRUPTED", "stopped by user"
    else:
        cat, hint = "UNKNOWN", "see the traceback in the run logs"
    return {"category": cat, "hint": hint, "error": text[:500]}
