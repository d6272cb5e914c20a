"""Evaluation: held-out loss / perplexity over a token stream (the
training-engine counterpart of the reference's `prime eval` surface)."""
from __future__ import annotations

import torch

from ..data import DataConfig, build_dataloader
from .llama import Llama


@torch.no_grad()
def evaluate_perplexity(
    model: Llama,
    data_cfg: DataConfig,
    n_batches: int = 10,
    device: torch.device | str = "cpu",
) -> dict:
    model.eval()
    loader = build_dataloader(data_cfg, model.cfg.vocab_size, shard=0, n_shards=1)
    total_loss, total_tokens = 0.0, 0
    for _ in range(n_batches):
        x, y = loader.next_batch(torch.device(device))
        loss = model.loss(x, y)
        ntok = y.numel()
        total_loss += float(loss) * ntok
        total_tokens += ntok
    mean = total_loss / max(1, total_tokens)
    return {
        "loss": mean,
        "perplexity": float(torch.exp(torch.tensor(mean))),
        "tokens": total_tokens,
        "batches": n_batches,
    }
