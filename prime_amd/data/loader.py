"""Data pipeline: synthetic token stream (benchmarks — no network in the
build/bench environment) and a memory-mapped token-file dataset for real
runs. Both are stateful (resumable from checkpoints) and shard by
(worker_id, n_workers, local DP rank) so every DiLoCo worker sees a
disjoint stream.
"""
from __future__ import annotations

from dataclasses import dataclass
from pathlib import Path

import numpy as np
import torch


@dataclass
class DataConfig:
    kind: str = "synthetic"        # synthetic | token_file
    path: str | None = None        # .bin/.npy uint16|uint32 token file
    seq_len: int = 2048
    micro_batch_size: int = 4
    seed: int = 1234
    shuffle: bool = True


class SyntheticTokens:
    """Deterministic pseudo-random token batches; infinite, resumable."""

    def __init__(self, cfg: DataConfig, vocab_size: int, shard: int, n_shards: int):
        self.cfg = cfg
        self.vocab = vocab_size
        self.shard = shard
        self.n_shards = n_shards
        self.batch_idx = 0

    def state_dict(self) -> dict:
        return {"batch_idx": self.batch_idx}

    def load_state_dict(self, sd: dict) -> None:
        self.batch_idx = int(sd["batch_idx"])

    def next_batch(self, device: torch.device) -> tuple[torch.Tensor, torch.Tensor]:
        g = torch.Generator()
        g.manual_seed(self.cfg.seed + self.batch_idx * self.n_shards + self.shard)
        self.batch_idx += 1
        b, s = self.cfg.micro_batch_size, self.cfg.seq_len
        toks = torch.randint(0, self.vocab, (b, s + 1), generator=g)
        x = toks[:, :-1].to(device, non_blocking=True)
        y = toks[:, 1:].to(device, non_blocking=True)
        return x, y


class TokenFileDataset:
    """Memory-mapped flat token file (.bin of uint16/uint32, e.g. a
    pre-tokenized corpus). Windows of seq_len+1 tokens are sampled via a
    seeded per-epoch permutation (shuffle=True) or sequentially; sharded
    by (shard, n_shards); resumable from {epoch, batch_idx}."""

    def __init__(self, cfg: DataConfig, vocab_size: int, shard: int, n_shards: int):
        if not cfg.path:
            raise ValueError("token_file dataset requires data.path")
        p = Path(cfg.path)
        dtype = np.uint16 if vocab_size <= 65535 else np.uint32
        # single flat .bin, or a directory of part_*.bin shards (the
        # streaming prepare-data writer's layout)
        if p.is_dir():
            parts = sorted(p.glob("*.bin"))
            if not parts:
                raise ValueError(f"no .bin shards under {p}")
            self.files = [np.memmap(f, dtype=dtype, mode="r") for f in parts]
        else:
            self.files = [np.memmap(p, dtype=dtype, mode="r")]
        self.cfg = cfg
        self.shard = shard
        self.n_shards = n_shards
        self.batch_idx = 0
        self.epoch = 0
        # per-file window table (windows never straddle shard files)
        self._file_windows = [max(0, (len(f) - 1) // cfg.seq_len) for f in self.files]
        self._file_base = np.cumsum([0] + self._file_windows)
        self.windows = int(self._file_base[-1])
        if self.windows < 1:
            raise ValueError(f"{p} too small for seq_len={cfg.seq_len}")
        self._perm: np.ndarray | None = None
        self._perm_epoch = -1

    def state_dict(self) -> dict:
        return {"batch_idx": self.batch_idx, "epoch": self.epoch}

    def load_state_dict(self, sd: dict) -> None:
        self.batch_idx = int(sd["batch_idx"])
        self.epoch = int(sd.get("epoch", 0))

    def _window(self, i: int) -> int:
        if not self.cfg.shuffle:
            return i % self.windows
        epoch = i // self.windows
        if epoch != self._perm_epoch:
            rng = np.random.default_rng(self.cfg.seed + epoch)
            self._perm = rng.permutation(self.windows)
            self._perm_epoch = epoch
        return int(self._perm[i % self.windows])

    def next_batch(self, device: torch.device) -> tuple[torch.Tensor, torch.Tensor]:
        b, s = self.cfg.micro_batch_size, self.cfg.seq_len
        xs, ys = [], []
        for i in range(b):
            gi = self.batch_idx * self.n_shards * b + self.shard * b + i
            w = self._window(gi)
            fi = int(np.searchsorted(self._file_base, w, side="right")) - 1
            start = (w - int(self._file_base[fi])) * s
            f = self.files[fi]
            chunk = torch.from_numpy(f[start : start + s + 1].astype(np.int64))
            if chunk.numel() < s + 1:  # last window of a shard: wrap pad
                pad = torch.from_numpy(f[: s + 1 - chunk.numel()].astype(np.int64))
                chunk = torch.cat([chunk, pad])
            xs.append(chunk[:-1])
            ys.append(chunk[1:])
        self.batch_idx += 1
        self.epoch = (self.batch_idx * self.n_shards * b) // self.windows
        x = torch.stack(xs).to(device, non_blocking=True)
        y = torch.stack(ys).to(device, non_blocking=True)
        return x, y


def build_dataloader(cfg: DataConfig, vocab_size: int, shard: int, n_shards: int):
    if cfg.kind == "synthetic":
        return SyntheticTokens(cfg, vocab_size, shard, n_shards)
    if cfg.kind == "token_file":
        return TokenFileDataset(cfg, vocab_size, shard, n_shards)
    raise ValueError(f"unknown data.kind '{cfg.kind}'")
