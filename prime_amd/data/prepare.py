"""Streaming / sharded corpus tokenization for `prime-amd prepare-data`.

The round-1 path read the whole text into memory and produced a single
.bin — fine for samples, unusable for a multi-GB corpus. This module
streams inputs in bounded blocks, tokenizes blocks in parallel worker
processes (order-preserving), and writes fixed-size output shards that
TokenFileDataset consumes directly (a directory of part_*.bin, or one
flat .bin when the corpus fits a single shard).
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import Iterator

import numpy as np

_BLOCK_BYTES = 8 << 20  # 8 MiB of UTF-8 per tokenization block


def iter_blocks(paths: list[Path], block_bytes: int = _BLOCK_BYTES) -> Iterator[str]:
    """Stream text files in ~block_bytes chunks, splitting only at line
    boundaries so tokens never straddle a block cut mid-word."""
    for p in paths:
        with open(p, encoding="utf-8", errors="replace") as fh:
            buf: list[str] = []
            size = 0
            for line in fh:
                buf.append(line)
                size += len(line)
                if size >= block_bytes:
                    yield "".join(buf)
                    buf, size = [], 0
            if buf:
                yield "".join(buf)


_TOK = None


def _worker_init(tokenizer_path: str) -> None:
    global _TOK
    from ..utils.tokenizer import load_tokenizer

    _TOK = load_tokenizer(tokenizer_path)


def _encode_block(text: str) -> np.ndarray:
    return np.asarray(_TOK.encode(text).ids, dtype=np.int64)


class ShardWriter:
    """Append token arrays, rolling to a new part_NNNN.bin every
    `shard_tokens`; a single-shard corpus is renamed to the flat `out`
    path so small cases keep the simple one-file layout."""

    def __init__(self, out: Path, dtype: np.dtype, shard_tokens: int):
        self.out = Path(out)
        self.dtype = dtype
        self.shard_tokens = shard_tokens
        self.dir = self.out if self.out.suffix == "" else self.out.parent
        self.stem = "part" if self.out.suffix == "" else self.out.stem
        self.dir.mkdir(parents=True, exist_ok=True)
        self.total = 0
        self._shard_idx = -1
        self._in_shard = 0
        self._fh = None
        self.paths: list[Path] = []

    def _roll(self) -> None:
        if self._fh:
            self._fh.close()
        self._shard_idx += 1
        p = self.dir / f"{self.stem}_{self._shard_idx:04d}.bin"
        self.paths.append(p)
        self._fh = open(p, "wb")
        self._in_shard = 0

    def write(self, ids: np.ndarray) -> None:
        off = 0
        while off < len(ids):
            if self._fh is None or self._in_shard >= self.shard_tokens:
                self._roll()
            take = min(len(ids) - off, self.shard_tokens - self._in_shard)
            ids[off : off + take].astype(self.dtype).tofile(self._fh)
            self._in_shard += take
            self.total += take
            off += take

    def close(self) -> list[Path]:
        if self._fh:
            self._fh.close()
        # single shard + flat out path requested: keep the simple layout
        if self.out.suffix and len(self.paths) == 1:
            os.replace(self.paths[0], self.out)
            self.paths = [self.out]
        return self.paths


def prepare_corpus(
    inputs: list[str | Path],
    out: str | Path,
    tokenizer_path: str,
    vocab_threshold: int = 65535,
    shard_tokens: int = 512 * 1024 * 1024,
    workers: int = 0,
) -> dict:
    """Tokenize text files into token shards. Returns a summary dict."""
    from ..utils.tokenizer import load_tokenizer

    paths = [Path(p) for p in inputs]
    for p in paths:
        if not p.exists():
            raise FileNotFoundError(p)
    vocab = load_tokenizer(tokenizer_path).get_vocab_size()
    dtype = np.uint16 if vocab <= vocab_threshold else np.uint32
    writer = ShardWriter(Path(out), dtype, shard_tokens)

    if workers and workers > 1:
        import multiprocessing as mp

        ctx = mp.get_context("spawn")
        with ctx.Pool(workers, initializer=_worker_init,
                      initargs=(tokenizer_path,)) as pool:
            for ids in pool.imap(_encode_block, iter_blocks(paths), chunksize=1):
                writer.write(ids)
    else:
        _worker_init(tokenizer_path)
        for block in iter_blocks(paths):
            writer.write(_encode_block(block))
    shards = writer.close()
    return {
        "tokens": writer.total,
        "vocab": vocab,
        "dtype": np.dtype(dtype).name,
        "shards": [str(s) for s in shards],
    }
