"""Optional tokenizer support for the serving CLI: loads a LOCAL
tokenizer.json (HuggingFace `tokenizers` format — the package ships in the
image; no network fetches)."""
from __future__ import annotations

from pathlib import Path


def load_tokenizer(path: str):
    p = Path(path)
    if not p.exists():
        raise FileNotFoundError(f"tokenizer file not found: {p}")
    try:
        from tokenizers import Tokenizer
    except ImportError as e:  # pragma: no cover
        raise RuntimeError("the `tokenizers` package is not available") from e
    return Tokenizer.from_file(str(p))


def encode(tok, text: str) -> list[int]:
    return tok.encode(text).ids


def decode(tok, ids: list[int]) -> str:
    return tok.decode(ids)
