"""CLI output helpers: JSON/table dual output with format validation and
a plain (no-color/no-style) mode.

Mirrors the reference CLI's output discipline (prime_cli
utils/display.py:13-57 — validate_output_format + json emission — and
utils/plain.py:25-245 — the plain-mode console) without pulling in rich:
every verb that lists or summarizes takes --json, and PRIME_AMD_PLAIN=1
(or --plain) strips ANSI styling for pipes/CI.
"""
from __future__ import annotations

import json
import os
from typing import Any, Iterable

import typer

VALID_FORMATS = ("table", "json")


def plain_mode() -> bool:
    return os.environ.get("PRIME_AMD_PLAIN", "0") == "1"


def set_plain(on: bool) -> None:
    os.environ["PRIME_AMD_PLAIN"] = "1" if on else "0"


def secho(message: str = "", **style: Any) -> None:
    """typer.secho that honors plain mode (drops fg/bold/etc.)."""
    if plain_mode():
        typer.echo(message)
    else:
        typer.secho(message, **style)


def validate_format(fmt: str) -> str:
    if fmt not in VALID_FORMATS:
        secho(f"invalid output format '{fmt}' (choose from: "
              f"{', '.join(VALID_FORMATS)})", fg="red")
        raise typer.Exit(2)
    return fmt


def emit_json(data: Any) -> None:
    typer.echo(json.dumps(data, indent=2, default=str))


def table(rows: Iterable[dict], columns: list[tuple[str, str, int]]) -> None:
    """Minimal fixed-width table: columns = [(key, header, width)]."""
    rows = list(rows)
    hdr = " ".join(f"{h:>{w}}" if w > 0 else f"{h:<{-w}}" for _, h, w in columns)
    typer.echo(hdr)
    for r in rows:
        cells = []
        for k, _, w in columns:
            v = r.get(k, "-")
            if isinstance(v, float):
                v = f"{v:,.4g}"
            cells.append(f"{v:>{w}}" if w > 0 else f"{v:<{-w}}")
        typer.echo(" ".join(cells))
