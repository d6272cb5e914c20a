"""Env-file / secret collection for run launches.

Mirrors the reference CLI's `-e/--env-file` handling (prime_cli
utils/env_vars.py:18-145): KEY=VALUE lines, `#` comments, optional
`export ` prefixes, quoted values, and `${VAR}` / `$VAR` expansion
against the parent environment — so W&B keys and similar secrets reach
the training processes without living in the TOML config.
"""
from __future__ import annotations

import os
import re
from pathlib import Path

_VAR_RE = re.compile(r"\$\{([A-Za-z_][A-Za-z0-9_]*)\}|\$([A-Za-z_][A-Za-z0-9_]*)")


class EnvFileError(ValueError):
    pass


def expand_vars(value: str, env: dict | None = None) -> str:
    """Expand ${VAR} and $VAR against `env` (default os.environ); unknown
    variables expand to the empty string, like a POSIX shell."""
    src = os.environ if env is None else env

    def sub(m: re.Match) -> str:
        name = m.group(1) or m.group(2)
        return str(src.get(name, ""))

    return _VAR_RE.sub(sub, value)


def parse_env_file(path: str | Path, env: dict | None = None) -> dict[str, str]:
    """Parse a dotenv-style file into a dict with expansion. Later lines
    may reference earlier ones (and the parent environment)."""
    out: dict[str, str] = {}
    base = dict(os.environ if env is None else env)
    for lineno, raw in enumerate(Path(path).read_text().splitlines(), 1):
        line = raw.strip()
        if not line or line.startswith("#"):
            continue
        if line.startswith("export "):
            line = line[len("export "):].lstrip()
        if "=" not in line:
            raise EnvFileError(f"{path}:{lineno}: expected KEY=VALUE, got {raw!r}")
        key, _, val = line.partition("=")
        key = key.strip()
        if not re.fullmatch(r"[A-Za-z_][A-Za-z0-9_]*", key):
            raise EnvFileError(f"{path}:{lineno}: invalid variable name {key!r}")
        val = val.strip()
        if len(val) >= 2 and val[0] == val[-1] and val[0] in "\"'":
            quoted = val[0]
            val = val[1:-1]
            if quoted == "'":
                out[key] = val  # single quotes: no expansion
                base[key] = val
                continue
        out[key] = expand_vars(val, base)
        base[key] = out[key]
    return out


def collect_env(env_files: list[str], extra: list[str] | None = None) -> dict[str, str]:
    """Merge env files (in order) plus explicit KEY=VALUE overrides."""
    merged: dict[str, str] = {}
    for f in env_files:
        merged.update(parse_env_file(f))
    for item in extra or []:
        if "=" not in item:
            raise EnvFileError(f"--env expects KEY=VALUE, got {item!r}")
        k, _, v = item.partition("=")
        merged[k.strip()] = expand_vars(v)
    return merged
