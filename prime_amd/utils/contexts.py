"""CLI configuration contexts, mirroring the reference's Config system
(reference: prime_cli core/config.py — persistent JSON config + named
environment context files with env-var precedence and path-traversal-safe
context names).

Layout:  ~/.prime_amd/config.json            (current context's settings)
         ~/.prime_amd/environments/<name>.json  (saved contexts)
Precedence: PRIME_AMD_* env vars > current config file > defaults.
PRIME_AMD_CONTEXT selects a named context without switching the file.
"""
from __future__ import annotations

import json
import os
import re
from pathlib import Path

_NAME_RE = re.compile(r"^[A-Za-z0-9._-]{1,64}$")

DEFAULTS = {
    "runs_dir": "~/.prime_amd/runs",
    "default_model": "llama_150m",
    "global_addr": "127.0.0.1",
    "global_port": 29777,
}

ENV_MAP = {
    "runs_dir": "PRIME_AMD_RUNS_DIR",
    "default_model": "PRIME_AMD_DEFAULT_MODEL",
    "global_addr": "PRIME_GLOBAL_ADDR",
    "global_port": "PRIME_GLOBAL_PORT",
}


def config_root() -> Path:
    return Path(os.environ.get("PRIME_AMD_HOME", "~/.prime_amd")).expanduser()


def _env_dir() -> Path:
    return config_root() / "environments"


def _check_name(name: str) -> str:
    if not _NAME_RE.match(name):
        raise ValueError(
            f"invalid context name '{name}' (alphanumeric, dot, dash, "
            "underscore; no path separators)"
        )
    return name


class Contexts:
    def __init__(self):
        self.root = config_root()
        self.cfg_path = self.root / "config.json"

    # ------------------------------------------------------------- values
    def current(self) -> dict:
        """Effective settings: defaults < context file < env vars."""
        out = dict(DEFAULTS)
        ctx = os.environ.get("PRIME_AMD_CONTEXT")
        path = (
            _env_dir() / f"{_check_name(ctx)}.json" if ctx else self.cfg_path
        )
        if path.exists():
            try:
                out.update(json.loads(path.read_text()))
            except json.JSONDecodeError:
                pass
        for key, env in ENV_MAP.items():
            if env in os.environ:
                out[key] = os.environ[env]
        return out

    def set(self, key: str, value) -> None:
        self.root.mkdir(parents=True, exist_ok=True)
        cfg = {}
        if self.cfg_path.exists():
            cfg = json.loads(self.cfg_path.read_text())
        cfg[key] = value
        self.cfg_path.write_text(json.dumps(cfg, indent=2))

    # ----------------------------------------------------------- contexts
    def save(self, name: str) -> Path:
        """Snapshot the current config file as a named context."""
        _check_name(name)
        _env_dir().mkdir(parents=True, exist_ok=True)
        data = "{}"
        if self.cfg_path.exists():
            data = self.cfg_path.read_text()
        p = _env_dir() / f"{name}.json"
        p.write_text(data)
        return p

    def use(self, name: str) -> None:
        """Switch the current config to a saved context."""
        _check_name(name)
        p = _env_dir() / f"{name}.json"
        if not p.exists():
            raise FileNotFoundError(f"no saved context '{name}'")
        self.root.mkdir(parents=True, exist_ok=True)
        self.cfg_path.write_text(p.read_text())

    def delete(self, name: str) -> None:
        _check_name(name)
        p = _env_dir() / f"{name}.json"
        if p.exists():
            p.unlink()

    def list(self) -> list[str]:
        if not _env_dir().exists():
            return []
        return sorted(p.stem for p in _env_dir().glob("*.json"))
