"""Structured JSON log lines, mirroring the reference's log format
({timestamp, level, message, type} — prime_cli commands/rl.py:153-196) with
client-side colorized rendering in the CLI."""
from __future__ import annotations

import json
import os
import sys
import time
from pathlib import Path
from typing import IO


class JsonLogger:
    def __init__(self, name: str = "prime_amd", file: IO | None = None,
                 log_path: str | Path | None = None, rank: int | None = None):
        self.name = name
        self.stream = file or sys.stderr
        self.rank = rank if rank is not None else int(os.environ.get("RANK", 0))
        self._fh = open(log_path, "a") if log_path else None

    def _emit(self, level: str, message: str, type_: str = "log", **extra) -> None:
        rec = {
            "timestamp": time.time(),
            "level": level,
            "message": message,
            "type": type_,
            "rank": self.rank,
        }
        if extra:
            rec.update(extra)
        line = json.dumps(rec)
        if self.rank == 0 or level in ("error", "warning"):
            print(line, file=self.stream, flush=True)
        if self._fh:
            self._fh.write(line + "\n")
            self._fh.flush()

    def info(self, msg: str, **kw) -> None:
        self._emit("info", msg, **kw)

    def warning(self, msg: str, **kw) -> None:
        self._emit("warning", msg, **kw)

    def error(self, msg: str, **kw) -> None:
        self._emit("error", msg, **kw)

    def progress(self, msg: str, **kw) -> None:
        self._emit("info", msg, type_="progress", **kw)

    def close(self) -> None:
        if self._fh:
            self._fh.close()
            self._fh = None


def render_log_line(line: str) -> str | None:
    """Pretty-print one JSON log line (CLI `logs` view); None = not JSON."""
    try:
        rec = json.loads(line)
    except json.JSONDecodeError:
        return None
    ts = time.strftime("%H:%M:%S", time.localtime(rec.get("timestamp", 0)))
    level = rec.get("level", "info").upper()
    return f"[{ts}] {level:7s} {rec.get('message', '')}"
