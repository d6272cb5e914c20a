"""Run wrapper invoked by the CLI launcher (and torchrun): executes the
trainer and maintains the run directory's status/result files so the
lifecycle verbs (list/logs/stop/restart) can inspect runs."""
from __future__ import annotations

import json
import os
import sys
import time
from pathlib import Path


def main() -> int:
    cfg_path, run_dir = sys.argv[1], Path(sys.argv[2])
    rank = int(os.environ.get("RANK", 0))
    from prime_amd.train import train_from_config
    from prime_amd.utils.config import load_config

    cfg = load_config(cfg_path)
    if rank == 0:
        (run_dir / "status.json").write_text(json.dumps(
            {"status": "RUNNING", "pid": os.getpid(), "started": time.time(),
             "steps_total": cfg.steps}))
    try:
        result = train_from_config(cfg, run_dir)
    except BaseException as e:  # noqa: BLE001
        if rank == 0:
            (run_dir / "status.json").write_text(json.dumps(
                {"status": "FAILED", "error": repr(e), "ended": time.time()}))
        raise
    if rank == 0:
        (run_dir / "status.json").write_text(json.dumps(
            {"status": "COMPLETED", "ended": time.time(), "result": result}))
    return 0


if __name__ == "__main__":
    sys.exit(main())
