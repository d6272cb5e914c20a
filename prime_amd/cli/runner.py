"""Run wrapper invoked by the CLI launcher (and torchrun): executes the
trainer and maintains the run directory's status/result files so the
lifecycle verbs (list/logs/stop/restart) can inspect runs."""
from __future__ import annotations

import json
import os
import signal
import sys
import time
from pathlib import Path


def main() -> int:
    cfg_path, run_dir = sys.argv[1], Path(sys.argv[2])
    rank = int(os.environ.get("RANK", 0))
    from prime_amd.train import Trainer
    from prime_amd.utils.config import load_config
    from prime_amd.utils.failures import classify_failure

    cfg = load_config(cfg_path)
    if rank == 0:
        (run_dir / "status.json").write_text(json.dumps(
            {"status": "RUNNING", "pid": os.getpid(),
             "pgid": os.getpgid(0), "started": time.time(),
             "steps_total": cfg.steps}))
    trainer = Trainer(cfg, run_dir)

    def _on_term(signum, frame):  # graceful: checkpoint at next step boundary
        trainer.stop_requested = True

    signal.signal(signal.SIGTERM, _on_term)
    try:
        result = trainer.run()
    except BaseException as e:  # noqa: BLE001
        if rank == 0:
            (run_dir / "status.json").write_text(json.dumps(
                {"status": "FAILED", "ended": time.time(),
                 "failure_analysis": classify_failure(e)}))
        raise
    finally:
        trainer.close()
    if rank == 0:
        status = "STOPPED" if trainer.stop_requested else "COMPLETED"
        (run_dir / "status.json").write_text(json.dumps(
            {"status": status, "ended": time.time(), "result": result}))
    return 0


if __name__ == "__main__":
    sys.exit(main())
