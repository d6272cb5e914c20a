"""Guards the driver contract: bench.py single- and multi-rank invocations
must emit exactly one well-formed JSON line from rank 0."""
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _check_payload(out: str, n_gpus: int):
    lines = [l for l in out.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out
    d = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == n_gpus
    assert d["scaling"] == "weak" and d["dtype"] == "bf16"
    assert d["config"]["model"] and d["config"]["seq_len"]
    assert d["value"] > 0
    return d


def test_bench_single_rank_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "llama_test", "--seq-len", "64", "--micro-batch", "2",
         "--h", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    _check_payload(out.stdout, 1)


def test_bench_two_ranks_cpu():
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29411", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--model", "llama_test", "--seq-len", "64",
         "--micro-batch", "2", "--h", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = _check_payload(out.stdout, 2)
    assert "diloco2" in d["config"]["parallelism"]
