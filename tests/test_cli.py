"""CLI tests with typer CliRunner invoking the real app (reference test
strategy: CLI-level golden output tests, no network/GPU)."""
import json
import time

import pytest
from typer.testing import CliRunner

from prime_amd.cli.main import app

runner = CliRunner()


@pytest.fixture()
def runs_dir(tmp_path, monkeypatch):
    monkeypatch.setenv("PRIME_AMD_RUNS_DIR", str(tmp_path / "runs"))
    return tmp_path / "runs"


def test_help():
    r = runner.invoke(app, ["--help"])
    assert r.exit_code == 0
    assert "train" in r.output


def test_train_init_and_validate(tmp_path, runs_dir):
    cfg = tmp_path / "t.toml"
    r = runner.invoke(app, ["train", "init", str(cfg), "--model", "llama_test"])
    assert r.exit_code == 0 and cfg.exists()
    # template must be loadable
    from prime_amd.utils.config import load_config

    assert load_config(cfg).model.name == "llama_test"
    # refuses overwrite
    r2 = runner.invoke(app, ["train", "init", str(cfg)])
    assert r2.exit_code == 1


def test_train_rejects_bad_config(tmp_path, runs_dir):
    bad = tmp_path / "bad.toml"
    bad.write_text("[model]\nnam = 'x'\n")
    r = runner.invoke(app, ["train", "run", str(bad)])
    assert r.exit_code == 2
    assert "model.nam" in r.output


def test_models_listing():
    r = runner.invoke(app, ["train", "models"])
    assert r.exit_code == 0
    assert "intellect_10b" in r.output
    assert "llama_70b" in r.output


def test_full_run_lifecycle(tmp_path, runs_dir):
    cfg = tmp_path / "run.toml"
    cfg.write_text(
        'run_name = "cli_e2e"\nsteps = 3\n'
        '[model]\nname = "llama_test"\nseq_len = 64\n'
        '[data]\nmicro_batch_size = 2\n'
        '[diloco]\nH = 2\n[metrics]\nlog_interval = 1\n'
    )
    r = runner.invoke(app, ["train", "run", str(cfg)])
    assert r.exit_code == 0, r.output
    assert "COMPLETED" in r.output

    r = runner.invoke(app, ["train", "list"])
    assert "cli_e2e" in r.output and "COMPLETED" in r.output

    r = runner.invoke(app, ["train", "list", "--json"])
    rows = json.loads(r.output)
    run_id = rows[0]["run"]

    r = runner.invoke(app, ["train", "metrics", run_id])
    assert r.exit_code == 0 and "LOSS" in r.output

    r = runner.invoke(app, ["train", "logs", run_id])
    assert r.exit_code == 0 and "starting run" in r.output

    r = runner.invoke(app, ["train", "checkpoints", run_id])
    assert r.exit_code == 0


def test_config_view_set(tmp_path, monkeypatch, runs_dir):
    monkeypatch.setenv("PRIME_AMD_HOME", str(tmp_path / "home2"))
    r = runner.invoke(app, ["config", "set", "default_model", "llama_1b"])
    assert r.exit_code == 0
    r = runner.invoke(app, ["config", "view"])
    assert "llama_1b" in r.output


def test_config_contexts(tmp_path, monkeypatch, runs_dir):
    monkeypatch.setenv("PRIME_AMD_HOME", str(tmp_path / "home"))
    r = runner.invoke(app, ["config", "set", "default_model", "llama_8b"])
    assert r.exit_code == 0
    r = runner.invoke(app, ["config", "save", "prod"])
    assert r.exit_code == 0
    runner.invoke(app, ["config", "set", "default_model", "llama_test"])
    r = runner.invoke(app, ["config", "view"])
    assert "llama_test" in r.output
    r = runner.invoke(app, ["config", "use", "prod"])
    assert r.exit_code == 0
    r = runner.invoke(app, ["config", "view"])
    assert "llama_8b" in r.output
    r = runner.invoke(app, ["config", "envs"])
    assert "prod" in r.output
    # env-var precedence
    monkeypatch.setenv("PRIME_AMD_DEFAULT_MODEL", "llama_70b")
    r = runner.invoke(app, ["config", "view"])
    assert "llama_70b" in r.output
    # path traversal rejected
    r = runner.invoke(app, ["config", "save", "../evil"])
    assert r.exit_code == 1


def test_eval_command(runs_dir):
    r = runner.invoke(app, ["eval", "--model", "llama_test", "--seq-len", "64",
                            "--batches", "2", "--micro-batch", "2", "--json"])
    assert r.exit_code == 0, r.output
    res = json.loads(r.output)
    assert res["tokens"] == 2 * 2 * 64
    assert res["perplexity"] > 1


def test_failure_classification():
    from prime_amd.utils.failures import NonFiniteLossError, classify_failure

    assert classify_failure(NonFiniteLossError("loss=nan"))["category"] == "NON_FINITE_LOSS"
    assert classify_failure(RuntimeError("HIP out of memory"))["category"] == "OOM"
    assert classify_failure(RuntimeError("NCCL watchdog timeout"))["category"] == "COMM"
    assert "hint" in classify_failure(ValueError("x"))


def test_graceful_sigterm_stop(tmp_path, runs_dir):
    import signal
    import subprocess
    import sys
    import time as _t

    cfg = tmp_path / "g.toml"
    cfg.write_text(
        'run_name = "graceful"\nsteps = 100000\n'
        '[model]\nname = "llama_test"\nseq_len = 64\n'
        '[data]\nmicro_batch_size = 2\n[diloco]\nH = 5\n'
        f'[checkpoint]\ninterval = 1\npath = "{tmp_path}/ck"\nasync_save = false\n'
        '[metrics]\nlog_interval = 1000000\n'
    )
    run_dir = tmp_path / "rd"
    run_dir.mkdir()
    proc = subprocess.Popen(
        [sys.executable, "-m", "prime_amd.cli.runner", str(cfg), str(run_dir)],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
    )
    # wait for some steps, then SIGTERM
    deadline = _t.time() + 90
    while _t.time() < deadline and not (run_dir / "metrics.jsonl").exists():
        if (run_dir / "status.json").exists():
            st = json.loads((run_dir / "status.json").read_text())
            if st.get("status") == "RUNNING":
                break
        _t.sleep(0.3)
    _t.sleep(3)  # let a few steps run
    proc.send_signal(signal.SIGTERM)
    rc = proc.wait(timeout=60)
    assert rc == 0
    st = json.loads((run_dir / "status.json").read_text())
    assert st["status"] == "STOPPED", st
    assert (tmp_path / "ck").exists()


def test_delete_run(tmp_path, runs_dir):
    cfg = tmp_path / "d.toml"
    cfg.write_text('run_name = "del_me"\nsteps = 1\n'
                   '[model]\nname = "llama_test"\nseq_len = 64\n'
                   '[data]\nmicro_batch_size = 1\n[metrics]\nlog_interval = 100\n')
    r = runner.invoke(app, ["train", "run", str(cfg)])
    assert r.exit_code == 0
    rows = json.loads(runner.invoke(app, ["train", "list", "--json"]).output)
    run_id = rows[0]["run"]
    r = runner.invoke(app, ["train", "delete", run_id])
    assert r.exit_code == 0
    assert json.loads(runner.invoke(app, ["train", "list", "--json"]).output) == []


def test_report_command(tmp_path, runs_dir):
    cfg = tmp_path / "r.toml"
    cfg.write_text('run_name = "rep"\nsteps = 2\n'
                   '[model]\nname = "llama_test"\nseq_len = 64\n'
                   '[data]\nmicro_batch_size = 2\n[metrics]\nlog_interval = 1\n')
    r = runner.invoke(app, ["train", "run", str(cfg)])
    assert r.exit_code == 0
    run_id = json.loads(runner.invoke(app, ["train", "list", "--json"]).output)[0]["run"]
    r = runner.invoke(app, ["train", "report", run_id])
    assert r.exit_code == 0
    assert "loss" in r.output and "tokens/sec" in r.output


def test_env_file_parsing(tmp_path):
    from prime_amd.utils.env_vars import EnvFileError, collect_env, parse_env_file

    f = tmp_path / ".env"
    f.write_text(
        "# comment\n"
        "export WANDB_API_KEY=abc123\n"
        "BASE=/data\n"
        'OUT="${BASE}/runs"\n'
        "LITERAL='${BASE}/raw'\n"
        "FROM_PARENT=${PRIME_TEST_PARENT}\n"
    )
    import os

    os.environ["PRIME_TEST_PARENT"] = "hello"
    try:
        env = parse_env_file(f)
    finally:
        del os.environ["PRIME_TEST_PARENT"]
    assert env["WANDB_API_KEY"] == "abc123"
    assert env["OUT"] == "/data/runs"
    assert env["LITERAL"] == "${BASE}/raw"  # single quotes: no expansion
    assert env["FROM_PARENT"] == "hello"
    merged = collect_env([str(f)], ["EXTRA=1"])
    assert merged["EXTRA"] == "1"
    bad = tmp_path / "bad.env"
    bad.write_text("NOT A LINE\n")
    import pytest

    with pytest.raises(EnvFileError):
        parse_env_file(bad)


def test_models_json_and_plain(tmp_path, monkeypatch):
    import json as _json

    from typer.testing import CliRunner

    from prime_amd.cli.main import app

    monkeypatch.setenv("PRIME_AMD_RUNS_DIR", str(tmp_path))
    r = CliRunner().invoke(app, ["train", "models", "--json"])
    assert r.exit_code == 0, r.output
    d = _json.loads(r.output)
    assert "intellect_10b" in d and d["intellect_10b"]["dim"] == 4096
    r = CliRunner().invoke(app, ["--plain", "train", "models"])
    assert r.exit_code == 0
    assert "\x1b[" not in r.output  # no ANSI styling in plain mode


def test_report_and_checkpoints_json(tmp_path, monkeypatch):
    import json as _json

    from typer.testing import CliRunner

    from prime_amd.cli.main import app

    monkeypatch.setenv("PRIME_AMD_RUNS_DIR", str(tmp_path))
    d = tmp_path / "myrun-abc123"
    d.mkdir(parents=True)
    (d / "status.json").write_text('{"status": "COMPLETED"}')
    (d / "metrics.jsonl").write_text(
        '{"step": 1, "loss": 2.0, "tokens_per_sec": 100.0}\n'
        '{"step": 2, "loss": 1.5, "tokens_per_sec": 110.0}\n'
    )
    r = CliRunner().invoke(app, ["train", "report", "myrun", "--json"])
    assert r.exit_code == 0, r.output
    rep = _json.loads(r.output)
    assert rep["status"] == "COMPLETED" and rep["loss"]["last"] == 1.5
    r = CliRunner().invoke(app, ["train", "checkpoints", "myrun", "--json"])
    assert r.exit_code == 0
    assert _json.loads(r.output) == []


def test_usage_and_top(tmp_path, monkeypatch):
    import json as _json

    from typer.testing import CliRunner

    from prime_amd.cli.main import app

    monkeypatch.setenv("PRIME_AMD_RUNS_DIR", str(tmp_path))
    d = tmp_path / "urun-xyz"
    d.mkdir(parents=True)
    (d / "status.json").write_text('{"status": "COMPLETED"}')
    (d / "metrics.jsonl").write_text(
        '{"step": 10, "loss": 2.0, "tokens_per_sec": 1000.0, '
        '"ms_per_step": 100.0, "mfu": 0.3}\n'
    )
    r = CliRunner().invoke(app, ["train", "usage", "urun", "--json"])
    assert r.exit_code == 0, r.output
    u = _json.loads(r.output)
    assert u["steps"] == 10 and u["gpu_seconds"] > 0
    r = CliRunner().invoke(app, ["--plain", "top", "--once"])
    assert r.exit_code == 0, r.output
    assert "urun-xyz" in r.output


def test_doctor(tmp_path, monkeypatch):
    import json as _json

    from typer.testing import CliRunner

    from prime_amd.cli.main import app

    r = CliRunner().invoke(app, ["doctor", "--json"])
    assert r.exit_code == 0, r.output
    checks = {c["check"]: c for c in _json.loads(r.output)}
    assert checks["hipcc"]["ok"]          # build image ships hipcc
    assert checks["torch"]["ok"]
    assert checks["kernel .so"]["ok"]     # fresh source-hash stamp
    assert not checks["gpu"]["ok"]        # no GPU in the build container


def test_list_output_format_validation(tmp_path, monkeypatch):
    from typer.testing import CliRunner

    from prime_amd.cli.main import app

    monkeypatch.setenv("PRIME_AMD_RUNS_DIR", str(tmp_path))
    r = CliRunner().invoke(app, ["train", "list", "--output", "yaml"])
    assert r.exit_code == 2
    assert "invalid output format" in r.output
    r = CliRunner().invoke(app, ["train", "list", "--output", "json"])
    assert r.exit_code == 0


def test_restart_completes_to_total(tmp_path, runs_dir):
    """restart resumes from the latest checkpoint and finishes the run's
    TOTAL step target (not cfg.steps more); a run_name containing the
    word 'resume' must not defeat the resume injection."""
    cfg = tmp_path / "run.toml"
    cfg.write_text(
        'run_name = "resume_edge"\nsteps = 4\n'
        '[model]\nname = "llama_test"\nseq_len = 64\n'
        '[data]\nmicro_batch_size = 2\n'
        '[diloco]\nH = 2\n'
        '[checkpoint]\ninterval = 1\nasync_save = false\n'
        '[metrics]\nlog_interval = 100\n'
    )
    r = runner.invoke(app, ["train", "run", str(cfg)])
    assert r.exit_code == 0, r.output
    rows = json.loads(runner.invoke(app, ["train", "list", "--json"]).output)
    run_id = rows[0]["run"]
    r = runner.invoke(app, ["train", "restart", run_id])
    assert r.exit_code == 0, r.output
    # restart of a completed run trains 0 further steps and re-completes
    rows = json.loads(runner.invoke(app, ["train", "list", "--json"]).output)
    latest = [x for x in rows if x["run"].startswith("resume_edge")]
    assert any(x["status"] == "COMPLETED" for x in latest)
    # the rewritten config must carry resume = "latest"
    import re as _re

    restarted = sorted(runs_dir.glob("*/config.restart.toml"))
    assert restarted and _re.search(r"(?m)^resume = \"latest\"",
                                    restarted[-1].read_text())
