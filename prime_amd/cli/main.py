"""prime-amd CLI — run lifecycle UX modeled on the reference's `prime
train` command family (reference: prime_cli commands/rl.py — TOML config
as default argument, strict schema with friendly errors, logs/metrics/
checkpoints/list/stop/restart verbs, JSON/table dual output).

Unlike the reference (a thin HTTPS client for a hosted service), runs here
execute locally: the launcher spawns `torch.distributed.run` with one rank
per GPU and tracks them in a run registry directory.
"""
from __future__ import annotations

import json
import os
import signal
import socket
import subprocess
import sys
import time
import uuid
from pathlib import Path
from typing import Optional

import typer
from typer.core import TyperGroup

from ..utils.config import ConfigError, default_config_toml, load_config
from ..utils.display import emit_json, secho, set_plain, table
from ..utils.logging import render_log_line


class DefaultRunGroup(TyperGroup):
    """Makes `prime-amd train cfg.toml` work without the `run` verb
    (reference: DefaultGroup in prime_cli commands/rl.py:1197-1244)."""

    def resolve_command(self, ctx, args):
        if args and args[0] not in self.commands and not args[0].startswith("-"):
            args = ["run", *args]
        return super().resolve_command(ctx, args)


app = typer.Typer(help="MI355X-native DiLoCo training engine", no_args_is_help=True)


@app.callback()
def _root(plain: bool = typer.Option(
        False, "--plain", help="no colors/styling (or PRIME_AMD_PLAIN=1)")):
    if plain:
        set_plain(True)

train_app = typer.Typer(help="Launch and manage training runs",
                        no_args_is_help=True, cls=DefaultRunGroup)
config_app = typer.Typer(help="CLI configuration", no_args_is_help=True)
app.add_typer(train_app, name="train")
app.add_typer(config_app, name="config")


def runs_root() -> Path:
    from ..utils.contexts import Contexts

    root = Path(str(Contexts().current()["runs_dir"])).expanduser()
    root.mkdir(parents=True, exist_ok=True)
    return root


def _find_run(ref: str) -> Path:
    root = runs_root()
    d = root / ref
    if d.exists():
        return d
    matches = [p for p in root.iterdir() if p.is_dir() and p.name.startswith(ref)]
    if len(matches) == 1:
        return matches[0]
    if not matches:
        secho(f"no run matching '{ref}'", fg="red")
        raise typer.Exit(1)
    secho(f"ambiguous run '{ref}': {[m.name for m in matches]}", fg="red")
    raise typer.Exit(1)


def _status(run_dir: Path) -> dict:
    f = run_dir / "status.json"
    if not f.exists():
        return {"status": "UNKNOWN"}
    st = json.loads(f.read_text())
    if st.get("status") == "RUNNING":
        pid = st.get("pid")
        if pid and not _pid_alive(pid):
            st["status"] = "DIED"
    return st


def _pid_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except OSError:
        return False


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


# ------------------------------------------------------------------ train
@train_app.command("run")
def run(
    config: str = typer.Argument(..., help="TOML run config"),
    nproc: int = typer.Option(0),
    detach: bool = typer.Option(False, "--detach", "-d"),
    env_file: list[str] = typer.Option([], "--env-file", "-e",
                                       help="dotenv file(s) with ${VAR} "
                                            "expansion (e.g. WANDB_API_KEY)"),
    env: list[str] = typer.Option([], "--env",
                                  help="extra KEY=VALUE for the run env"),
):
    from ..utils.env_vars import EnvFileError, collect_env

    try:
        cfg = load_config(config)
    except ConfigError as e:
        secho(str(e), fg="red")
        raise typer.Exit(2)
    try:
        run_env = collect_env(env_file, env)
    except (EnvFileError, OSError) as e:
        secho(str(e), fg="red")
        raise typer.Exit(2)
    run_id = f"{cfg.run_name}-{uuid.uuid4().hex[:6]}"
    run_dir = runs_root() / run_id
    run_dir.mkdir(parents=True)
    (run_dir / "config.toml").write_text(Path(config).read_text())

    if nproc <= 0:
        nproc = cfg.parallel.worker_size
        try:
            import torch

            if torch.cuda.is_available():
                nproc = max(nproc, 1)
        except ImportError:
            pass
    cmd = [sys.executable, "-m", "prime_amd.cli.runner", str(run_dir / "config.toml"), str(run_dir)]
    if nproc > 1:
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={nproc}",
            "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
            "-m", "prime_amd.cli.runner", str(run_dir / "config.toml"), str(run_dir),
        ]
    secho(f"run {run_id}: {' '.join(cmd)}", fg="cyan")
    log = open(run_dir / "launcher.log", "w")
    child_env = {**os.environ, **run_env}
    proc = subprocess.Popen(cmd, stdout=log, stderr=subprocess.STDOUT,
                            start_new_session=True, env=child_env)
    (run_dir / "launcher.pid").write_text(str(proc.pid))
    (run_dir / "status.json").write_text(json.dumps(
        {"status": "STARTING", "pid": proc.pid, "started": time.time()}))
    if detach:
        typer.echo(f"started (pid {proc.pid}); follow with: prime-amd train logs {run_id} -f")
        return
    try:
        rc = proc.wait()
    except KeyboardInterrupt:
        secho("interrupt: stopping run", fg="yellow")
        os.killpg(proc.pid, signal.SIGTERM)
        rc = proc.wait()
    st = _status(run_dir)
    color = "green" if st.get("status") == "COMPLETED" else "red"
    secho(f"run {run_id}: {st.get('status')} (rc={rc})", fg=color)
    if st.get("status") == "COMPLETED":
        typer.echo(json.dumps(st.get("result", {}), indent=2))


@train_app.command("init")
def init(
    path: str = typer.Argument("train.toml"),
    model: str = typer.Option("llama_150m", help="model preset"),
):
    """Write a config template (reference: config template generator)."""
    p = Path(path)
    if p.exists():
        secho(f"{p} exists; not overwriting", fg="red")
        raise typer.Exit(1)
    p.write_text(default_config_toml(model))
    secho(f"wrote {p}", fg="green")


@train_app.command("list")
def list_runs(json_out: bool = typer.Option(False, "--json"),
              output: str = typer.Option(
                  "table", "--output", "-o",
                  help="table | json (reference: display.py format "
                       "validation)")):
    from ..utils.display import validate_format

    if validate_format(output) == "json":
        json_out = True
    rows = []
    for d in sorted(runs_root().iterdir()):
        if not d.is_dir():
            continue
        st = _status(d)
        last = _last_metrics(d)
        status = st.get("status", "?")
        fa = st.get("failure_analysis")
        if fa:
            status = f"{status}:{fa['category']}"
        rows.append({
            "run": d.name, "status": status,
            "step": last.get("step", "-"), "loss": last.get("loss", "-"),
            "tok/s": last.get("tokens_per_sec", "-"),
        })
    if json_out:
        typer.echo(json.dumps(rows, indent=2))
        return
    if not rows:
        typer.echo("no runs")
        return
    table(rows, [("run", "RUN", -40), ("status", "STATUS", -22),
                 ("step", "STEP", 8), ("loss", "LOSS", 10),
                 ("tok/s", "TOK/S", 12)])


def _last_metrics(run_dir: Path) -> dict:
    f = run_dir / "metrics.jsonl"
    if not f.exists():
        return {}
    lines = f.read_text().splitlines()
    return json.loads(lines[-1]) if lines else {}


@train_app.command("logs")
def logs(
    run: str = typer.Argument(...),
    follow: bool = typer.Option(False, "--follow", "-f"),
    rank: int = typer.Option(0),
    raw: bool = typer.Option(False, help="raw JSON lines"),
    component: Optional[str] = typer.Option(
        None, help="only lines of this type (progress/checkpoint/result/...)"),
):
    d = _find_run(run)
    f = d / f"rank{rank}.log"
    if not f.exists():
        f = d / "launcher.log"
    if not f.exists():
        secho("no logs yet", fg="yellow")
        raise typer.Exit(1)
    with open(f) as fh:
        while True:
            line = fh.readline()
            if line:
                if component:
                    try:
                        if json.loads(line).get("type") != component:
                            continue
                    except (ValueError, AttributeError):
                        continue
                if raw:
                    typer.echo(line.rstrip())
                else:
                    rendered = render_log_line(line)
                    typer.echo(rendered if rendered else line.rstrip())
            elif follow and _status(d).get("status") in ("RUNNING", "STARTING"):
                time.sleep(0.5)
            else:
                break


@train_app.command("metrics")
def metrics(run: str = typer.Argument(...), last: int = typer.Option(10),
            json_out: bool = typer.Option(False, "--json")):
    d = _find_run(run)
    from ..utils.metrics import read_metrics

    rows = read_metrics(d / "metrics.jsonl")[-last:]
    if json_out:
        typer.echo(json.dumps(rows, indent=2))
        return
    if not rows:
        typer.echo("no metrics yet")
        return
    hdr = f"{'STEP':>8} {'LOSS':>10} {'TOK/S':>12} {'MFU':>7} {'MS/STEP':>9} {'OUTER':>6}"
    typer.echo(hdr)
    for r in rows:
        typer.echo(f"{r.get('step', 0):>8} {r.get('loss', float('nan')):>10.4f} "
                   f"{r.get('tokens_per_sec', 0):>12,.0f} {r.get('mfu', 0):>7.3f} "
                   f"{r.get('ms_per_step', 0):>9.1f} {r.get('outer_steps', 0):>6}")


@train_app.command("checkpoints")
def checkpoints(run: str = typer.Argument(...),
                json_out: bool = typer.Option(False, "--json")):
    d = _find_run(run)
    ck = d / "ckpt"
    rows = []
    if ck.exists():
        for tag in sorted(ck.iterdir()):
            if tag.is_dir() and not tag.is_symlink():
                files = list(tag.glob("*.pt"))
                rows.append({"tag": tag.name, "files": len(files),
                             "bytes": sum(f.stat().st_size for f in files)})
    if json_out:
        emit_json(rows)
        return
    if not rows:
        typer.echo("no checkpoints")
        return
    for r in rows:
        typer.echo(f"{r['tag']:16s} {r['files']} file(s) {r['bytes']/1e9:8.2f} GB")


@train_app.command("stop")
def stop(run: str = typer.Argument(...)):
    d = _find_run(run)
    st = _status(d)
    pid = st.get("pid")
    if not pid or not _pid_alive(pid):
        secho("not running", fg="yellow")
        return
    # exact recorded process group (never pattern-based): signalling the
    # pgid reaches EVERY rank, and the trainer's per-step consensus stop
    # makes them exit at the same boundary with a checkpoint.
    try:
        os.killpg(st.get("pgid") or pid, signal.SIGTERM)
    except OSError:
        os.kill(pid, signal.SIGTERM)
    secho(f"stop requested for {d.name} (graceful: checkpoints first)", fg="green")


@train_app.command("delete")
def delete(run: str = typer.Argument(...),
           force: bool = typer.Option(False, "--force", "-f")):
    """Delete a run's directory (refuses while RUNNING unless --force)."""
    import shutil

    d = _find_run(run)
    st = _status(d)
    if st.get("status") in ("RUNNING", "STARTING") and not force:
        secho("run appears to be RUNNING; stop it first or use --force", fg="red")
        raise typer.Exit(1)
    shutil.rmtree(d)
    secho(f"deleted {d.name}", fg="green")


@train_app.command("restart")
def restart(run: str = typer.Argument(...), detach: bool = typer.Option(False, "-d")):
    """Restart a run from its latest checkpoint."""
    d = _find_run(run)
    cfg_path = d / "config.toml"
    if not cfg_path.exists():
        secho("run has no config.toml", fg="red")
        raise typer.Exit(1)
    import re

    text = cfg_path.read_text()
    # line-anchored: a substring check would false-positive on e.g.
    # run_name = "resume_test" and silently restart from scratch
    has_resume = re.search(r"(?m)^\s*resume\s*=", text)
    if "[checkpoint]" in text and not has_resume:
        text = text.replace("[checkpoint]", '[checkpoint]\nresume = "latest"')
    elif "[checkpoint]" not in text:
        text += '\n[checkpoint]\nresume = "latest"\n'
    new_cfg = d / "config.restart.toml"
    new_cfg.write_text(text)
    run_cmd(str(new_cfg), detach)


def run_cmd(config: str, detach: bool) -> None:
    run(config, nproc=0, detach=detach, env_file=[], env=[])


@train_app.command("report")
def report(run: str = typer.Argument(...),
           json_out: bool = typer.Option(False, "--json")):
    """Summarize a run: status, loss trajectory, throughput, failures."""
    from ..utils.metrics import read_metrics

    d = _find_run(run)
    st = _status(d)
    rows = read_metrics(d / "metrics.jsonl")
    if json_out:
        losses = [r["loss"] for r in rows if "loss" in r]
        tps = [r["tokens_per_sec"] for r in rows if "tokens_per_sec" in r]
        emit_json({
            "run": d.name, "status": st.get("status"),
            "failure_analysis": st.get("failure_analysis"),
            "steps_logged": [rows[0]["step"], rows[-1]["step"]] if rows else None,
            "loss": {"first": losses[0], "min": min(losses),
                     "last": losses[-1]} if losses else None,
            "tokens_per_sec": {"mean": sum(tps) / len(tps),
                               "last": tps[-1]} if tps else None,
            "result": st.get("result"),
        })
        return
    secho(f"run {d.name}: {st.get('status', '?')}", bold=True)
    fa = st.get("failure_analysis")
    if fa:
        secho(f"  failure: {fa['category']} — {fa['hint']}", fg="red")
    if not rows:
        typer.echo("  no metrics recorded")
        return
    losses = [r["loss"] for r in rows if "loss" in r]
    tps = [r["tokens_per_sec"] for r in rows if "tokens_per_sec" in r]
    mfus = [r["mfu"] for r in rows if "mfu" in r]
    typer.echo(f"  steps logged : {rows[0]['step']} .. {rows[-1]['step']}")
    if losses:
        typer.echo(f"  loss         : first {losses[0]:.4f}  min {min(losses):.4f}  last {losses[-1]:.4f}")
    if tps:
        typer.echo(f"  tokens/sec   : mean {sum(tps)/len(tps):,.0f}  last {tps[-1]:,.0f}")
    if mfus:
        typer.echo(f"  MFU          : mean {sum(mfus)/len(mfus):.3f}  last {mfus[-1]:.3f}")
    res = st.get("result")
    if res:
        typer.echo(f"  result       : {json.dumps(res)}")


@train_app.command("models")
def models(json_out: bool = typer.Option(False, "--json")):
    """List model presets (reference: `prime train models`)."""
    from ..models import CONFIGS

    if json_out:
        emit_json({name: c.to_dict() for name, c in CONFIGS.items()})
        return
    for name, c in CONFIGS.items():
        typer.echo(f"{name:16s} {c.n_params()/1e9:7.2f}B params  dim={c.dim} "
                   f"layers={c.n_layers} heads={c.n_heads}/{c.n_kv_heads} "
                   f"vocab={c.vocab_size}")


@train_app.command("usage")
def usage(run: str = typer.Argument(...),
          json_out: bool = typer.Option(False, "--json")):
    """Resource usage for a run: tokens, wall time, GPU-seconds, FLOPs
    (engine-side counterpart of the reference's billing/usage views)."""
    from ..models import CONFIGS
    from ..utils.metrics import model_flops_per_token, read_metrics

    d = _find_run(run)
    st = _status(d)
    rows = read_metrics(d / "metrics.jsonl")
    cfg_txt = (d / "config.toml")
    n_gpus, model_name, seq_len = 1, None, 2048
    if cfg_txt.exists():
        from ..utils.config import load_config

        try:
            c = load_config(cfg_txt)
            n_gpus = c.parallel.worker_size
            model_name, seq_len = c.model.name, c.model.seq_len
        except Exception:  # noqa: BLE001 — usage stays best-effort
            pass
    if not rows:
        secho("no metrics recorded", fg="yellow")
        raise typer.Exit(0)
    steps = rows[-1].get("step", 0)
    tps = [r["tokens_per_sec"] for r in rows if "tokens_per_sec" in r]
    ms = [r["ms_per_step"] for r in rows if "ms_per_step" in r]
    mfus = [r["mfu"] for r in rows if "mfu" in r]
    wall_s = sum(msv for msv in ms) / 1000.0 * (steps / max(1, len(ms)))
    tokens = int(sum(tps) / max(1, len(tps)) * wall_s) if tps else 0
    flops = None
    if model_name in CONFIGS:
        flops = model_flops_per_token(CONFIGS[model_name], seq_len) * tokens
    data = {
        "run": d.name, "status": st.get("status"), "model": model_name,
        "steps": steps, "est_tokens": tokens,
        "est_wall_seconds": round(wall_s, 1),
        "gpu_seconds": round(wall_s * n_gpus, 1),
        "est_pflops": round(flops / 1e15, 1) if flops else None,
        "mean_mfu": round(sum(mfus) / len(mfus), 4) if mfus else None,
    }
    if json_out:
        emit_json(data)
        return
    for k, v in data.items():
        typer.echo(f"  {k:18s}: {v}")


@app.command("top")
def top(interval: float = typer.Option(2.0, help="refresh seconds"),
        once: bool = typer.Option(False, help="render once and exit")):
    """Live dashboard over all runs (an engine-side sliver of the
    reference's Lab TUI: statuses, loss, throughput, MFU, refreshed)."""
    from ..utils.display import plain_mode

    def rows_now():
        out = []
        for d in sorted(runs_root().iterdir()):
            if not d.is_dir():
                continue
            st = _status(d)
            last = _last_metrics(d)
            out.append({
                "run": d.name, "status": st.get("status", "?"),
                "step": last.get("step", "-"),
                "loss": last.get("loss"), "tok_s": last.get("tokens_per_sec"),
                "mfu": last.get("mfu"), "outer": last.get("outer_steps", "-"),
            })
        return out

    def fmt(v, spec):
        return format(v, spec) if isinstance(v, (int, float)) else str(v or "-")

    use_rich = not plain_mode() and not once
    if use_rich:
        try:
            from rich.live import Live
            from rich.table import Table
        except ImportError:
            use_rich = False
    if not use_rich:
        for r in rows_now():
            typer.echo(f"{r['run']:40s} {r['status']:12s} "
                       f"{fmt(r['step'], '>8')} {fmt(r['loss'], '.4f'):>10} "
                       f"{fmt(r['tok_s'], ',.0f'):>12} {fmt(r['mfu'], '.3f'):>7}")
        return

    def build():
        t = Table(title="prime-amd runs")
        for col in ("run", "status", "step", "loss", "tok/s", "MFU", "outer"):
            t.add_column(col)
        for r in rows_now():
            style = {"RUNNING": "green", "COMPLETED": "cyan",
                     "FAILED": "red", "DIED": "red"}.get(r["status"].split(":")[0])
            t.add_row(r["run"], r["status"], str(r["step"]),
                      fmt(r["loss"], ".4f"), fmt(r["tok_s"], ",.0f"),
                      fmt(r["mfu"], ".3f"), str(r["outer"]), style=style)
        return t

    with Live(build(), refresh_per_second=4) as live:
        try:
            while True:
                time.sleep(interval)
                live.update(build())
        except KeyboardInterrupt:
            pass


# ----------------------------------------------------------------- config
# named contexts with env-var precedence (reference: core/config.py)
from ..utils.contexts import Contexts  # noqa: E402


@config_app.command("view")
def config_view():
    """Effective settings (defaults < context < PRIME_AMD_* env vars)."""
    typer.echo(json.dumps(Contexts().current(), indent=2))


@config_app.command("set")
def config_set(key: str, value: str):
    Contexts().set(key, value)
    secho(f"{key} = {value}", fg="green")


@config_app.command("save")
def config_save(name: str):
    """Save the current config as a named context."""
    try:
        p = Contexts().save(name)
    except ValueError as e:
        secho(str(e), fg="red")
        raise typer.Exit(1)
    secho(f"saved context '{name}' -> {p}", fg="green")


@config_app.command("use")
def config_use(name: str):
    """Switch to a saved context."""
    try:
        Contexts().use(name)
    except (ValueError, FileNotFoundError) as e:
        secho(str(e), fg="red")
        raise typer.Exit(1)
    secho(f"now using context '{name}'", fg="green")


@config_app.command("envs")
def config_envs():
    """List saved contexts."""
    for name in Contexts().list():
        typer.echo(name)


@config_app.command("delete")
def config_delete(name: str):
    try:
        Contexts().delete(name)
    except ValueError as e:
        secho(str(e), fg="red")
        raise typer.Exit(1)
    secho(f"deleted context '{name}'", fg="green")


# ----------------------------------------------------------- prepare-data
@app.command("prepare-data")
def prepare_data(
    inputs: list[str] = typer.Argument(..., help="input UTF-8 text file(s)"),
    out: str = typer.Option(..., "--out", "-o",
                            help="output .bin, or a directory for shards"),
    tokenizer: str = typer.Option(..., help="local tokenizer.json"),
    vocab_threshold: int = typer.Option(65535, help="uint16 if vocab fits"),
    shard_tokens: int = typer.Option(512 * 1024 * 1024,
                                     help="tokens per output shard"),
    workers: int = typer.Option(0, help="parallel tokenizer processes"),
    json_out: bool = typer.Option(False, "--json"),
):
    """Tokenize text into token shards for the token_file dataloader.
    Streams inputs in bounded blocks (multi-GB corpora never load into
    memory), tokenizes blocks in parallel workers, and rolls output
    shards every --shard-tokens."""
    from ..data.prepare import prepare_corpus

    try:
        res = prepare_corpus(inputs, out, tokenizer,
                             vocab_threshold=vocab_threshold,
                             shard_tokens=shard_tokens, workers=workers)
    except FileNotFoundError as e:
        secho(f"input not found: {e}", fg="red")
        raise typer.Exit(1)
    if json_out:
        emit_json(res)
    else:
        secho(f"wrote {res['tokens']:,} tokens (vocab {res['vocab']}, "
              f"{res['dtype']}) to {len(res['shards'])} shard(s)", fg="green")


# ------------------------------------------------------------------- eval
@app.command("eval")
def eval_cmd(
    model: str = typer.Option("llama_150m"),
    data: Optional[str] = typer.Option(None, help="token file (.bin); default synthetic"),
    seq_len: int = typer.Option(1024),
    batches: int = typer.Option(10),
    micro_batch: int = typer.Option(4),
    checkpoint: Optional[str] = typer.Option(None, help="checkpoint dir"),
    json_out: bool = typer.Option(False, "--json"),
):
    """Held-out loss / perplexity (reference: prime eval surface)."""
    import torch

    from ..data import DataConfig
    from ..models import build_model
    from ..models.evaluate import evaluate_perplexity

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    m = build_model(model)
    if dev == "cuda":
        m = m.to(dev, dtype=torch.bfloat16)
        m.reset_rope(dev)
    if checkpoint:
        from ..ckpt import CheckpointManager
        from ..parallel.flat import FlatParamSpace

        flat = FlatParamSpace(m)
        payload = CheckpointManager(checkpoint).load(map_location=dev)
        if payload is None:
            secho("no checkpoint found", fg="red")
            raise typer.Exit(1)
        flat.load_flat_(payload["tensors"]["master32"].to(dev))
    dc = DataConfig(kind="token_file" if data else "synthetic", path=data,
                    seq_len=seq_len, micro_batch_size=micro_batch)
    res = evaluate_perplexity(m, dc, n_batches=batches, device=dev)
    if json_out:
        typer.echo(json.dumps(res))
    else:
        typer.echo(f"loss {res['loss']:.4f}  ppl {res['perplexity']:.2f}  "
                   f"({res['tokens']:,} tokens)")


# --------------------------------------------------------------- generate
@app.command("generate")
def generate_cmd(
    model: str = typer.Option("llama_150m", help="model preset"),
    prompt_tokens: str = typer.Option("1,2,3,4", help="comma-separated token ids"),
    prompt: Optional[str] = typer.Option(None, help="text prompt (needs --tokenizer)"),
    tokenizer: Optional[str] = typer.Option(None, help="local tokenizer.json"),
    max_new: int = typer.Option(32),
    temperature: float = typer.Option(0.0),
    top_k: int = typer.Option(0),
    seed: int = typer.Option(0),
    checkpoint: Optional[str] = typer.Option(None, help="checkpoint dir (latest tag)"),
):
    """Generate tokens with the KV-cache decode path (greedy by default)."""
    import torch

    from ..models import build_model
    from ..models.generate import generate

    tok = None
    if tokenizer:
        from ..utils.tokenizer import load_tokenizer

        tok = load_tokenizer(tokenizer)
    torch.manual_seed(seed)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    m = build_model(model)
    if dev == "cuda":
        m = m.to(dev, dtype=torch.bfloat16)
    m.reset_rope(dev)
    if checkpoint:
        from ..ckpt import CheckpointManager
        from ..parallel.flat import FlatParamSpace

        flat = FlatParamSpace(m)
        payload = CheckpointManager(checkpoint).load(map_location=dev)
        if payload is None:
            secho("no checkpoint found", fg="red")
            raise typer.Exit(1)
        flat.load_flat_(payload["tensors"]["master32"].to(dev))
    if prompt is not None:
        if tok is None:
            secho("--prompt requires --tokenizer", fg="red")
            raise typer.Exit(2)
        from ..utils.tokenizer import encode

        ids = encode(tok, prompt)
    else:
        ids = [int(t) for t in prompt_tokens.split(",")]
    toks = torch.tensor([ids], device=dev)
    out = generate(m, toks, max_new, temperature=temperature, top_k=top_k,
                   seed=seed)
    if tok is not None:
        from ..utils.tokenizer import decode as tok_decode

        typer.echo(tok_decode(tok, [int(t) for t in out[0]]))
    else:
        typer.echo(",".join(str(int(t)) for t in out[0]))


# ------------------------------------------------------------------ export
@app.command("export")
def export_cmd(
    checkpoint: str = typer.Argument(..., help="checkpoint dir"),
    out: str = typer.Argument("model.safetensors"),
    model: str = typer.Option("llama_150m"),
):
    """Export a checkpoint's weights to safetensors (per-parameter bf16)."""
    from ..ckpt.manager import export_safetensors

    try:
        n = export_safetensors(checkpoint, model, out)
    except FileNotFoundError as e:
        secho(str(e), fg="red")
        raise typer.Exit(1)
    secho(f"wrote {n} tensors to {out}", fg="green")


@app.command("import")
def import_cmd(
    weights: str = typer.Argument(..., help="input .safetensors"),
    ckpt_dir: str = typer.Argument(..., help="output checkpoint dir"),
    model: str = typer.Option(..., help="model preset the tensors match"),
    lenient: bool = typer.Option(False, help="keep random init for params "
                                             "missing from the file"),
):
    """Pack pretrained safetensors weights into a resumable/servable
    checkpoint (inverse of `export`)."""
    from ..ckpt.manager import import_safetensors

    try:
        n = import_safetensors(weights, model, ckpt_dir, strict=not lenient)
    except (ValueError, FileNotFoundError) as e:
        secho(str(e), fg="red")
        raise typer.Exit(1)
    secho(f"imported {n} tensors into {ckpt_dir}", fg="green")


# ------------------------------------------------------------------ store
@app.command("store")
def store_cmd(port: int = typer.Option(29777), addr: str = typer.Option("0.0.0.0")):
    """Host a standalone TCPStore for the elastic cross-worker fabric
    (workers point PRIME_GLOBAL_ADDR/PORT here; survives worker churn)."""
    import time as _time

    from torch.distributed import TCPStore

    # the reference MUST be held: an unbound TCPStore master is garbage-
    # collected immediately and the listening socket closes (caught by
    # tests/test_elastic_fsdp.py::test_store_host_failover)
    store = TCPStore(addr, port, is_master=True, wait_for_workers=False)
    secho(f"elastic store listening on {addr}:{port} (ctrl-c to stop)", fg="green")
    try:
        while True:
            _time.sleep(3600)
    except KeyboardInterrupt:
        pass
    finally:
        del store


# ----------------------------------------------------------------- doctor
@app.command("doctor")
def doctor(json_out: bool = typer.Option(False, "--json")):
    """Validate the environment (reference: `prime lab doctor`): toolchain,
    kernel library freshness, GPU/RCCL readiness, elastic store reach."""
    import shutil as _shutil

    checks = []

    def add(name, ok, detail):
        checks.append({"check": name, "ok": bool(ok), "detail": detail})

    hipcc = _shutil.which("hipcc")
    add("hipcc", hipcc is not None, hipcc or "not on PATH (kernel rebuilds impossible)")
    try:
        from ..ops.build import LIB_PATH, needs_build

        stale = needs_build()
        add("kernel .so", LIB_PATH.exists() and not stale,
            f"{LIB_PATH} " + ("fresh (source-hash match)" if not stale
                              else "missing/stale - will rebuild on first use"))
    except Exception as e:  # noqa: BLE001
        add("kernel .so", False, str(e))
    try:
        import torch

        add("torch", True, f"{torch.__version__} (HIP {torch.version.hip})")
        cuda = torch.cuda.is_available()
        detail = "no GPU visible (CPU/gloo plumbing only)"
        if cuda:
            prop = torch.cuda.get_device_properties(0)
            detail = (f"{torch.cuda.device_count()}x {prop.name} "
                      f"({prop.total_memory / 2**30:.0f} GiB, {prop.gcnArchName})")
            ok_arch = "gfx950" in prop.gcnArchName
            add("gfx950", ok_arch, prop.gcnArchName)
        add("gpu", cuda, detail)
        import torch.distributed as dist

        add("rccl backend", dist.is_nccl_available(),
            "torch.distributed nccl(=RCCL) " +
            ("available" if dist.is_nccl_available() else "MISSING"))
        add("ipc mode", os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY") == "0",
            "HSA_ENABLE_IPC_MODE_LEGACY=" +
            os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "<unset>") +
            " (0 required for multi-process GPU work on dmabuf-only hosts)")
    except Exception as e:  # noqa: BLE001
        add("torch", False, str(e))
    addr = os.environ.get("PRIME_GLOBAL_ADDR")
    if addr:
        port = int(os.environ.get("PRIME_GLOBAL_PORT", 29777))
        try:
            with socket.create_connection((addr, port), timeout=3):
                add("elastic store", True, f"{addr}:{port} reachable")
        except OSError as e:
            add("elastic store", False, f"{addr}:{port} unreachable ({e})")
    if json_out:
        emit_json(checks)
    else:
        for c in checks:
            mark, color = ("ok", "green") if c["ok"] else ("!!", "yellow")
            secho(f"  [{mark}] {c['check']:14s} {c['detail']}", fg=color)
    if not all(c["ok"] for c in checks if c["check"] in ("hipcc", "torch")):
        raise typer.Exit(1)


# ------------------------------------------------------------------ serve
@app.command("serve")
def serve_cmd(
    model: str = typer.Option("llama_150m"),
    checkpoint: Optional[str] = typer.Option(None, help="checkpoint dir"),
    tokenizer: Optional[str] = typer.Option(None, help="local tokenizer.json"),
    host: str = typer.Option("127.0.0.1"),
    port: int = typer.Option(8392),
):
    """Serve the model over an OpenAI-style HTTP API (engine-side
    counterpart of the reference's hosted inference surface):
    GET /v1/models, POST /v1/completions, POST /v1/chat/completions."""
    import uvicorn

    from ..serve import create_app, load_model_for_serving

    tok = None
    if tokenizer:
        from ..utils.tokenizer import load_tokenizer

        tok = load_tokenizer(tokenizer)
    try:
        m = load_model_for_serving(model, checkpoint)
    except FileNotFoundError as e:
        secho(str(e), fg="red")
        raise typer.Exit(1)
    secho(f"serving {model} on http://{host}:{port} (ctrl-c to stop)", fg="green")
    uvicorn.run(create_app(m, model, tok), host=host, port=port,
                log_level="warning")


# ------------------------------------------------------------------ bench
@app.command("bench")
def bench(steps: int = 10, warmup: int = 3, model: str = "intellect_10b",
          seq_len: int = 2048, micro_batch: int = 8,
          fp8: bool = typer.Option(False, help="opt-in fp8 GEMM tiers"),
          fp8_dgrad: bool = False, fp8_wgrad: bool = False):
    """Run the flagship benchmark (see bench.py for the driver contract)."""
    import bench as bench_mod  # noqa: F401 — repo-root bench
    cmd = [sys.executable, str(Path(bench_mod.__file__)),
           "--steps", str(steps), "--warmup", str(warmup),
           "--model", model, "--seq-len", str(seq_len),
           "--micro-batch", str(micro_batch)]
    if fp8:
        cmd.append("--fp8")
    if fp8_dgrad:
        cmd.append("--fp8-dgrad")
    if fp8_wgrad:
        cmd.append("--fp8-wgrad")
    subprocess.run(cmd, check=True)


def main() -> None:
    app()


if __name__ == "__main__":
    main()
