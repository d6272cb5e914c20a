"""Strict TOML run configs, mirroring the reference CLI's config discipline:
pydantic models with extra="forbid" and friendly error rendering
(reference: packages/prime/src/prime_cli/commands/rl.py:765-827 — strict
schema + two-phase lenient peek + human-readable validation errors).
"""
from __future__ import annotations

try:
    import tomllib  # py>=3.11
except ModuleNotFoundError:  # py3.10: tomli is API-compatible
    import tomli as tomllib
from pathlib import Path
from typing import Literal

from pydantic import BaseModel, ConfigDict, Field, ValidationError


class _Strict(BaseModel):
    model_config = ConfigDict(extra="forbid")


class ModelConfig(_Strict):
    name: str = "llama_150m"
    seq_len: int = Field(2048, gt=0)
    activation_checkpointing: bool = False
    # opt-in mixed precision: forward linears in OCP fp8-e4m3 (per-tensor
    # dynamic scaling; master/grads/backward stay bf16/fp32) — ~1.8x the
    # bf16 GEMM rate on MI355X. The headline bench stays bf16.
    fp8: bool = False
    # second fp8 tier: dX (dgrad) GEMMs in e5m2 x e4m3 as well — dW and
    # the optimizer stay bf16/fp32 (the highest-risk path never drops)
    fp8_dgrad: bool = False
    # third tier (full TE-style recipe): dW GEMMs in fp8 too; the fp32
    # master still integrates every update
    fp8_wgrad: bool = False
    overrides: dict = Field(default_factory=dict)


class DataSection(_Strict):
    kind: Literal["synthetic", "token_file"] = "synthetic"
    path: str | None = None
    micro_batch_size: int = Field(4, gt=0)
    grad_accum: int = Field(1, gt=0)
    seed: int = 1234
    shuffle: bool = True
    # explicit DiLoCo-worker data index; elastic mode derives one from the
    # join sequence when unset so workers never train identical streams
    worker_index: int | None = None


class OptimConfig(_Strict):
    lr: float = 3e-4
    betas: tuple[float, float] = (0.9, 0.95)
    eps: float = 1e-8
    weight_decay: float = 0.1
    warmup_steps: int = Field(100, ge=0)
    lr_decay_steps: int | None = None
    min_lr_ratio: float = Field(0.1, ge=0, le=1)
    grad_clip: float = Field(1.0, ge=0)


class DilocoConfig(_Strict):
    enabled: bool = True
    H: int = Field(100, gt=0)
    outer_lr: float = 0.7
    outer_momentum: float = 0.9
    quant_int8: bool = True
    outer_device: Literal["auto", "gpu", "host"] = "auto"


class ParallelConfig(_Strict):
    worker_size: int = Field(1, gt=0)  # GPUs per DiLoCo worker
    fsdp: bool = False                # shard params across the worker
    # sequence (context) parallelism: the worker's ranks each hold
    # seq_len tokens of one (worker_size x seq_len)-token context;
    # attention spans the full context via all-to-alls (Ulysses style)
    seq_parallel: bool = False
    backend: str | None = None        # nccl | gloo (default: auto)
    elastic: bool = False
    heartbeat_interval: float = Field(5.0, gt=0)
    heartbeat_timeout: float = Field(30.0, gt=0)


class CheckpointConfig(_Strict):
    interval: int = Field(0, ge=0)    # outer steps between checkpoints; 0=off
    path: str | None = None
    remote_path: str | None = None    # secondary copy (NFS/fuse mount)
    keep: int = Field(3, ge=0)
    async_save: bool = True
    resume: str | None = None


class MetricsConfig(_Strict):
    log_interval: int = Field(10, gt=0)
    jsonl: bool = True
    wandb: bool = False
    torch_profiler_steps: int = Field(0, ge=0)  # trace steps [3, 3+N) to <run>/trace.json


class TrainConfig(_Strict):
    run_name: str = "run"
    steps: int = Field(100, ge=0)
    seed: int = 1234
    device: str | None = None         # cuda | cpu (default: auto)
    model: ModelConfig = Field(default_factory=ModelConfig)
    data: DataSection = Field(default_factory=DataSection)
    optim: OptimConfig = Field(default_factory=OptimConfig)
    diloco: DilocoConfig = Field(default_factory=DilocoConfig)
    parallel: ParallelConfig = Field(default_factory=ParallelConfig)
    checkpoint: CheckpointConfig = Field(default_factory=CheckpointConfig)
    metrics: MetricsConfig = Field(default_factory=MetricsConfig)


class ConfigError(Exception):
    pass


def _friendly(e: ValidationError, path: str) -> str:
    lines = [f"invalid config {path}:"]
    for err in e.errors():
        loc = ".".join(str(x) for x in err["loc"]) or "<root>"
        lines.append(f"  - {loc}: {err['msg']}")
    return "\n".join(lines)


def load_config(path: str | Path) -> TrainConfig:
    p = Path(path)
    if not p.exists():
        raise ConfigError(f"config file not found: {p}")
    try:
        raw = tomllib.loads(p.read_text())
    except tomllib.TOMLDecodeError as e:
        raise ConfigError(f"TOML parse error in {p}: {e}") from e
    try:
        return TrainConfig.model_validate(raw)
    except ValidationError as e:
        raise ConfigError(_friendly(e, str(p))) from e


def default_config_toml(name: str = "llama_150m") -> str:
    """Template generator (reference: generate_rl_config_template)."""
    return f'''# prime_amd training run config
run_name = "{name}"
steps = 1000

[model]
name = "{name}"
seq_len = 2048
activation_checkpointing = false

[data]
kind = "synthetic"          # synthetic | token_file
micro_batch_size = 4
grad_accum = 1

[optim]
lr = 3e-4
warmup_steps = 100
grad_clip = 1.0

[diloco]
enabled = true
H = 100
outer_lr = 0.7
outer_momentum = 0.9
quant_int8 = true

[parallel]
worker_size = 1             # GPUs per DiLoCo worker
fsdp = false
elastic = false

[checkpoint]
interval = 0                # outer steps between checkpoints (0 = off)
keep = 3
async_save = true

[metrics]
log_interval = 10
'''
