"""Training loop: DiLoCo inner/outer over the fused-kernel Llama.

torchrun-able (one process per GPU, RANK/WORLD_SIZE/LOCAL_RANK from env) and
callable in-process for tests. The same loop runs the CPU/gloo plumbing
config (BASELINE.json config 1) and the MI355X configs.
"""
from __future__ import annotations

import math
import os
import time
from pathlib import Path

import torch

from .ckpt.manager import CheckpointManager
from .data import DataConfig, build_dataloader
from .models import build_model
from .parallel.diloco import DilocoOptimizer
from .parallel.flat import FlatParamSpace, FusedAdamW
from .parallel.mesh import ElasticDeviceMesh, MeshConfig
from .utils.config import TrainConfig
from .utils.logging import JsonLogger
from .utils.metrics import MetricsWriter, model_flops_per_token, mfu


def lr_at(step: int, cfg) -> float:
    """Warmup + cosine decay to min_lr_ratio."""
    if step < cfg.warmup_steps:
        return cfg.lr * (step + 1) / max(1, cfg.warmup_steps)
    if not cfg.lr_decay_steps:
        return cfg.lr
    t = min(1.0, (step - cfg.warmup_steps) / max(1, cfg.lr_decay_steps - cfg.warmup_steps))
    return cfg.lr * (cfg.min_lr_ratio + (1 - cfg.min_lr_ratio) * 0.5 * (1 + math.cos(math.pi * t)))


class Trainer:
    def __init__(self, cfg: TrainConfig, run_dir: str | Path | None = None):
        self.cfg = cfg
        # fp8's weight cache is (data_ptr, shape)-keyed like the W^T
        # cache: FSDP's rotating gather pools would alias it (same pool
        # pointer and shape for different layers within one step)
        if cfg.model.fp8 and cfg.parallel.fsdp and cfg.parallel.worker_size > 1:
            raise ValueError("model.fp8 is not supported with parallel.fsdp "
                             "yet (per-step weight caches need stable "
                             "parameter storage)")
        self.stop_requested = False
        self.run_dir = Path(run_dir) if run_dir else Path("runs") / cfg.run_name
        self.run_dir.mkdir(parents=True, exist_ok=True)

        world = int(os.environ.get("WORLD_SIZE", 1))
        self.mesh = ElasticDeviceMesh(MeshConfig(
            # elastic mode: this torchrun job IS one worker; the
            # cross-worker layer is the dynamic TCPStore fabric
            worker_size=world if cfg.parallel.elastic else cfg.parallel.worker_size,
            backend=cfg.parallel.backend,
            quant_outer=cfg.diloco.quant_int8,
        ))
        dev = cfg.device or ("cuda" if torch.cuda.is_available() else "cpu")
        self.device = self.mesh.device if dev == "cuda" else torch.device(dev)

        self.log = JsonLogger(log_path=self.run_dir / f"rank{self.mesh.rank}.log",
                              rank=self.mesh.rank)
        self.metrics = MetricsWriter(
            self.run_dir / "metrics.jsonl" if self.mesh.rank == 0 else None
        )
        from .utils.metrics import WandbWriter

        self.wandb = WandbWriter(
            cfg.metrics.wandb and self.mesh.rank == 0, cfg.run_name,
            {"model": cfg.model.name, "H": cfg.diloco.H},
        )

        torch.manual_seed(cfg.seed)
        # construct directly on the target device: CPU-side init of a 10B
        # model takes minutes; on-GPU init is seconds
        with torch.device(self.device):
            self.model = build_model(
                cfg.model.name,
                activation_checkpointing=cfg.model.activation_checkpointing,
                **cfg.model.overrides,
            )
        self.model_cfg = self.model.cfg
        if self.device.type == "cuda":
            self.model = self.model.to(dtype=torch.bfloat16)
        self.model.reset_rope(self.device)  # tables stay fp32

        self.fsdp = cfg.parallel.fsdp and cfg.parallel.worker_size > 1
        self.seq_par = cfg.parallel.seq_parallel and cfg.parallel.worker_size > 1
        if self.seq_par:
            if self.fsdp:
                raise ValueError("seq_parallel and fsdp are mutually "
                                 "exclusive inner-parallelism modes")
            W = cfg.parallel.worker_size
            if self.model_cfg.n_heads % W or self.model_cfg.n_kv_heads % W:
                raise ValueError(
                    f"seq_parallel needs n_heads ({self.model_cfg.n_heads}) and "
                    f"n_kv_heads ({self.model_cfg.n_kv_heads}) divisible by "
                    f"worker_size ({W})")
            if self.model_cfg.max_seq < cfg.model.seq_len * W:
                raise ValueError(
                    f"seq_parallel full context {cfg.model.seq_len * W} exceeds "
                    f"model max_seq {self.model_cfg.max_seq}")
            for blk in self.model.layers:
                blk.attn.sp_group = self.mesh.local_group
            self.model.sp_pos_offset = self.mesh.worker_rank * cfg.model.seq_len
        if self.fsdp:
            from .parallel.fsdp import ShardedParamSpace

            if not cfg.model.activation_checkpointing:
                self.model.activation_checkpointing = True
            self.flat = ShardedParamSpace(self.model, self.mesh)
            # shard-aligned broadcast across workers (rank i of each worker
            # holds the same shard); same-seed init already matches, this
            # guards against nondeterministic init kernels
            if self.mesh.outer_group is not None:
                import torch.distributed as dist

                dist.broadcast(self.flat.flat_w, src=self.mesh.worker_rank,
                               group=self.mesh.outer_group)
                self.flat.master32.copy_(self.flat.flat_w.float())
        else:
            # broadcast initial params so every rank starts identical
            self.flat = FlatParamSpace(self.model)
            if self.mesh.initialized:
                import torch.distributed as dist

                dist.broadcast(self.flat.flat_w, src=0)
                self.flat.master32.copy_(self.flat.flat_w.float())

        # layout-tuned linear backward (cached W^T) needs stable param
        # storage: FSDP re-materializes block weights into rotating pool
        # buffers, so only the non-sharded path may cache by tensor id
        from . import ops as _ops

        _ops.set_linear_tuned(self.device.type == "cuda" and not self.fsdp)
        _ops.set_linear_fp8(cfg.model.fp8 and self.device.type == "cuda",
                            dgrad=cfg.model.fp8_dgrad,
                            wgrad=cfg.model.fp8_wgrad)

        self.inner = FusedAdamW(
            self.flat, lr=cfg.optim.lr, betas=tuple(cfg.optim.betas),
            eps=cfg.optim.eps, weight_decay=cfg.optim.weight_decay,
        )
        self.elastic = None
        self.shard_client = None
        if cfg.parallel.elastic and self.mesh.is_leader:
            from .parallel.elastic import ElasticWorker

            self.elastic = ElasticWorker(
                worker_name=cfg.run_name,
                host_store=os.environ.get("PRIME_GLOBAL_HOST", "0") == "1",
                heartbeat_interval=cfg.parallel.heartbeat_interval,
                heartbeat_timeout=cfg.parallel.heartbeat_timeout,
                ckpt_provider=lambda: self.diloco.live_state(),
            )
        elif cfg.parallel.elastic and self.fsdp:
            # non-leader FSDP rank: own store client + shard ckpt server +
            # shard-aligned cross-worker ring (see parallel/elastic.py)
            from .parallel.elastic import ElasticShardClient

            self.shard_client = ElasticShardClient(
                shard_rank=self.mesh.worker_rank,
                ckpt_provider=lambda: self.diloco.live_state(),
            )
        self.diloco = DilocoOptimizer(
            self.flat, self.mesh, self.inner,
            outer_lr=cfg.diloco.outer_lr,
            outer_momentum=cfg.diloco.outer_momentum,
            H=cfg.diloco.H if cfg.diloco.enabled else 10**9,
            outer_device=cfg.diloco.outer_device,
            elastic=self.elastic,
            elastic_mode=cfg.parallel.elastic,
            shard_client=self.shard_client,
            sharded=self.fsdp,
        )
        if cfg.parallel.elastic:
            if self.diloco.init_bootstrap():
                self.log.info(
                    "live-recovered from peer at outer step "
                    f"{self.diloco.outer_step_count}"
                )

        # DiLoCo-worker data index: distinct workers must see distinct
        # streams. Elastic workers are separate torchrun jobs with identical
        # rank/world, so fold a worker-unique index (config override, else
        # the elastic join sequence) into the seed.
        widx = cfg.data.worker_index
        if widx is None and cfg.parallel.elastic:
            widx = self.elastic.join_seq if self.elastic is not None else 0
            if self.mesh.local_group is not None:
                import torch.distributed as dist

                t = torch.tensor([widx], dtype=torch.int64, device=self.device)
                # elastic: this whole torchrun job is one worker, leader = 0
                dist.broadcast(t, src=0, group=self.mesh.local_group)
                widx = int(t[0])
        sp_W = cfg.parallel.worker_size if self.seq_par else 1
        data_cfg = DataConfig(
            kind=cfg.data.kind, path=cfg.data.path,
            # SP: every rank of the worker loads the SAME full-context
            # rows and trains on its own seq_len-token slice
            seq_len=cfg.model.seq_len * sp_W,
            micro_batch_size=cfg.data.micro_batch_size,
            seed=cfg.data.seed + 100003 * (widx or 0),
            shuffle=cfg.data.shuffle,
        )
        self.data = build_dataloader(
            data_cfg, self.model_cfg.vocab_size,
            shard=self.mesh.worker_id if self.seq_par else self.mesh.rank,
            n_shards=max(1, self.mesh.n_workers if self.seq_par
                         else self.mesh.world_size),
        )

        self.ckpt = None
        if cfg.checkpoint.interval > 0 or cfg.checkpoint.resume:
            self.ckpt = CheckpointManager(
                cfg.checkpoint.path or (self.run_dir / "ckpt"),
                keep=cfg.checkpoint.keep, async_save=cfg.checkpoint.async_save,
                worker_id=self.mesh.worker_id, is_leader=self.mesh.is_leader,
                remote_path=cfg.checkpoint.remote_path,
                # FSDP: optimizer/outer state is per-shard -> every rank saves
                shard_rank=self.mesh.worker_rank if self.fsdp else 0,
            )
        self.step_count = 0
        if cfg.checkpoint.resume and self.ckpt:
            self._resume(cfg.checkpoint.resume)

        self.tokens_per_step = (
            cfg.data.micro_batch_size * cfg.data.grad_accum * cfg.model.seq_len
            * max(1, self.mesh.world_size)
        )
        self.flops_per_token = model_flops_per_token(self.model_cfg, cfg.model.seq_len)

    # ---------------------------------------------------------------- steps
    def train_step(self) -> torch.Tensor:
        cfg = self.cfg
        self.inner.lr = lr_at(self.step_count, cfg.optim)
        self.flat.zero_grad()
        loss_acc = None
        for _ in range(cfg.data.grad_accum):
            x, y = self.data.next_batch(self.device)
            if self.seq_par:
                sl = cfg.model.seq_len
                r = self.mesh.worker_rank
                x = x[:, r * sl : (r + 1) * sl]
                y = y[:, r * sl : (r + 1) * sl]
            loss = self.model.loss(x, y) / cfg.data.grad_accum
            loss.backward()
            loss_acc = loss.detach() if loss_acc is None else loss_acc + loss.detach()
        if self.fsdp:
            self.flat.finalize_grads()  # sharded units were reduce-scattered in bwd
        else:
            self.mesh.local_allreduce_grad(self.flat.flat_grad)
        gscale = None
        if cfg.optim.grad_clip > 0:
            if self.device.type == "cuda" and not self.fsdp:
                # deferred clip: scale folded into the fused AdamW pass
                gscale = self.flat.grad_clip_scale(cfg.optim.grad_clip)
            else:
                self.flat.clip_grad_norm_(cfg.optim.grad_clip)
        did_outer = self.diloco.step(gscale)
        self.step_count += 1
        if did_outer:
            self._maybe_checkpoint()
        return loss_acc

    def _maybe_checkpoint(self) -> None:
        if not self.ckpt or self.cfg.checkpoint.interval <= 0:
            return
        if self.diloco.outer_step_count % self.cfg.checkpoint.interval != 0:
            return
        if not (self.mesh.is_leader or self.fsdp):
            return  # FSDP: every rank owns a distinct shard and must save
        self.save_checkpoint()

    def save_checkpoint(self) -> None:
        t0 = time.perf_counter()
        tensors = {
            "master32": self.flat.master32,
            "theta_outer": self.diloco.theta_outer,
            "outer_buf": self.diloco.outer_buf,
            "adam_m": self.inner.m,
            "adam_v": self.inner.v,
        }
        meta = {
            "inner_step": self.diloco.inner_step_count,
            "outer_step": self.diloco.outer_step_count,
            "adam_step": self.inner.step_count,
            "step_count": self.step_count,
            "data_state": self.data.state_dict(),
            "model": self.model_cfg.name,
        }
        self.ckpt.save(self.diloco.outer_step_count, tensors, meta)
        self.log.info(
            f"checkpoint staged at outer step {self.diloco.outer_step_count} "
            f"({time.perf_counter() - t0:.3f}s to stage)", type_="checkpoint",
        )

    def _resume(self, tag: str) -> None:
        payload = self.ckpt.load(None if tag == "latest" else tag,
                                 map_location=self.device)
        if payload is None:
            self.log.warning(f"no checkpoint found for resume='{tag}'")
            return
        t = payload["tensors"]
        m = payload["meta"]
        self.flat.load_flat_(t["master32"].to(self.flat.master32.device))
        self.diloco.theta_outer.copy_(t["theta_outer"])
        self.diloco.outer_buf.copy_(t["outer_buf"])
        self.inner.m.copy_(t["adam_m"])
        self.inner.v.copy_(t["adam_v"])
        self.inner.step_count = m["adam_step"]
        self.diloco.inner_step_count = m["inner_step"]
        self.diloco.outer_step_count = m["outer_step"]
        self.step_count = m["step_count"]
        self.data.load_state_dict(m["data_state"])
        self.log.info(f"resumed from outer step {m['outer_step']} (inner {m['inner_step']})")

    # ----------------------------------------------------------------- run
    def run(self) -> dict:
        cfg = self.cfg
        self.log.info(
            f"starting run '{cfg.run_name}': model={self.model_cfg.name} "
            f"({self.model_cfg.n_params()/1e6:.1f}M params), "
            f"workers={self.mesh.n_workers}x{cfg.parallel.worker_size}, "
            f"H={cfg.diloco.H if cfg.diloco.enabled else 'off'}, device={self.device}"
        )
        profiler = None
        if cfg.metrics.torch_profiler_steps > 0 and self.mesh.rank == 0:
            from torch.profiler import ProfilerActivity, profile, schedule

            profiler = profile(
                activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                schedule=schedule(wait=2, warmup=1,
                                  active=cfg.metrics.torch_profiler_steps),
                on_trace_ready=lambda p: p.export_chrome_trace(
                    str(self.run_dir / "trace.json")),
            )
            profiler.start()
        t_start = time.perf_counter()
        window_t0, window_steps = t_start, 0
        last_loss = float("nan")
        loss = None
        executed = 0  # steps run THIS session (early stop / resume aware)
        # cfg.steps is the run's TOTAL step target: a resumed run trains
        # only the remainder, so a crash-restart loop converges on the
        # configured total instead of extending it every restart
        for _ in range(max(0, cfg.steps - self.step_count)):
            if profiler is not None:
                profiler.step()
            stop = self.stop_requested
            if self.mesh.initialized and self.mesh.world_size > 1:
                # consensus stop: SIGTERM lands on ranks at different
                # steps; without agreement the early stopper leaves peers
                # hanging in the next collective
                import torch.distributed as dist

                t = torch.tensor([1 if stop else 0], device=self.device
                                 if self.device.type == "cuda" else "cpu")
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
                stop = bool(int(t[0]))
            if stop:
                self.log.warning("stop requested: saving checkpoint and exiting")
                self.stop_requested = True
                if self.ckpt:
                    self.save_checkpoint()
                break
            loss = self.train_step()
            executed += 1
            window_steps += 1
            if self.step_count % cfg.metrics.log_interval == 0:
                if self.device.type == "cuda":
                    torch.cuda.synchronize()
                now = time.perf_counter()
                dt = now - window_t0
                tps = self.tokens_per_step * window_steps / dt
                tps_gpu = tps / max(1, self.mesh.world_size)
                last_loss = float(loss)
                if last_loss != last_loss or last_loss in (float("inf"),
                                                           float("-inf")):
                    from .utils.failures import NonFiniteLossError

                    raise NonFiniteLossError(
                        f"loss is {last_loss} at step {self.step_count}"
                    )
                self.wandb.write(
                    self.step_count, loss=last_loss, tokens_per_sec=tps,
                    mfu=mfu(tps_gpu, self.flops_per_token),
                )
                self.metrics.write(
                    self.step_count, loss=last_loss, tokens_per_sec=tps,
                    tokens_per_sec_per_gpu=tps_gpu,
                    mfu=mfu(tps_gpu, self.flops_per_token),
                    lr=self.inner.lr, ms_per_step=1000 * dt / window_steps,
                    outer_steps=self.diloco.outer_step_count,
                )
                self.log.progress(
                    f"step {self.step_count}/{cfg.steps} loss={last_loss:.4f} "
                    f"tok/s={tps:,.0f} mfu={mfu(tps_gpu, self.flops_per_token):.3f}"
                )
                window_t0, window_steps = now, 0
        if profiler is not None:
            profiler.stop()
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        total_t = time.perf_counter() - t_start
        if loss is not None:
            last_loss = float(loss)
        if self.ckpt:
            self.ckpt.wait()
        result = {
            "steps": self.step_count,
            "outer_steps": self.diloco.outer_step_count,
            "loss": last_loss,
            "total_time_s": total_t,
            "tokens_per_sec": self.tokens_per_step * executed / total_t,
        }
        self.log.info(f"run complete: {result}", type_="result")
        return result

    def close(self, destroy_pg: bool = True) -> None:
        self.wandb.close()
        if self.elastic is not None:
            self.elastic.close(leaving=True)
        if self.shard_client is not None:
            self.shard_client.close()
        if self.ckpt:
            self.ckpt.wait()
        self.metrics.close()
        self.log.close()
        if destroy_pg:
            self.mesh.destroy()


def train_from_config(cfg: TrainConfig, run_dir=None) -> dict:
    tr = Trainer(cfg, run_dir)
    try:
        return tr.run()
    finally:
        tr.close()


def main() -> None:
    import argparse

    from .utils.config import load_config

    ap = argparse.ArgumentParser(description="prime_amd trainer (torchrun-able)")
    ap.add_argument("config", help="TOML run config")
    ap.add_argument("--run-dir", default=None)
    args = ap.parse_args()
    cfg = load_config(args.config)
    train_from_config(cfg, args.run_dir)


if __name__ == "__main__":
    main()
