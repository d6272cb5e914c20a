"""Async distributed checkpointing.

GPU path: device tensors are staged into reusable PINNED host buffers with
hipMemcpyAsync on a dedicated side stream (no stall of the compute stream),
then a background thread waits on the HIP event and persists with
torch.save. Checkpoints are taken at outer-step boundaries so every worker
snapshot is outer-consistent; retention keeps the newest `keep` tags.

Layout:  <path>/step_<outer>/worker<id>.pt   + meta.json
         <path>/latest -> step_<outer>       (symlink)
Each DiLoCo worker's leader rank persists {master32/theta_outer/outer_buf,
inner m/v/step, dataloader, config}; non-leader ranks of a sharded worker
persist their own shard file.
"""
from __future__ import annotations

import json
import os
import shutil
import threading
import time
from pathlib import Path

import torch


class _PinnedStager:
    """Reusable pinned-host mirror of a dict of device tensors. Pinning is
    best-effort: a 10B checkpoint stages ~200 GB — if the pin fails (or
    host RAM is tight) the buffer falls back to pageable memory (slower
    D2H, still async-ish) instead of risking the OOM killer."""

    def __init__(self):
        self.buffers: dict[str, torch.Tensor] = {}
        self._pinned_bytes = 0  # cumulative: per-tensor checks overcommit

    def _alloc(self, t: torch.Tensor) -> torch.Tensor:
        try:
            avail = None
            for line in open("/proc/meminfo"):
                if line.startswith("MemAvailable:"):
                    avail = int(line.split()[1]) * 1024
                    break
            need = t.numel() * t.element_size()
            if avail is None or self._pinned_bytes + need < 0.3 * avail:
                buf = torch.empty_like(t, device="cpu", pin_memory=True)
                self._pinned_bytes += need
                return buf
        except (OSError, RuntimeError):
            pass
        return torch.empty_like(t, device="cpu")

    def stage(self, state: dict[str, torch.Tensor], stream: torch.cuda.Stream) -> dict:
        out = {}
        with torch.cuda.stream(stream):
            for k, t in state.items():
                buf = self.buffers.get(k)
                if buf is None or buf.shape != t.shape or buf.dtype != t.dtype:
                    buf = self._alloc(t)
                    self.buffers[k] = buf
                buf.copy_(t, non_blocking=True)
                out[k] = buf
        return out

    def stage_cpu(self, state: dict[str, torch.Tensor]) -> dict:
        """Synchronous copy of CPU tensors into reusable staging buffers so
        the caller may keep mutating the originals while a background
        thread serializes the snapshot."""
        out = {}
        for k, t in state.items():
            buf = self.buffers.get(k)
            if buf is None or buf.shape != t.shape or buf.dtype != t.dtype:
                buf = torch.empty_like(t, device="cpu")
                self.buffers[k] = buf
            buf.copy_(t)
            out[k] = buf
        return out


class CheckpointManager:
    def __init__(self, path: str | Path, keep: int = 3, async_save: bool = True,
                 worker_id: int = 0, is_leader: bool = True,
                 remote_path: str | Path | None = None, shard_rank: int = 0):
        self.root = Path(path)
        self.keep = keep
        self.async_save = async_save
        self.worker_id = worker_id
        self.is_leader = is_leader
        # FSDP: every rank of a worker persists its own shard file
        self.shard_rank = shard_rank
        self.remote = Path(remote_path) if remote_path else None
        self._stager = _PinnedStager()
        self._stream = None
        self._pending: threading.Thread | None = None
        self.root.mkdir(parents=True, exist_ok=True)

    # ------------------------------------------------------------- save
    @staticmethod
    def _fs_is_ram(path: Path) -> bool:
        try:
            rp = os.path.realpath(path)
            best, best_t = "", ""
            for line in open("/proc/mounts"):
                parts = line.split()
                if len(parts) >= 3 and (rp == parts[1] or
                                        rp.startswith(parts[1].rstrip("/") + "/")):
                    if len(parts[1]) > len(best):
                        best, best_t = parts[1], parts[2]
            return best_t in ("tmpfs", "ramfs")
        except OSError:
            return False

    def _check_capacity(self, tag_dir: Path, need: int) -> None:
        """Refuse loudly instead of killing the node: a 10B checkpoint is
        ~200 GB — written to a tmpfs path it silently eats host RAM until
        the OOM killer takes the machine down (observed on a soak run)."""
        try:
            free = shutil.disk_usage(tag_dir).free
        except OSError:
            return
        kind = "tmpfs (RAM-backed!)" if self._fs_is_ram(tag_dir) else "disk"
        if need * 1.05 > free:
            raise RuntimeError(
                f"checkpoint needs ~{need / 2**30:.1f} GiB but {tag_dir} "
                f"({kind}) has {free / 2**30:.1f} GiB free — refusing to "
                "write; point checkpoint.path at a filesystem with room"
            )
        if self._fs_is_ram(tag_dir):
            avail = None
            try:
                for line in open("/proc/meminfo"):
                    if line.startswith("MemAvailable:"):
                        avail = int(line.split()[1]) * 1024
                        break
            except OSError:
                pass
            if avail is not None and need * 1.2 > avail:
                raise RuntimeError(
                    f"checkpoint path {tag_dir} is RAM-backed (tmpfs) and "
                    f"~{need / 2**30:.1f} GiB would exhaust MemAvailable "
                    f"({avail / 2**30:.1f} GiB) — refusing; use a disk path"
                )

    def save(self, outer_step: int, tensors: dict[str, torch.Tensor],
             meta: dict) -> None:
        """Snapshot `tensors` (+ JSON-serializable `meta`)."""
        self.wait()
        tag_dir = self.root / f"step_{outer_step}"
        tag_dir.mkdir(parents=True, exist_ok=True)
        fname = tag_dir / self._fname()
        self._check_capacity(
            tag_dir, sum(t.numel() * t.element_size() for t in tensors.values()))

        on_gpu = any(t.is_cuda for t in tensors.values())
        if on_gpu and self.async_save:
            if self._stream is None:
                self._stream = torch.cuda.Stream()
            self._stream.wait_stream(torch.cuda.current_stream())
            host = self._stager.stage(
                {k: t for k, t in tensors.items() if t.is_cuda}, self._stream
            )
            # CPU tensors (host-offloaded theta_outer/outer_buf) are copied
            # into reusable staging buffers: a later outer step must not
            # mutate what the background torch.save is serializing.
            host.update(self._stager.stage_cpu(
                {k: t for k, t in tensors.items() if not t.is_cuda}
            ))
            ev = torch.cuda.Event()
            ev.record(self._stream)
            # The compute stream must not mutate master32/m/v (the very next
            # train_step's fused AdamW does) while the side stream is still
            # reading them for the D2H staging copies.
            torch.cuda.current_stream().wait_event(ev)

            def _persist():
                ev.synchronize()
                torch.save({"tensors": host, "meta": meta}, fname)
                self._finalize(tag_dir, outer_step, meta)

            self._pending = threading.Thread(target=_persist, daemon=True)
            self._pending.start()
        else:
            host = {k: t.detach().cpu() for k, t in tensors.items()}
            torch.save({"tensors": host, "meta": meta}, fname)
            self._finalize(tag_dir, outer_step, meta)

    def _fname(self) -> str:
        if self.shard_rank:
            return f"worker{self.worker_id}_shard{self.shard_rank}.pt"
        return f"worker{self.worker_id}.pt"

    def _finalize(self, tag_dir: Path, outer_step: int, meta: dict) -> None:
        if self.remote is not None:
            # remote copy (e.g. NFS / fuse-mounted object store): performed
            # on the background persist thread, never the training thread
            dst = self.remote / tag_dir.name
            dst.mkdir(parents=True, exist_ok=True)
            for f in tag_dir.glob(self._fname()):
                shutil.copy2(f, dst / f.name)
        if self.is_leader and self.worker_id == 0:
            (tag_dir / "meta.json").write_text(
                json.dumps({"outer_step": outer_step, "time": time.time(), **{
                    k: v for k, v in meta.items()
                    if isinstance(v, (int, float, str, bool, type(None)))
                }})
            )
            if self.remote is not None and (tag_dir / "meta.json").exists():
                shutil.copy2(tag_dir / "meta.json",
                             self.remote / tag_dir.name / "meta.json")
            latest = self.root / "latest"
            tmp = self.root / ".latest.tmp"
            if tmp.is_symlink() or tmp.exists():
                tmp.unlink()
            tmp.symlink_to(tag_dir.name)
            os.replace(tmp, latest)
            self._retain()

    def _retain(self) -> None:
        tags = sorted(
            (d for d in self.root.iterdir() if d.is_dir() and d.name.startswith("step_")),
            key=lambda d: int(d.name.split("_")[1]),
        )
        for d in tags[: -self.keep] if self.keep > 0 else []:
            shutil.rmtree(d, ignore_errors=True)

    def wait(self) -> None:
        if self._pending is not None:
            self._pending.join()
            self._pending = None

    # ------------------------------------------------------------- load
    def latest_tag(self) -> Path | None:
        latest = self.root / "latest"
        if latest.exists():
            return latest.resolve()
        tags = sorted(
            (d for d in self.root.iterdir() if d.is_dir() and d.name.startswith("step_")),
            key=lambda d: int(d.name.split("_")[1]),
        )
        return tags[-1] if tags else None

    def load(self, tag: str | Path | None = None, map_location="cpu") -> dict | None:
        tag_dir = Path(tag) if tag else self.latest_tag()
        if tag_dir is None or not tag_dir.exists():
            return None
        fname = tag_dir / self._fname()
        if not fname.exists():
            # joining worker: adopt any peer's snapshot (live recovery fallback)
            cands = sorted(tag_dir.glob("worker*.pt"))
            if not cands:
                return None
            fname = cands[0]
        return torch.load(fname, map_location=map_location, weights_only=False)


def export_safetensors(ckpt_dir, model_name: str, out_path,
                       overrides: dict | None = None) -> int:
    """Explode a checkpoint's flat fp32 master into per-parameter bf16
    tensors and write a safetensors file (interop export). Returns the
    number of tensors written."""
    from safetensors.torch import save_file

    from ..models import build_model
    from ..parallel.flat import FlatParamSpace

    mgr = CheckpointManager(ckpt_dir)
    payload = mgr.load()
    if payload is None:
        raise FileNotFoundError(f"no checkpoint under {ckpt_dir}")
    model = build_model(model_name, **(overrides or {}))
    flat = FlatParamSpace(model)
    flat.load_flat_(payload["tensors"]["master32"])
    tensors = {}
    for n, _ in flat.params:
        o, k, shp = flat.offsets[n]
        tensors[n] = flat.flat_w[o : o + k].view(shp).to(torch.bfloat16).contiguous()
    save_file(tensors, str(out_path), metadata={"model": model_name})
    return len(tensors)


def import_safetensors(in_path, model_name: str, ckpt_dir,
                       overrides: dict | None = None,
                       strict: bool = True) -> int:
    """Inverse of export_safetensors: pack per-parameter tensors from a
    safetensors file (e.g. published pretrained weights) into a flat-
    master checkpoint this engine resumes/serves from. Tensors must match
    the named config's parameter names/shapes (tied lm_head may be
    absent). Returns the number of tensors consumed."""
    from safetensors.torch import load_file

    from ..models import build_model
    from ..parallel.flat import FlatParamSpace

    tensors = load_file(str(in_path))
    model = build_model(model_name, **(overrides or {}))
    flat = FlatParamSpace(model)
    seen = 0
    missing = []
    for n, _ in flat.params:
        o, k, shp = flat.offsets[n]
        src = tensors.get(n)
        if src is None and n == "lm_head.weight" and model.cfg.tie_embeddings:
            src = tensors.get("tok_embeddings.weight")
        if src is None:
            missing.append(n)
            continue
        if tuple(src.shape) != tuple(shp):
            raise ValueError(f"{n}: shape {tuple(src.shape)} != {tuple(shp)}")
        flat.master32[o : o + k].copy_(src.flatten().float())
        seen += 1
    if missing and strict:
        raise ValueError(f"missing tensors for {len(missing)} params "
                         f"(first: {missing[:3]}); use strict=False to "
                         "keep random init for them")
    flat.flat_w.copy_(flat.master32.to(flat.flat_w.dtype))
    mgr = CheckpointManager(ckpt_dir, async_save=False)
    mgr.save(0, {
        "master32": flat.master32,
        "theta_outer": flat.master32.clone(),
        "outer_buf": torch.zeros_like(flat.master32),
        "adam_m": torch.zeros_like(flat.master32),
        "adam_v": torch.zeros_like(flat.master32),
    }, {
        "inner_step": 0, "outer_step": 0, "adam_step": 0, "step_count": 0,
        "data_state": {"batch_idx": 0}, "model": model_name,
        "imported_from": str(in_path),
    })
    return seen
