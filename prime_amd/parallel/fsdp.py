"""FSDP2-style intra-worker parameter sharding.

MI355X-first design, built directly on RCCL-over-xGMI collectives rather
than a port of torch FSDP:
  - the SHARD UNIT is one transformer block: big contiguous all-gathers
    (tens of MB) that match xGMI's per-link bandwidth profile, not
    per-tensor chatter.
  - each rank permanently owns one contiguous flat buffer holding
    [block0_shard | block1_shard | ... | replicated(embed/head/norm)] in
    bf16, plus fp32 master/m/v of the same shape: the fused AdamW remains
    ONE kernel launch over everything this rank owns.
  - forward/backward: a block's params are materialized by
    all_gather(+repoint views) right before use and dropped right after;
    backward re-materializes via the activation-checkpoint recompute (fsdp
    requires activation_checkpointing=True) and reduce-scatters the
    block's grads from per-param post-accumulate-grad hooks once the last
    grad of the unit lands.
  - gloo fallbacks (CPU plumbing tests): all_gather into chunk views and
    all_reduce+slice instead of the *_into_tensor fused collectives.

The resulting object quacks like FlatParamSpace (flat_w / flat_grad /
master32 / numel_padded / load_flat_), so FusedAdamW and DilocoOptimizer
work unchanged on the shard space — DiLoCo's outer ring then runs over
the mesh's shard-aligned outer groups.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from .mesh import ElasticDeviceMesh

ALIGN = 64


def _all_gather_flat(full: torch.Tensor, shard: torch.Tensor, group,
                     async_op: bool = False):
    if dist.get_backend(group) == "nccl":
        return dist.all_gather_into_tensor(full, shard, group=group,
                                           async_op=async_op)
    W = dist.get_world_size(group)
    chunks = list(full.chunk(W))
    return dist.all_gather(chunks, shard.contiguous(), group=group,
                           async_op=async_op)


def _reduce_scatter_flat(out_shard: torch.Tensor, grad_full: torch.Tensor,
                         group, async_op: bool = False):
    if dist.get_backend(group) == "nccl":
        return dist.reduce_scatter_tensor(out_shard, grad_full, group=group,
                                          async_op=async_op)
    # gloo: all_reduce then slice (no reduce_scatter); the slice-copy must
    # happen after completion, so async falls back to sync here
    dist.all_reduce(grad_full, group=group)
    W = dist.get_world_size(group)
    r = dist.get_rank(group)
    n = grad_full.numel() // W
    out_shard.copy_(grad_full[r * n : (r + 1) * n])
    return None


class _Unit:
    """One shard unit (a transformer block)."""

    def __init__(self, idx: int, module: nn.Module, params: list[tuple[str, nn.Parameter]]):
        self.idx = idx
        self.module = module
        self.params = params
        self.offsets: dict[str, tuple[int, int, torch.Size]] = {}
        off = 0
        for n, p in params:
            k = p.numel()
            self.offsets[n] = (off, k, p.shape)
            off += (k + ALIGN - 1) // ALIGN * ALIGN
        self.numel = off            # unpadded-to-world unit size
        self.numel_padded = off     # set by the space (pad to W*ALIGN)
        self.shard_off = 0          # offset of my shard in the space flat_w
        self.shard_len = 0
        self.names_by_id = {id(p): n for n, p in params}
        self.pending_grads = 0


class ShardedParamSpace:
    """Block-sharded parameter space over the worker's local group."""

    def __init__(self, model, mesh: ElasticDeviceMesh):
        if mesh.local_group is None:
            raise ValueError("fsdp requires worker_size > 1")
        if not getattr(model, "activation_checkpointing", False):
            raise ValueError(
                "fsdp requires activation_checkpointing=True: block weights "
                "are re-gathered during the checkpoint recompute; a plain "
                "backward would read released weight buffers"
            )
        self.mesh = mesh
        self.group = mesh.local_group
        self.W = mesh.cfg.worker_size
        self.r = mesh.worker_rank
        self.module = model
        self.device = next(model.parameters()).device
        self.dtype = next(model.parameters()).dtype

        block_param_ids = set()
        self.units: list[_Unit] = []
        for i, blk in enumerate(model.layers):
            params = [(f"layers.{i}.{n}", p) for n, p in blk.named_parameters()]
            for _, p in params:
                block_param_ids.add(id(p))
            self.units.append(_Unit(i, blk, params))
        # replicated remainder (embeddings, final norm, lm_head)
        seen = set()
        self.repl_params: list[tuple[str, nn.Parameter]] = []
        for n, p in model.named_parameters():
            if id(p) in block_param_ids or id(p) in seen or not p.requires_grad:
                continue
            seen.add(id(p))
            self.repl_params.append((n, p))

        # ---- layout: [unit shards ...][replicated]
        off = 0
        for u in self.units:
            u.numel_padded = ((u.numel + self.W * ALIGN - 1) // (self.W * ALIGN)) * (self.W * ALIGN)
            u.shard_len = u.numel_padded // self.W
            u.shard_off = off
            off += u.shard_len
        self.repl_off = off
        self.repl_offsets: dict[str, tuple[int, int, torch.Size]] = {}
        for n, p in self.repl_params:
            k = p.numel()
            self.repl_offsets[n] = (off, k, p.shape)
            off += (k + ALIGN - 1) // ALIGN * ALIGN
        self.numel_padded = off

        self.flat_w = torch.zeros(off, device=self.device, dtype=self.dtype)
        self.flat_grad = torch.zeros(off, device=self.device, dtype=self.dtype)
        self._stub = torch.empty(0, device=self.device, dtype=self.dtype)

        # fill shards: build each unit's full flat once, slice my shard
        for u in self.units:
            full = torch.zeros(u.numel_padded, device=self.device, dtype=self.dtype)
            for n, p in u.params:
                o, k, _ = u.offsets[n]
                full[o : o + k].copy_(p.detach().flatten())
            self.flat_w[u.shard_off : u.shard_off + u.shard_len].copy_(
                full[self.r * u.shard_len : (self.r + 1) * u.shard_len]
            )
            for _, p in u.params:
                p.data = self._stub
            del full
        for n, p in self.repl_params:
            o, k, shp = self.repl_offsets[n]
            self.flat_w[o : o + k].copy_(p.detach().flatten())
            p.data = self.flat_w[o : o + k].view(shp)
            p.grad = self.flat_grad[o : o + k].view(shp)
        self.master32 = self.flat_w.float()

        # gather/grad staging pools (rotating; 2 covers fwd prefetch later)
        max_unit = max(u.numel_padded for u in self.units)
        self._full_pool = [
            torch.empty(max_unit, device=self.device, dtype=self.dtype) for _ in range(2)
        ]
        self._grad_pool = [
            torch.zeros(max_unit, device=self.device, dtype=self.dtype) for _ in range(2)
        ]
        self._rs_tmp = [
            torch.empty(max_unit // self.W, device=self.device, dtype=self.dtype)
            for _ in range(2)
        ]
        # comm/compute overlap state
        self._prefetch: dict[int, object] = {}   # unit idx -> in-flight gather
        self._last_gathered = -1
        self._pending_rs: list[tuple[object, _Unit, int]] = []
        self._install_hooks()

    # -------------------------------------------------------------- hooks
    def _install_hooks(self) -> None:
        for u in self.units:
            u.module.forward = self._wrap_forward(u, u.module.forward)
            for _, p in u.params:
                p.register_post_accumulate_grad_hook(self._make_grad_hook(u))

    def _wrap_forward(self, unit: _Unit, orig):
        # Works with non-reentrant activation checkpointing (enforced by the
        # Trainer): BOTH the dropping first pass and the recompute pass run
        # gather -> orig -> release. Saved weight views stay valid through
        # the unit's backward because the 2-buffer parity pool only reuses a
        # buffer two units later, after this unit's backward has completed
        # (autograd executes block backwards sequentially).
        def fwd(*args, **kw):
            self.gather_unit(unit)
            unit.pending_grads = len(unit.params)
            out = orig(*args, **kw)
            self.release_unit(unit)
            return out

        return fwd

    def _make_grad_hook(self, unit: _Unit):
        def hook(p):
            # copy this param's grad into the unit's contiguous staging
            # buffer; reduce-scatter once the unit's last grad landed
            name = unit.names_by_id[id(p)]
            o, k, _ = unit.offsets[name]
            g = self._grad_pool[unit.idx % 2]
            g[o : o + k].copy_(p.grad.flatten())
            p.grad = None
            unit.pending_grads -= 1
            if unit.pending_grads == 0:
                self.reduce_scatter_unit(unit)

        return hook

    # ---------------------------------------------------------- gather/free
    def gather_unit(self, unit: _Unit) -> None:
        """Materialize the unit; overlap: wait on a prefetched gather when
        one is in flight, and issue the next predicted unit's all-gather
        (idx+1 while ascending through forward, idx-1 through the backward
        recompute walk) so it lands under this unit's compute."""
        # unit idx-2 shares this unit's grad-pool parity: its in-flight
        # reduce-scatter must complete before this unit's backward hooks
        # overwrite the buffer (gather precedes the unit's backward, so
        # draining here orders the streams correctly)
        self._drain_rs(unit.idx % 2)
        work = self._prefetch.pop(unit.idx, None)
        if work is not None:
            work.wait()
        else:
            self._issue_gather(unit, async_op=False)
        full = self._full_pool[unit.idx % 2]
        for n, p in unit.params:
            o, k, shp = unit.offsets[n]
            p.data = full[o : o + k].view(shp)
        # predict + prefetch (buffer-parity safe: pool[i%2] is only reused
        # two gather steps later, after that unit's backward completed)
        if unit.idx > self._last_gathered:
            nxt = unit.idx + 1
        else:
            nxt = unit.idx - 1
        self._last_gathered = unit.idx
        if 0 <= nxt < len(self.units) and nxt not in self._prefetch:
            w = self._issue_gather(self.units[nxt], async_op=True)
            if w is not None:
                self._prefetch[nxt] = w

    def _issue_gather(self, unit: _Unit, async_op: bool):
        full = self._full_pool[unit.idx % 2][: unit.numel_padded]
        shard = self.flat_w[unit.shard_off : unit.shard_off + unit.shard_len]
        return _all_gather_flat(full, shard, self.group, async_op=async_op)

    def release_unit(self, unit: _Unit) -> None:
        for _, p in unit.params:
            p.data = self._stub

    def reduce_scatter_unit(self, unit: _Unit) -> None:
        """Issue the unit's grad reduce-scatter async (RCCL) and complete it
        when its tmp buffer is next needed (or at finalize_grads) — the
        collective overlaps the previous unit's recompute+backward."""
        tmp_idx = unit.idx % 2
        self._drain_rs(tmp_idx)
        g = self._grad_pool[tmp_idx][: unit.numel_padded]
        out = self._rs_tmp[tmp_idx][: unit.shard_len]
        work = _reduce_scatter_flat(out, g, self.group, async_op=True)
        if work is None:  # gloo sync fallback already filled `out`
            self._apply_rs(unit, out)
        else:
            self._pending_rs.append((work, unit, tmp_idx))

    def _apply_rs(self, unit: _Unit, out: torch.Tensor) -> None:
        # average over the worker's ranks + accumulate (grad_accum)
        self.flat_grad[unit.shard_off : unit.shard_off + unit.shard_len].add_(
            out, alpha=1.0 / self.W
        )

    def _drain_rs(self, tmp_idx: int | None = None) -> None:
        keep = []
        for work, unit, ti in self._pending_rs:
            if tmp_idx is None or ti == tmp_idx:
                work.wait()
                self._apply_rs(unit, self._rs_tmp[ti][: unit.shard_len])
            else:
                keep.append((work, unit, ti))
        self._pending_rs = keep

    # -------------------------------------------- FlatParamSpace interface
    def zero_grad(self) -> None:
        self.flat_grad.zero_()

    def finalize_grads(self) -> None:
        """Drain in-flight reduce-scatters, then average the replicated
        region across the worker."""
        self._drain_rs()
        # complete (not discard) any stale prefetch: a dangling in-flight
        # gather landing after the optimizer step would overwrite a fresh
        # gather with stale weights
        for w in self._prefetch.values():
            w.wait()
        self._prefetch.clear()
        repl = self.flat_grad[self.repl_off :]
        dist.all_reduce(repl, group=self.group)
        repl.div_(self.W)

    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        sh = self.flat_grad[: self.repl_off].float()
        repl = self.flat_grad[self.repl_off :].float()
        sq = sh.pow(2).sum()
        t = sq.clone()
        dist.all_reduce(t, group=self.group)
        total = t + repl.pow(2).sum()
        norm = total.sqrt()
        scale = (max_norm / (norm + 1e-6)).clamp(max=1.0)
        self.flat_grad.mul_(scale.to(self.flat_grad.dtype))
        return norm

    def state_dict(self) -> dict:
        return {"flat_w": self.flat_w, "master32": self.master32}

    def load_flat_(self, master32: torch.Tensor) -> None:
        self.master32.copy_(master32)
        self.flat_w.copy_(self.master32.to(self.dtype))
