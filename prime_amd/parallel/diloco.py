"""DiLoCo: inner AdamW + every-H outer Nesterov over int8-ring-averaged
pseudo-gradients.

Outer state placement is MI355X-deliberate:
  - "gpu": theta_outer/momentum resident in HBM — right for <~2B-param
    models where 3 extra fp32 copies are cheap.
  - "host": theta_outer/momentum in PINNED host DRAM, streamed through the
    outer step in chunks over PCIe with hipMemcpyAsync (north star: pinned
    hipMemcpyAsync to host DRAM). This frees 3x40 GB of HBM on the 10B
    config (the AdamW fp32 state already takes 120 GB of the 288 GB) at the
    cost of ~160 GB of PCIe traffic every H=100 steps (measured 1.75 s at
    10B with per-direction copy streams — full-duplex PCIe — overlapping
    chunk i+1's upload and i-1's drain under chunk i's compute; amortized
    ~1.6% of step time). Each chunk: {h2d theta/buf -> pseudograd -> int8
    ring all-reduce -> fused Nesterov -> d2h theta/buf}.
  - "auto": host when the model exceeds ~4B params on CUDA, else gpu.
"""
from __future__ import annotations

import torch

from .. import ops
from ..ops import QBLK
from .flat import FlatParamSpace, FusedAdamW
from .mesh import ElasticDeviceMesh

# chunks must divide by 64*QBLK so any worker count <=64 can ring them
_RING_ALIGN = 64 * QBLK
_HOST_THRESHOLD = 4_000_000_000  # params; above this "auto" offloads to host


def _mem_available_bytes() -> int | None:
    try:
        for line in open("/proc/meminfo"):
            if line.startswith("MemAvailable:"):
                return int(line.split()[1]) * 1024
    except OSError:
        pass
    return None


def _disk_backed_dir() -> str | None:
    """First candidate temp dir NOT on tmpfs/ramfs (a RAM-backed 'file'
    would defeat the low-RAM fallback)."""
    import os

    mounts = []
    try:
        for line in open("/proc/mounts"):
            parts = line.split()
            if len(parts) >= 3:
                mounts.append((parts[1], parts[2]))
    except OSError:
        return None
    mounts.sort(key=lambda m: -len(m[0]))  # longest prefix first

    def fstype(path: str) -> str:
        rp = os.path.realpath(path)
        for mp, t in mounts:
            if rp == mp or rp.startswith(mp.rstrip("/") + "/") or mp == "/":
                return t
        return "?"

    for cand in (os.environ.get("TMPDIR"), "/var/tmp", "/tmp", "."):
        if cand and os.path.isdir(cand) and os.access(cand, os.W_OK):
            if fstype(cand) not in ("tmpfs", "ramfs"):
                return cand
    return None


def host_outer_buffer(k: int, world: int) -> torch.Tensor:
    """Allocate fp32 host outer state (theta/momentum), tiered by what
    the HOST can hold — N ranks x 84 GB can exceed a node's RAM, and an
    OOM-killed rank ends a scale run:
      1. pinned pages (full PCIe rate) when RAM is plentiful
      2. pageable RAM when it fits
      3. a disk-file-backed mapping (OS pages it) as the floor"""
    import os
    import tempfile

    want_pin = os.environ.get("PRIME_AMD_OUTER_PIN", "auto")
    need = 2 * k * 4 * world  # theta+buf across all local ranks
    avail = _mem_available_bytes()
    forced_file = os.environ.get("PRIME_AMD_OUTER_FILEBACKED", "0") == "1"
    if want_pin != "0" and not forced_file:
        if want_pin == "1" or (avail is not None and need < 0.5 * avail):
            try:
                return torch.zeros(k, dtype=torch.float32, pin_memory=True)
            except RuntimeError:
                pass
    if not forced_file and (avail is None or need < 0.7 * avail):
        return torch.zeros(k, dtype=torch.float32)
    d = _disk_backed_dir()
    if d is None:
        return torch.zeros(k, dtype=torch.float32)  # best effort
    f = tempfile.NamedTemporaryFile(prefix="prime_outer_", dir=d, delete=False)
    path = f.name
    f.truncate(k * 4)
    f.close()
    t = torch.from_file(path, shared=True, size=k, dtype=torch.float32)
    os.unlink(path)  # inode lives until the mapping drops; pages to disk
    t.zero_()
    return t


class DilocoOptimizer:
    def __init__(
        self,
        flat: FlatParamSpace,
        mesh: ElasticDeviceMesh,
        inner: FusedAdamW,
        outer_lr: float = 0.7,
        outer_momentum: float = 0.9,
        H: int = 100,
        outer_device: str = "auto",
        chunk_elems: int = 256 * 1024 * 1024,  # 1 GB fp32 per streamed chunk
        elastic=None,  # ElasticWorker (leader rank only)
        elastic_mode: bool = False,  # True on ALL ranks of an elastic worker
        shard_client=None,  # ElasticShardClient (non-leader FSDP ranks)
        sharded: bool = False,  # flat is a ShardedParamSpace (FSDP)
    ):
        self.flat = flat
        self.mesh = mesh
        self.inner = inner
        self.outer_lr = outer_lr
        self.outer_momentum = outer_momentum
        self.H = H
        self.elastic = elastic
        self.elastic_mode = elastic_mode or elastic is not None
        self.shard_client = shard_client
        self.sharded = sharded
        self._view = None
        self._scale = 1.0  # this worker's outer-average pre-scale
        import threading

        self._outer_lock = threading.Lock()
        self.inner_step_count = 0
        self.outer_step_count = 0

        n = flat.numel_padded
        on_cuda = flat.device.type == "cuda"
        if outer_device == "auto":
            outer_device = "host" if (on_cuda and n > _HOST_THRESHOLD) else "gpu"
        if not on_cuda:
            outer_device = "gpu"  # plumbing path: state sits with the params
        self.outer_device = outer_device

        if outer_device == "gpu":
            self._n_comm = (n + _RING_ALIGN - 1) // _RING_ALIGN * _RING_ALIGN
            dev = flat.device
            self.theta_outer = torch.zeros(self._n_comm, device=dev, dtype=torch.float32)
            self.theta_outer[:n].copy_(flat.master32)
            self.outer_buf = torch.zeros_like(self.theta_outer)
            self.delta = torch.zeros_like(self.theta_outer)
        else:
            self.chunk = (chunk_elems // _RING_ALIGN) * _RING_ALIGN
            self._n_comm = (n + self.chunk - 1) // self.chunk * self.chunk

            def _host_buf(k: int) -> torch.Tensor:
                import os

                world = int(os.environ.get("WORLD_SIZE", 1))
                return host_outer_buffer(k, world)

            self.theta_outer = _host_buf(self._n_comm)
            # chunked D2H init (no 40 GB host temp from master32.to("cpu"))
            for c0 in range(0, n, self.chunk):
                c1 = min(c0 + self.chunk, n)
                self.theta_outer[c0:c1].copy_(flat.master32[c0:c1])
            self.outer_buf = _host_buf(self._n_comm)
            dev = flat.device
            # double-buffered chunk scratch: h2d of chunk i+1 and d2h of
            # chunk i-1 run on a dedicated copy stream under chunk i's
            # pseudograd/ring/Nesterov (hipMemcpyAsync + pinned host)
            self._g_theta = [torch.zeros(self.chunk, device=dev, dtype=torch.float32) for _ in range(2)]
            self._g_buf = [torch.zeros(self.chunk, device=dev, dtype=torch.float32) for _ in range(2)]
            self._g_delta = torch.zeros(self.chunk, device=dev, dtype=torch.float32)
            # separate streams per direction: PCIe Gen5 is full duplex and
            # the SDMA engines are per-direction — one shared stream would
            # serialize h2d and d2h at 2x the wall time
            self._h2d_stream = torch.cuda.Stream(device=dev)
            self._d2h_stream = torch.cuda.Stream(device=dev)
            self._h2d_ev = [torch.cuda.Event(), torch.cuda.Event()]
            self._cmp_ev = [torch.cuda.Event(), torch.cuda.Event()]
            self._d2h_ev = [torch.cuda.Event(), torch.cuda.Event()]

    # ----------------------------------------------------------------- step
    def step(self, gscale=None) -> bool:
        """One inner step (grads already populated + locally averaged).
        Returns True when an outer sync happened."""
        self.inner.step(gscale)
        self.inner_step_count += 1
        if self.inner_step_count % self.H != 0:
            return False
        self.outer_step()
        return True

    def outer_step(self) -> None:
        if self.elastic_mode and not self._elastic_boundary():
            return  # evicted this boundary: all ranks skip together
        with self._outer_lock:
            if self.outer_device == "gpu":
                self._outer_step_resident()
            else:
                self._outer_step_streamed()
        ops.invalidate_wt_cache()  # nesterov_outer rewrote flat_w in place
        self.outer_step_count += 1

    def _elastic_boundary(self) -> bool:
        """Outer-boundary membership agreement for elastic workers.

        The leader runs the TCPStore sync (possibly getting evicted and
        rejoining); every local rank then learns the SAME branch via an
        object broadcast — without it, non-leader ranks of a multi-rank
        worker would walk into the chunk loop's collectives alone and
        deadlock. Sharded (FSDP) ranks additionally build their own
        shard-aligned cross-worker gloo group. Returns False when this
        boundary must be skipped (just evicted)."""
        from .elastic import ElasticView, EvictedError

        info = None
        if self.elastic is not None:
            try:
                steps_this_round = self.inner_step_count - getattr(
                    self, "_last_boundary_step", 0)
                self._view = self.elastic.sync(
                    contribution=min(1.0, steps_this_round / max(1, self.H)))
                self._last_boundary_step = self.inner_step_count
                v = self._view
                info = {"evicted": False, "bootstrapped": False,
                        "epoch": v.epoch, "world": v.world,
                        "my_index": v.my_index, "scale": v.my_scale(),
                        "wid": self.elastic.wid}
            except EvictedError:
                # stalled past the heartbeat timeout and got evicted:
                # re-register, adopt a live peer's outer state, skip this
                # boundary, and contribute again from the next one
                self.elastic.rejoin()
                payload = self.elastic.bootstrap_from_peer()
                boot = payload is not None
                if boot:
                    self.load_bootstrap(payload)
                self.rejoined = getattr(self, "rejoined", 0) + 1
                self._view = None
                info = {"evicted": True, "bootstrapped": boot,
                        "wid": self.elastic.wid}
        lg = self.mesh.local_group
        src = self.mesh.worker_id * self.mesh.cfg.worker_size
        if lg is not None:
            import torch.distributed as dist

            obj = [info]
            dist.broadcast_object_list(obj, src=src, group=lg)
            info = obj[0]
            if self.shard_client is not None and info and info.get("wid"):
                self.shard_client.set_wid(info["wid"])
        if info is None:  # single-rank non-leader cannot happen; be safe
            return True
        self._scale = float(info.get("scale", 1.0))
        if info["evicted"]:
            if info["bootstrapped"] and lg is not None:
                if self.sharded:
                    if self.shard_client is not None:
                        payload = self.shard_client.bootstrap_shard()
                        if payload is not None:
                            self.load_bootstrap(payload)
                else:
                    self._bcast_outer_state(src)
            self._last_boundary_step = self.inner_step_count
            return False
        # sharded non-leader ranks: build this epoch's shard-aligned group
        if (self.sharded and self.shard_client is not None
                and info["world"] > 1):
            pg = self.shard_client.build_pg(info["epoch"], info["my_index"],
                                            info["world"])
            self._view = ElasticView(info["epoch"], [""] * info["world"],
                                     info["my_index"], pg, None)
        elif self.elastic is None:
            self._view = (ElasticView(info["epoch"], [""], 0, None, None)
                          if info["world"] <= 1 else self._view)
        return True

    def _bcast_outer_state(self, src: int) -> None:
        """Replicate the leader's (just-bootstrapped) outer state to the
        other ranks of a DP elastic worker, so a rejoin cannot leave leader
        and non-leader parameters diverged."""
        import torch.distributed as dist

        lg = self.mesh.local_group
        dev = self.flat.device
        n = self.flat.numel_padded
        counts = torch.tensor([self.inner_step_count, self.outer_step_count],
                              dtype=torch.int64, device=dev)
        dist.broadcast(counts, src=src, group=lg)
        self.inner_step_count = int(counts[0])
        self.outer_step_count = int(counts[1])
        self._last_boundary_step = self.inner_step_count
        if self.outer_device == "gpu":
            dist.broadcast(self.theta_outer, src=src, group=lg)
            dist.broadcast(self.outer_buf, src=src, group=lg)
        else:
            # host-offloaded state: chunk through the device scratch
            scratch = self._g_theta[0]
            for c0 in range(0, self._n_comm, self.chunk):
                c1 = min(c0 + self.chunk, self._n_comm)
                k = c1 - c0
                scratch[:k].copy_(self.theta_outer[c0:c1])
                dist.broadcast(scratch[:k], src=src, group=lg)
                self.theta_outer[c0:c1].copy_(scratch[:k])
                scratch[:k].copy_(self.outer_buf[c0:c1])
                dist.broadcast(scratch[:k], src=src, group=lg)
                self.outer_buf[c0:c1].copy_(scratch[:k])
        th = self.theta_outer[:n]
        self.flat.load_flat_(th.to(self.flat.master32.device)
                             if th.device != self.flat.master32.device else th)

    def init_bootstrap(self) -> bool:
        """Startup live recovery: the leader pulls a live peer's outer
        state; a multi-rank worker then syncs so every rank starts
        consistent (FSDP ranks fetch their own shard from the peer's
        matching rank). Returns True when state was adopted."""
        info = None
        if self.elastic is not None:
            payload = self.elastic.bootstrap_from_peer()
            boot = payload is not None
            if boot:
                self.load_bootstrap(payload)
            info = {"bootstrapped": boot, "wid": self.elastic.wid}
        lg = self.mesh.local_group
        if self.elastic_mode and lg is not None:
            import torch.distributed as dist

            src = self.mesh.worker_id * self.mesh.cfg.worker_size
            obj = [info]
            dist.broadcast_object_list(obj, src=src, group=lg)
            info = obj[0]
            if self.shard_client is not None and info and info.get("wid"):
                self.shard_client.set_wid(info["wid"])
            if info and info["bootstrapped"]:
                if self.sharded:
                    if self.shard_client is not None:
                        payload = self.shard_client.bootstrap_shard()
                        if payload is not None:
                            self.load_bootstrap(payload)
                else:
                    self._bcast_outer_state(src)
        return bool(info and info["bootstrapped"])

    def _allreduce(self, delta_chunk: torch.Tensor) -> None:
        """Average a (padded) delta chunk across workers.

        Elastic mode: the worker's leader runs the int8 gloo ring over the
        current epoch's cross-worker group, then the result fans out to the
        worker's other ranks over the local (RCCL) group. Static mode: the
        mesh's pre-built outer group (int8 ring over RCCL P2P)."""
        if self.elastic_mode:
            # who rings cross-worker: the leader always; with FSDP every
            # rank rings its OWN shard over its shard-aligned group
            rings = self.elastic is not None or (
                self.sharded and self.shard_client is not None)
            v = self._view
            if rings and v is not None and v.world > 1:
                from .elastic import ring_allreduce_int8_pg

                if self._scale != 1.0:
                    delta_chunk.mul_(self._scale)  # weighted outer average
                if delta_chunk.is_cuda:
                    # reusable pinned bounce buffer: the gloo ring reads
                    # host memory; a fresh pageable .to("cpu") per chunk
                    # both allocates and slows the D2H path
                    buf = getattr(self, "_elastic_host_buf", None)
                    if buf is None or buf.numel() < delta_chunk.numel():
                        try:
                            buf = torch.empty(delta_chunk.numel(),
                                              dtype=delta_chunk.dtype,
                                              pin_memory=True)
                        except RuntimeError:
                            buf = torch.empty(delta_chunk.numel(),
                                              dtype=delta_chunk.dtype)
                        self._elastic_host_buf = buf
                    host = buf[: delta_chunk.numel()]
                    host.copy_(delta_chunk)  # sync D2H (ring needs it now)
                    ring_allreduce_int8_pg(host, v.pg, v.my_index, v.world)
                    delta_chunk.copy_(host, non_blocking=True)
                else:
                    ring_allreduce_int8_pg(delta_chunk, v.pg, v.my_index, v.world)
            if not self.sharded and self.mesh.local_group is not None:
                import torch.distributed as dist

                src = self.mesh.worker_id * self.mesh.cfg.worker_size
                dist.broadcast(delta_chunk, src=src, group=self.mesh.local_group)
        else:
            self.mesh.outer_allreduce_avg(delta_chunk)

    def _outer_step_resident(self) -> None:
        f = self.flat
        n = f.numel_padded
        theta, delta = self.theta_outer, self.delta
        if f.device.type == "cuda":
            ops.pseudograd(theta[:n], f.master32, delta[:n])
            if n < delta.numel():
                delta[n:].zero_()
            self._allreduce(delta)
            ops.nesterov_outer(
                theta[:n], f.master32, f.flat_w, self.outer_buf[:n], delta[:n],
                lr=self.outer_lr, mu=self.outer_momentum,
            )
        else:
            delta[:n] = theta[:n] - f.master32
            if n < delta.numel():
                delta[n:].zero_()
            self._allreduce(delta)
            buf = self.outer_buf[:n]
            buf.mul_(self.outer_momentum).add_(delta[:n])
            theta[:n].add_(delta[:n] + self.outer_momentum * buf, alpha=-self.outer_lr)
            f.load_flat_(theta[:n])

    def _outer_step_streamed(self) -> None:
        """Chunked outer step with pinned-host theta/momentum, pipelined:
        the copy stream uploads chunk i+1 and drains chunk i-1 while the
        compute stream runs chunk i's {pseudograd, ring, Nesterov}."""
        f = self.flat
        n = f.numel_padded
        gd = self._g_delta
        chunks = list(range(0, self._n_comm, self.chunk))
        up, down = self._h2d_stream, self._d2h_stream
        main = torch.cuda.current_stream(f.device)

        def h2d(i: int) -> None:
            c0 = chunks[i]
            c1 = min(c0 + self.chunk, self._n_comm)
            b = i % 2
            with torch.cuda.stream(up):
                if i >= 2:  # buffer reuse: chunk i-2's d2h must have drained
                    up.wait_event(self._d2h_ev[b])
                self._g_theta[b][: c1 - c0].copy_(self.theta_outer[c0:c1], non_blocking=True)
                self._g_buf[b][: c1 - c0].copy_(self.outer_buf[c0:c1], non_blocking=True)
                self._h2d_ev[b].record(up)

        up.wait_stream(main)  # master32 writes (inner steps) visible
        h2d(0)
        for i, c0 in enumerate(chunks):
            c1 = min(c0 + self.chunk, self._n_comm)
            k = c1 - c0
            live = max(0, min(n, c1) - c0)  # elements backed by real params
            b = i % 2
            gt, gb = self._g_theta[b], self._g_buf[b]
            main.wait_event(self._h2d_ev[b])
            if live > 0:
                ops.pseudograd(gt[:live], f.master32[c0:c0 + live], gd[:live])
            if live < k:
                gd[live:k].zero_()
            if i + 1 < len(chunks):
                # buffer (i+1)%2 is free: chunk i-1's d2h was enqueued on the
                # copy stream before this h2d (stream order preserves it)
                h2d(i + 1)
            self._allreduce(gd[:k])
            if live > 0:
                ops.nesterov_outer(
                    gt[:live], f.master32[c0:c0 + live],
                    f.flat_w[c0:c0 + live], gb[:live], gd[:live],
                    lr=self.outer_lr, mu=self.outer_momentum,
                )
            self._cmp_ev[b].record(main)
            with torch.cuda.stream(down):
                down.wait_event(self._cmp_ev[b])
                self.theta_outer[c0:c1].copy_(gt[:k], non_blocking=True)
                self.outer_buf[c0:c1].copy_(gb[:k], non_blocking=True)
                self._d2h_ev[b].record(down)
        torch.cuda.synchronize()

    # ------------------------------------------------------------ ckpt
    def live_state(self) -> dict:
        """Consistent CPU snapshot for live peer recovery (taken under the
        outer-step lock so a joiner never sees a half-applied outer step)."""
        with self._outer_lock:
            return {
                "theta_outer": self.theta_outer.detach().to("cpu").clone(),
                "outer_buf": self.outer_buf.detach().to("cpu").clone(),
                "inner_step": self.inner_step_count,
                "outer_step": self.outer_step_count,
            }

    def load_bootstrap(self, payload: dict) -> None:
        """Adopt a peer's live state (joining worker)."""
        n = self.flat.numel_padded
        theta = payload["theta_outer"]
        with self._outer_lock:
            k = min(theta.numel(), self.theta_outer.numel())
            self.theta_outer[:k].copy_(theta[:k].to(self.theta_outer.device))
            self.outer_buf[:k].copy_(payload["outer_buf"][:k].to(self.outer_buf.device))
            self.inner_step_count = int(payload["inner_step"])
            self.outer_step_count = int(payload["outer_step"])
            # the adopted step count is the peer's, not steps WE trained:
            # without resetting the boundary marker the first outer sync
            # would clamp this joiner's contribution to 1.0
            self._last_boundary_step = self.inner_step_count
            th = self.theta_outer[:n]
            self.flat.load_flat_(th.to(self.flat.master32.device)
                                 if th.device != self.flat.master32.device else th)

    def state_dict(self) -> dict:
        return {
            "theta_outer": self.theta_outer,
            "outer_buf": self.outer_buf,
            "inner_step": self.inner_step_count,
            "outer_step": self.outer_step_count,
            "inner": self.inner.state_dict(),
        }

    def load_state_dict(self, sd: dict) -> None:
        self.theta_outer.copy_(sd["theta_outer"])
        self.outer_buf.copy_(sd["outer_buf"])
        self.inner_step_count = int(sd["inner_step"])
        self.outer_step_count = int(sd["outer_step"])
        self.inner.load_state_dict(sd["inner"])
