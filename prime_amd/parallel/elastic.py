"""Elastic cross-worker layer: dynamic DiLoCo worker membership over a
TCPStore, with heartbeats, dead-peer eviction, live peer checkpoint
recovery, and per-epoch gloo group rebuild (SURVEY.md §B1.3 — the
ElasticDeviceMesh fault-tolerance semantics rebuilt MI355X-first: RCCL
stays intra-worker; the elastic cross-worker fabric is TCP/gloo).

Protocol (per outer-step boundary; "arbiter" = lowest-id live leader):
  1. every worker leader posts   ready/<n>/<wid>
  2. the arbiter waits until every LIVE member (fresh heartbeat, not
     leaving) has posted, evicting stale members as it waits, then
     atomically (compare_set) publishes   view/<n> = sorted member list
  3. everyone builds a fresh ProcessGroupGloo on PrefixStore(pg/<n>/)
     with rank = index in the view and runs the int8 ring over it
  4. arbiter failover: if view/<n> hasn't appeared after a grace period,
     the next-lowest live leader attempts the write (first writer wins)

Joiners: register, pull {theta_outer, outer_buf, outer_step} from any
live peer's checkpoint server (a TCP thread each leader runs), train a
full inner round, then enter the protocol at their first boundary — no
partial-round dilution of the outer average.
"""
from __future__ import annotations

import io
import json
import os
import socket
import struct
import threading
import time
from dataclasses import dataclass, field
from datetime import timedelta
from typing import Callable, Optional

import torch
from torch.distributed import PrefixStore, ProcessGroupGloo, TCPStore


def _now() -> float:
    return time.monotonic()


@dataclass
class ElasticView:
    epoch: int
    members: list[str]
    my_index: int
    pg: Optional[ProcessGroupGloo]
    weights: list[float] | None = None  # per-member contribution weights

    @property
    def world(self) -> int:
        return len(self.members)

    def my_scale(self) -> float:
        """Pre-scale for this worker's delta so the ring's plain average
        yields the WEIGHTED mean: delta_i * (w_i * W / sum(w))."""
        if not self.weights:
            return 1.0
        tot = sum(self.weights)
        if tot <= 0:
            return 1.0
        return self.weights[self.my_index] * len(self.weights) / tot


@dataclass
class _PeerTrack:
    last_val: bytes = b""
    t_changed: float = field(default_factory=_now)


class EvictedError(RuntimeError):
    """This worker was evicted (e.g. long stall); caller must re-join."""


class ElasticWorker:
    """One DiLoCo worker's leader-side handle on the elastic fabric."""

    def __init__(
        self,
        addr: str = None,
        port: int = None,
        worker_name: str | None = None,
        host_store: bool = False,
        heartbeat_interval: float = 2.0,
        heartbeat_timeout: float = 15.0,
        ckpt_provider: Callable[[], dict] | None = None,
        host_ip: str | None = None,
    ):
        self.addr = addr or os.environ.get("PRIME_GLOBAL_ADDR", "127.0.0.1")
        self.port = int(port or os.environ.get("PRIME_GLOBAL_PORT", 29777))
        self.hb_interval = heartbeat_interval
        self.hb_timeout = heartbeat_timeout
        self.host_ip = host_ip or os.environ.get("PRIME_HOST_IP", "127.0.0.1")
        self.store = TCPStore(
            self.addr, self.port, is_master=host_store, wait_for_workers=False,
            timeout=timedelta(seconds=60),
        )
        # unique ordered worker id: join sequence number + name
        seq = int(self.store.add("join_seq", 1))
        self.join_seq = seq  # worker-unique (data sharding uses it)
        self.wid = f"{seq:06d}-{worker_name or os.getpid()}"
        self.epoch = int(self._get_str("epoch", "0"))
        self._hb_seq = 0
        self._peer_tracks: dict[str, _PeerTrack] = {}
        self._stop = threading.Event()
        self._ckpt_provider = ckpt_provider
        self._ckpt_srv: Optional[socket.socket] = None
        self._register()
        self._hb_thread = threading.Thread(target=self._hb_loop, daemon=True)
        self._hb_thread.start()
        if ckpt_provider is not None:
            self._start_ckpt_server()

    # ---------------------------------------------------------- store utils
    def _get_str(self, key: str, default: str) -> str:
        if self.store.check([key]):
            return self.store.get(key).decode()
        return default

    def _register(self) -> None:
        self.store.set(f"members/{self.wid}", "1")
        self.store.set(f"hb/{self.wid}", "0")

    def _index_cas(self, transform) -> list[str]:
        """Apply a pure transform to the member_index mirror with a
        compare_set retry loop — a plain get/set read-modify-write lets two
        concurrent joiners each read the old index and overwrite the
        other's registration (this key is the membership source of truth)."""
        for _ in range(64):
            exists = self.store.check(["member_index"])
            cur = self.store.get("member_index").decode() if exists else "[]"
            new = json.dumps(transform(json.loads(cur)))
            if new == cur:
                return json.loads(cur)
            got = self.store.compare_set("member_index", cur if exists else "", new)
            if got.decode() == new:
                return json.loads(new)
        raise RuntimeError("member_index compare_set did not converge")

    def members(self) -> list[str]:
        """Sorted registry (join order == lexicographic by construction)."""
        # TCPStore has no key listing; keep a mirror index
        try:
            idx = self._index_cas(lambda i: sorted(set(i) | {self.wid}))
        except Exception:  # noqa: BLE001 — store host gone: reconnect once
            if not self.reconnect():
                raise
            idx = self._index_cas(lambda i: sorted(set(i) | {self.wid}))
        live = [w for w in idx if self.store.check([f"members/{w}"])]
        if live != idx:
            dead = set(idx) - set(live)
            live = self._index_cas(lambda i: sorted(set(i) - dead))
            live = [w for w in live if self.store.check([f"members/{w}"])]
        return live

    # ----------------------------------------------------------- reconnect
    def reconnect(self, attempts: int = 5, backoff: float = 2.0) -> bool:
        """The TCPStore host went away (e.g. a standalone `prime-amd
        store` was restarted): build a fresh client against the same
        address and re-register under a new wid. The restarted registry
        starts empty; membership reconverges at the next boundary as
        every surviving worker re-registers."""
        import time as _time

        for i in range(attempts):
            try:
                store = TCPStore(self.addr, self.port, is_master=False,
                                 wait_for_workers=False,
                                 timeout=timedelta(seconds=10))
                seq = int(store.add("join_seq", 1))
                self.store = store
                self.join_seq = seq
                self.wid = f"{seq:06d}-re-{self.wid.split('-', 1)[1]}"
                self._peer_tracks.clear()
                self.epoch = int(self._get_str("epoch", "0"))
                self._register()
                if self._ckpt_srv is not None:
                    host, port = self._ckpt_srv.getsockname()
                    self.store.set(f"ckptsrv/{self.wid}", f"{host}:{port}")
                    self.store.set(f"ckptsrv/{self.wid}/r0", f"{host}:{port}")
                return True
            except Exception:  # noqa: BLE001 — host still down; back off
                _time.sleep(backoff * (i + 1))
        return False

    # ------------------------------------------------------------ heartbeat
    def _hb_loop(self) -> None:
        while not self._stop.wait(self.hb_interval):
            self._hb_seq += 1
            try:
                self.store.set(f"hb/{self.wid}", str(self._hb_seq))
            except Exception:  # noqa: BLE001 — store host gone; the sync
                # path owns reconnection; just keep trying the (possibly
                # replaced) store handle
                continue

    def _is_stale(self, wid: str) -> bool:
        try:
            val = self.store.get(f"hb/{wid}") if self.store.check([f"hb/{wid}"]) else b""
        except Exception:  # noqa: BLE001
            val = b""
        tr = self._peer_tracks.setdefault(wid, _PeerTrack(val, _now()))
        if val != tr.last_val:
            tr.last_val = val
            tr.t_changed = _now()
        return (_now() - tr.t_changed) > self.hb_timeout

    def _is_leaving(self, wid: str) -> bool:
        return self.store.check([f"leaving/{wid}"])

    def _evict(self, wid: str) -> None:
        for k in (f"members/{wid}", f"hb/{wid}", f"ckptsrv/{wid}"):
            try:
                self.store.delete_key(k)
            except Exception:  # noqa: BLE001
                pass

    # -------------------------------------------------------- boundary sync
    def sync(self, build_pg: bool = True, contribution: float = 1.0) -> ElasticView:
        """Called by every leader at an outer-step boundary; returns the
        agreed membership view and a fresh gloo group over it.
        `contribution` (e.g. inner steps completed this round / H) becomes
        this worker's weight in the outer average — a freshly-joined
        worker with a partial round contributes proportionally."""
        while True:
            n = self.epoch + 1
            try:
                self.store.set(f"ready/{n}/{self.wid}",
                               f"{max(0.0, contribution):.6f}")
                view_key = f"view/{n}"
                t0 = _now()
                members: list[str] = []
                while True:
                    if self.store.check([view_key]):
                        members = json.loads(self.store.get(view_key).decode())
                        break
                    if self._try_arbiter(n, t0):
                        members = json.loads(self.store.get(view_key).decode())
                        break
                    time.sleep(0.05)
            except Exception as e:  # noqa: BLE001
                # store host gone (possibly mid-wait): reconnect to a
                # restarted `prime-amd store` and redo the boundary from
                # the new registry's epoch
                if self.reconnect():
                    continue
                raise RuntimeError(
                    "elastic store unreachable and reconnect failed — "
                    "run a standalone registry (`prime-amd store`) so "
                    "worker churn cannot take the store down, then "
                    f"restart workers against it. ({e})"
                ) from e
            self.epoch = n
            self.store.set("epoch", str(n))
            if self.wid not in members:
                if not self.store.check([f"members/{self.wid}"]):
                    raise EvictedError(f"worker {self.wid} was evicted at epoch {n}")
                continue  # missed this round's cut; try next boundary
            pg = None
            if build_pg and len(members) > 1:
                # shard-aligned prefix: rank r of every worker rings over
                # pg/<epoch>/r<r>/ — the leader is shard 0; non-leader ranks
                # of an FSDP worker build theirs via ElasticShardClient
                pg = ProcessGroupGloo(
                    PrefixStore(f"pg/{n}/r0/", self.store),
                    members.index(self.wid), len(members),
                )
            weights = []
            for w in members:
                try:
                    weights.append(float(self.store.get(f"ready/{n}/{w}").decode()))
                except Exception:  # noqa: BLE001 — legacy "1" or missing
                    weights.append(1.0)
            return ElasticView(n, members, members.index(self.wid), pg, weights)

    def _try_arbiter(self, n: int, t_wait_start: float) -> bool:
        """Attempt the arbiter role; returns True once view/<n> exists."""
        live = [
            w for w in self.members()
            if not self._is_leaving(w) and (w == self.wid or not self._is_stale(w))
        ]
        if not live:
            live = [self.wid]
        # am I a candidate yet? lowest waits 0, next waits grace, ...
        grace = 2.0 * self.hb_timeout
        try:
            rank_among = live.index(self.wid)
        except ValueError:
            return False
        if (_now() - t_wait_start) < grace * rank_among:
            return False
        ready = [w for w in live if self.store.check([f"ready/{n}/{w}"])]
        stale = [
            w for w in self.members()
            if w != self.wid and self._is_stale(w) and not self.store.check([f"ready/{n}/{w}"])
        ]
        for w in stale:
            self._evict(w)
        if set(ready) >= set(live) - set(stale):
            payload = json.dumps(sorted(set(ready) - set(stale)))
            self.store.compare_set(f"view/{n}", "", payload)
            return True
        return False

    # ------------------------------------------------- live ckpt recovery
    def _start_ckpt_server(self) -> None:
        srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind((self.host_ip, 0))
        srv.listen(4)
        self._ckpt_srv = srv
        host, port = srv.getsockname()
        self.store.set(f"ckptsrv/{self.wid}", f"{host}:{port}")
        # shard-aligned alias: the leader serves shard 0 of an FSDP worker
        self.store.set(f"ckptsrv/{self.wid}/r0", f"{host}:{port}")

        def _serve():
            while not self._stop.is_set():
                try:
                    srv.settimeout(1.0)
                    conn, _ = srv.accept()
                except socket.timeout:
                    continue
                except OSError:
                    return
                try:
                    payload = self._ckpt_provider()
                    buf = io.BytesIO()
                    torch.save(payload, buf)
                    raw = buf.getvalue()
                    conn.sendall(struct.pack("<Q", len(raw)))
                    conn.sendall(raw)
                except Exception:  # noqa: BLE001 — keep serving other peers
                    pass
                finally:
                    conn.close()

        threading.Thread(target=_serve, daemon=True).start()

    def bootstrap_from_peer(self) -> dict | None:
        """Fetch live state from any other member's checkpoint server.
        Returns None when this worker is alone (cold start)."""
        for wid in self.members():
            if wid == self.wid or self._is_stale(wid) or self._is_leaving(wid):
                continue
            key = f"ckptsrv/{wid}"
            if not self.store.check([key]):
                continue
            host, port = self.store.get(key).decode().rsplit(":", 1)
            try:
                with socket.create_connection((host, int(port)), timeout=30) as c:
                    hdr = _recv_exact(c, 8)
                    (length,) = struct.unpack("<Q", hdr)
                    raw = _recv_exact(c, length)
                return torch.load(io.BytesIO(raw), map_location="cpu", weights_only=False)
            except OSError:
                continue
        return None

    # -------------------------------------------------------------- rejoin
    def rejoin(self) -> None:
        """Re-register after an eviction (e.g. a long stall): new worker id,
        fresh heartbeat identity, re-advertised checkpoint server. The
        caller should re-bootstrap state from a peer before contributing."""
        seq = int(self.store.add("join_seq", 1))
        self.join_seq = seq
        old = self.wid
        self.wid = f"{seq:06d}-rejoin-{old.split('-', 1)[1]}"
        self._peer_tracks.clear()
        self.epoch = int(self._get_str("epoch", "0"))
        self._register()
        if self._ckpt_srv is not None:
            host, port = self._ckpt_srv.getsockname()
            self.store.set(f"ckptsrv/{self.wid}", f"{host}:{port}")
            self.store.set(f"ckptsrv/{self.wid}/r0", f"{host}:{port}")

    # --------------------------------------------------------------- leave
    def close(self, leaving: bool = True) -> None:
        if leaving:
            try:
                self.store.set(f"leaving/{self.wid}", "1")
                self._evict(self.wid)
            except Exception:  # noqa: BLE001
                pass
        self._stop.set()
        if self._ckpt_srv is not None:
            try:
                self._ckpt_srv.close()
            except OSError:
                pass


class ElasticShardClient:
    """A non-leader rank's handle on the elastic fabric (FSDP workers).

    The leader (ElasticWorker) owns membership; this client lets shard
    rank r of the worker (a) build the shard-aligned cross-worker gloo
    group pg/<epoch>/r<r>/ once the leader has broadcast the agreed view,
    (b) serve its OWN shard for live peer recovery under
    ckptsrv/<leader-wid>/r<r>, and (c) bootstrap its shard from the
    matching rank of any live peer worker (peers must share worker_size)."""

    def __init__(self, shard_rank: int, addr: str = None, port: int = None,
                 ckpt_provider: Callable[[], dict] | None = None,
                 host_ip: str | None = None):
        self.addr = addr or os.environ.get("PRIME_GLOBAL_ADDR", "127.0.0.1")
        self.port = int(port or os.environ.get("PRIME_GLOBAL_PORT", 29777))
        self.host_ip = host_ip or os.environ.get("PRIME_HOST_IP", "127.0.0.1")
        self.shard_rank = shard_rank
        self.wid: str | None = None  # leader's wid, set via set_wid()
        self.store = TCPStore(self.addr, self.port, is_master=False,
                              wait_for_workers=False,
                              timeout=timedelta(seconds=60))
        self._stop = threading.Event()
        self._ckpt_provider = ckpt_provider
        self._ckpt_srv: Optional[socket.socket] = None
        if ckpt_provider is not None:
            self._start_ckpt_server()

    def set_wid(self, wid: str) -> None:
        """(Re)publish this shard's checkpoint server under the leader's
        current wid (changes after an eviction+rejoin)."""
        if wid == self.wid:
            return
        self.wid = wid
        if self._ckpt_srv is not None:
            host, port = self._ckpt_srv.getsockname()
            self.store.set(f"ckptsrv/{wid}/r{self.shard_rank}", f"{host}:{port}")

    def build_pg(self, epoch: int, my_index: int, world: int) -> ProcessGroupGloo:
        return ProcessGroupGloo(
            PrefixStore(f"pg/{epoch}/r{self.shard_rank}/", self.store),
            my_index, world,
        )

    def _start_ckpt_server(self) -> None:
        srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind((self.host_ip, 0))
        srv.listen(4)
        self._ckpt_srv = srv  # store key published by set_wid()

        def _serve():
            while not self._stop.is_set():
                try:
                    srv.settimeout(1.0)
                    conn, _ = srv.accept()
                except socket.timeout:
                    continue
                except OSError:
                    return
                try:
                    payload = self._ckpt_provider()
                    buf = io.BytesIO()
                    torch.save(payload, buf)
                    raw = buf.getvalue()
                    conn.sendall(struct.pack("<Q", len(raw)))
                    conn.sendall(raw)
                except Exception:  # noqa: BLE001 — keep serving other peers
                    pass
                finally:
                    conn.close()

        threading.Thread(target=_serve, daemon=True).start()

    def bootstrap_shard(self) -> dict | None:
        """Fetch this shard's live state from the matching rank of any
        live peer worker."""
        try:
            exists = self.store.check(["member_index"])
            idx = json.loads(self.store.get("member_index").decode()) if exists else []
        except Exception:  # noqa: BLE001
            return None
        for wid in idx:
            # skip ourselves and evicted members (their shard ckptsrv keys
            # outlive _evict, which only knows the leader-level key)
            if wid == self.wid or not self.store.check([f"members/{wid}"]):
                continue
            key = f"ckptsrv/{wid}/r{self.shard_rank}"
            if not self.store.check([key]):
                continue
            host, port = self.store.get(key).decode().rsplit(":", 1)
            try:
                with socket.create_connection((host, int(port)), timeout=30) as c:
                    hdr = _recv_exact(c, 8)
                    (length,) = struct.unpack("<Q", hdr)
                    raw = _recv_exact(c, length)
                return torch.load(io.BytesIO(raw), map_location="cpu", weights_only=False)
            except OSError:
                continue
        return None

    def close(self) -> None:
        self._stop.set()
        if self._ckpt_srv is not None:
            try:
                self._ckpt_srv.close()
            except OSError:
                pass


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    chunks = []
    got = 0
    while got < n:
        c = sock.recv(min(1 << 20, n - got))
        if not c:
            raise OSError("peer closed during checkpoint transfer")
        chunks.append(c)
        got += len(c)
    return b"".join(chunks)


# ---------------------------------------------------- ring over a raw pg
def ring_allreduce_int8_pg(delta: torch.Tensor, pg, rank: int, world: int,
                           average: bool = True) -> None:
    """int8 ring all-reduce over a raw ProcessGroup (pg.send/pg.recv,
    group-relative ranks) — the elastic path's transport. Same algorithm
    as prime_amd.parallel.ring (which needs a registered group)."""
    from .. import ops
    from ..ops import QBLK
    from .ring import _dequant_add, _pad_run, _quant

    if world == 1:
        return
    n = delta.numel()
    if _pad_run(delta, world * QBLK,
                lambda b: ring_allreduce_int8_pg(b, pg, rank, world, average)):
        return
    part = n // world
    parts = [delta[i * part : (i + 1) * part] for i in range(world)]
    nxt, prv = (rank + 1) % world, (rank - 1) % world
    recv_q = torch.empty(part, dtype=torch.int8, device=delta.device)
    recv_s = torch.empty(part // QBLK, dtype=torch.float32, device=delta.device)
    for step in range(world - 1):
        send_idx = (rank - step) % world
        recv_idx = (rank - step - 1) % world
        q, s = _quant(parts[send_idx])
        q, s = q.contiguous(), s.contiguous()
        works = [pg.send([q], nxt, 0), pg.send([s], nxt, 1),
                 pg.recv([recv_q], prv, 0), pg.recv([recv_s], prv, 1)]
        for w in works:
            w.wait()
        _dequant_add(recv_q, recv_s, parts[recv_idx], accumulate=True)
    own = (rank + 1) % world
    if average:
        parts[own].div_(world)
    send_q, send_s = _quant(parts[own])
    send_q, send_s = send_q.contiguous(), send_s.contiguous()
    _dequant_add(send_q, send_s, parts[own], accumulate=False)
    for step in range(world - 1):
        recv_idx = (rank - step) % world
        works = [pg.send([send_q], nxt, 0), pg.send([send_s], nxt, 1),
                 pg.recv([recv_q], prv, 0), pg.recv([recv_s], prv, 1)]
        for w in works:
            w.wait()
        _dequant_add(recv_q, recv_s, parts[recv_idx], accumulate=False)
        send_q, recv_q = recv_q.clone(), send_q
        send_s, recv_s = recv_s.clone(), send_s
