"""Elastic multi-rank (FSDP) workers on CPU/gloo: shard-aligned
cross-worker rings, the leader->local eviction broadcast (regression for
the non-leader deadlock), per-shard live recovery, and concurrent-join
stress on the membership index."""
import multiprocessing as mp
import os
import time

import pytest
import torch

from tests.conftest import free_port
from tests.test_elastic import _entry


def _spawn(fns_envs, timeout=300):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = []
    for i, (fn, env, args) in enumerate(fns_envs):
        p = ctx.Process(target=_entry, args=(fn, i, env, args, q))
        p.start()
        procs.append(p)
    results, errs = {}, []
    for _ in range(len(fns_envs)):
        kind, i, payload = q.get(timeout=timeout)
        if kind == "ok":
            results[i] = payload
        else:
            errs.append((i, payload))
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if errs:
        raise AssertionError(f"elastic fsdp workers failed: {errs}")
    return results


# ------------------------------------------------- 2 workers x FSDP(2)
def _fsdp_trainer(gport, host, steps, run_tag):
    from prime_amd.train import Trainer
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )

    cfg = TrainConfig(
        run_name=f"elf_{run_tag}",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=64,
                          activation_checkpointing=True),
        diloco=DilocoConfig(H=2),
        parallel=ParallelConfig(elastic=True, fsdp=True, worker_size=2,
                                heartbeat_interval=0.3,
                                heartbeat_timeout=10.0),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/elfsdp_{run_tag}")
    import torch.distributed as dist

    # both workers registered before the first boundary
    if tr.elastic is not None:
        while len(tr.elastic.members()) < 2:
            time.sleep(0.05)
    dist.barrier()
    res = tr.run()
    shard_head = tr.flat.flat_w[:16].tolist()
    outer_head = tr.diloco.theta_outer[:16].tolist()
    rank = tr.mesh.worker_rank
    tr.close()
    return {"rank": rank, "shard": shard_head, "outer": outer_head,
            "outer_steps": res["outer_steps"]}


def test_elastic_fsdp_two_by_two():
    gport = free_port()
    pA, pB = free_port(), free_port()
    base = {"PRIME_GLOBAL_PORT": gport, "MASTER_ADDR": "127.0.0.1",
            "WORLD_SIZE": 2}
    out = _spawn([
        (_fsdp_trainer, {**base, "RANK": 0, "LOCAL_RANK": 0,
                         "MASTER_PORT": pA, "PRIME_GLOBAL_HOST": 1},
         (gport, True, 4, "a0")),
        (_fsdp_trainer, {**base, "RANK": 1, "LOCAL_RANK": 1,
                         "MASTER_PORT": pA}, (gport, False, 4, "a1")),
        (_fsdp_trainer, {**base, "RANK": 0, "LOCAL_RANK": 0,
                         "MASTER_PORT": pB}, (gport, False, 4, "b0")),
        (_fsdp_trainer, {**base, "RANK": 1, "LOCAL_RANK": 1,
                         "MASTER_PORT": pB}, (gport, False, 4, "b1")),
    ])
    by = {(0, "a"): out[0], (1, "a"): out[1], (0, "b"): out[2], (1, "b"): out[3]}
    assert out[0]["outer_steps"] == 2
    # shard-aligned: rank r of worker A converged with rank r of worker B
    for r in (0, 1):
        assert by[(r, "a")]["shard"] == by[(r, "b")]["shard"], r
        assert by[(r, "a")]["outer"] == by[(r, "b")]["outer"], r
    # distinct shards are actually distinct (not a broadcast-flattened copy)
    assert by[(0, "a")]["shard"] != by[(1, "a")]["shard"]


# --------------------------- multi-rank eviction (deadlock regression)
def _mk_worker(gport, mesh_host: bool):
    """Build a 2-rank FSDP worker's DilocoOptimizer (CPU/gloo)."""
    import torch

    from prime_amd.models import build_model
    from prime_amd.parallel.diloco import DilocoOptimizer
    from prime_amd.parallel.elastic import ElasticShardClient, ElasticWorker
    from prime_amd.parallel.flat import FusedAdamW
    from prime_amd.parallel.fsdp import ShardedParamSpace
    from prime_amd.parallel.mesh import ElasticDeviceMesh, MeshConfig

    torch.manual_seed(0)
    mesh = ElasticDeviceMesh(MeshConfig(worker_size=2))
    model = build_model("llama_test", activation_checkpointing=True)
    flat = ShardedParamSpace(model, mesh)
    el, sc = None, None
    if mesh.is_leader:
        el = ElasticWorker(port=gport, host_store=mesh_host, worker_name="w",
                           heartbeat_interval=0.2, heartbeat_timeout=1.5,
                           ckpt_provider=lambda: dl.live_state())
    else:
        sc = ElasticShardClient(shard_rank=mesh.worker_rank, port=gport,
                                ckpt_provider=lambda: dl.live_state())
    dl = DilocoOptimizer(flat, mesh, FusedAdamW(flat), H=1, elastic=el,
                         elastic_mode=True, shard_client=sc, sharded=True)
    return mesh, el, sc, dl


def _stall_fsdp_worker(gport, rank):
    import threading

    import torch.distributed as dist

    mesh, el, sc, dl = _mk_worker(gport, mesh_host=False)
    store = el.store if el is not None else sc.store
    if el is not None:
        while len(el.members()) < 2:
            time.sleep(0.05)
        # publish wid so the shard client can register its ckpt server
    dist.barrier()
    dl.outer_step()                    # boundary 1: both workers present
    # stall the whole job past the heartbeat timeout
    if el is not None:
        el._stop.set()
        el._hb_thread.join()
        store.set("t/stalling", "1")
    store.wait(["t/evicted"])
    if el is not None:
        el._stop.clear()
        el._hb_thread = threading.Thread(target=el._hb_loop, daemon=True)
        el._hb_thread.start()
    dist.barrier()
    # the deadlock regression: BOTH ranks must return from this call
    dl.outer_step()                    # leader: evicted -> rejoin+bootstrap
    if el is not None:
        assert getattr(dl, "rejoined", 0) == 1
        store.set("t/rejoined", "1")
    dist.barrier()
    dl.outer_step()                    # participates again (world 2)
    if el is not None:
        store.set("t/staller_done", "1")
    out = {"rank": rank, "outer": dl.theta_outer[:8].tolist(),
           "shard": dl.flat.flat_w[:8].tolist()}
    dist.barrier()
    (el or sc).close()
    dist.destroy_process_group()
    return out


def _surv_fsdp_worker(gport, rank):
    import torch.distributed as dist

    mesh, el, sc, dl = _mk_worker(gport, mesh_host=(rank == 0))
    store = el.store if el is not None else sc.store
    if el is not None:
        while len(el.members()) < 2:
            time.sleep(0.05)
    dist.barrier()
    dl.outer_step()                    # boundary 1
    store.wait(["t/stalling"])
    time.sleep(2.0)                    # staller heartbeat goes stale
    dist.barrier()
    dl.outer_step()                    # evicts the staller (world 1)
    store.set("t/evicted", "1")
    store.wait(["t/rejoined"])
    dist.barrier()
    dl.outer_step()                    # world 2 again
    store.wait(["t/staller_done"])
    out = {"rank": rank, "outer": dl.theta_outer[:8].tolist(),
           "shard": dl.flat.flat_w[:8].tolist()}
    dist.barrier()
    (el or sc).close()
    dist.destroy_process_group()
    return out


def test_multirank_eviction_no_deadlock():
    gport = free_port()
    pA, pB = free_port(), free_port()
    base = {"PRIME_GLOBAL_PORT": gport, "MASTER_ADDR": "127.0.0.1",
            "WORLD_SIZE": 2}
    out = _spawn([
        (_surv_fsdp_worker, {**base, "RANK": 0, "MASTER_PORT": pA,
                             "PRIME_GLOBAL_HOST": 1}, (gport, 0)),
        (_surv_fsdp_worker, {**base, "RANK": 1, "MASTER_PORT": pA}, (gport, 1)),
        (_stall_fsdp_worker, {**base, "RANK": 0, "MASTER_PORT": pB}, (gport, 0)),
        (_stall_fsdp_worker, {**base, "RANK": 1, "MASTER_PORT": pB}, (gport, 1)),
    ], timeout=300)
    # after rejoin + final boundary, matching shards agree across workers
    surv = {o["rank"]: o for o in (out[0], out[1])}
    stall = {o["rank"]: o for o in (out[2], out[3])}
    for r in (0, 1):
        assert surv[r]["outer"] == stall[r]["outer"], r


# ------------------------------------------------ concurrent join stress
def _joiner(gport, host, n_members, idx):
    from prime_amd.parallel.elastic import ElasticWorker

    w = ElasticWorker(port=gport, host_store=host, worker_name="j",
                      heartbeat_interval=0.5, heartbeat_timeout=30.0)
    t0 = time.monotonic()
    while len(w.members()) < n_members:
        if time.monotonic() - t0 > 60:
            raise AssertionError(f"only {len(w.members())} registered")
        time.sleep(0.01)
    got = w.members()
    w.store.set(f"jdone/{idx}", "1")
    if host:  # the store host must outlive every polling peer
        w.store.wait([f"jdone/{i}" for i in range(n_members)])
    w.close(leaving=False)
    return {"n": len(got)}


def test_concurrent_join_stress():
    """8 workers register simultaneously; the member_index compare_set
    loop must not lose any registration (RMW-race regression)."""
    gport = free_port()
    jobs = [(_joiner, {}, (gport, i == 0, 8, i)) for i in range(8)]
    out = _spawn(jobs, timeout=120)
    for i in range(8):
        assert out[i]["n"] == 8


# --------------------------------------------- store-host failover
def _store_proc(port, ready_evt, stop_evt):
    from torch.distributed import TCPStore

    store = TCPStore("127.0.0.1", port, is_master=True, wait_for_workers=False)
    ready_evt.set()
    stop_evt.wait(120)
    del store


def _failover_worker(gport, idx):
    from prime_amd.parallel.elastic import ElasticWorker

    w = ElasticWorker(port=gport, host_store=False, worker_name=f"f{idx}",
                      heartbeat_interval=0.3, heartbeat_timeout=10.0)
    while len(w.members()) < 2:
        time.sleep(0.05)
    v1 = w.sync()
    # coordinate the store restart through files (the store dies)
    import pathlib

    flagdir = pathlib.Path("/tmp/prime_amd_test/failover")
    flagdir.mkdir(parents=True, exist_ok=True)
    (flagdir / f"synced{idx}").write_text("1")
    while not (flagdir / "store_restarted").exists():
        time.sleep(0.1)
    # next boundary: the old store is gone -> reconnect path
    t0 = time.monotonic()
    while len(w.members()) < 2:  # both re-registered on the new store
        if time.monotonic() - t0 > 60:
            raise AssertionError("peers did not reconverge")
        time.sleep(0.05)
    v2 = w.sync()
    w.close(leaving=False)
    return {"w1": v1.world, "w2": v2.world, "rewid": w.wid.split("-")[1]}


def test_store_host_failover():
    """Kill the standalone registry mid-run, restart it empty on the same
    port: workers reconnect, re-register and reconverge to world=2."""
    import multiprocessing as mp
    import pathlib
    import shutil

    shutil.rmtree("/tmp/prime_amd_test/failover", ignore_errors=True)
    gport = free_port()
    ctx = mp.get_context("spawn")
    ready, stop = ctx.Event(), ctx.Event()
    store_p = ctx.Process(target=_store_proc, args=(gport, ready, stop))
    store_p.start()
    assert ready.wait(30)

    q = ctx.Queue()
    procs = []
    for i in range(2):
        pr = ctx.Process(target=_entry, args=(_failover_worker, i, {}, (gport, i), q))
        pr.start()
        procs.append(pr)
    flagdir = pathlib.Path("/tmp/prime_amd_test/failover")
    t0 = time.monotonic()
    while not all((flagdir / f"synced{i}").exists() for i in range(2)):
        assert time.monotonic() - t0 < 60
        time.sleep(0.1)
    store_p.terminate()
    store_p.join(10)
    ready2, stop2 = ctx.Event(), ctx.Event()
    store_p2 = ctx.Process(target=_store_proc, args=(gport, ready2, stop2))
    store_p2.start()
    assert ready2.wait(30)
    (flagdir / "store_restarted").write_text("1")

    results, errs = {}, []
    for _ in range(2):
        kind, i, payload = q.get(timeout=180)
        if kind == "ok":
            results[i] = payload
        else:
            errs.append((i, payload))
    for pr in procs:
        pr.join(timeout=30)
        if pr.is_alive():
            pr.terminate()
    stop2.set()
    store_p2.terminate()
    assert not errs, errs
    for i in (0, 1):
        assert results[i]["w1"] == 2 and results[i]["w2"] == 2
        assert results[i]["rewid"] == "re"  # went through reconnect()
