"""Elastic fabric tests (CPU): membership sync, join with live peer
recovery, dead-peer eviction, int8 ring over the rebuilt gloo group."""
import json
import multiprocessing as mp
import os
import time

import pytest
import torch

from tests.conftest import free_port


def _spawn(fns_envs, timeout=180):
    """Run [(fn, env, args)] in separate spawn processes; collect results."""
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
        (results if kind == "ok" else errs.__class__)  # noqa
        if kind == "ok":
            results[i] = payload
        else:
            errs.append((i, payload))
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if errs:
        raise AssertionError(f"elastic workers failed: {errs}")
    return results


def _entry(fn, idx, env, args, q):
    os.environ.update({k: str(v) for k, v in env.items()})
    try:
        q.put(("ok", idx, fn(*args)))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", idx, f"{e}\n{traceback.format_exc()}"))


# --------------------------------------------------------------- workers
def _basic_worker(host, port, n_syncs):
    from prime_amd.parallel.elastic import ElasticWorker, ring_allreduce_int8_pg
    from prime_amd.ops import QBLK

    w = ElasticWorker(port=port, host_store=host, worker_name="t",
                      heartbeat_interval=0.3, heartbeat_timeout=5.0,
                      ckpt_provider=lambda: {"x": torch.ones(4)})
    # wait until both registered so the first view contains both
    while len(w.members()) < 2:
        time.sleep(0.05)
    worlds = []
    val = None
    for _ in range(n_syncs):
        view = w.sync()
        worlds.append(view.world)
        t = torch.full((view.world * QBLK,), float(view.my_index + 1))
        if view.pg is not None:
            ring_allreduce_int8_pg(t, view.pg, view.my_index, view.world)
        val = t[:4].tolist()
    w.close()
    return {"worlds": worlds, "val": val, "wid": w.wid}


def test_two_workers_sync_and_ring():
    port = free_port()
    out = _spawn([
        (_basic_worker, {}, (True, port, 2)),
        (_basic_worker, {}, (False, port, 2)),
    ])
    assert out[0]["worlds"] == [2, 2]
    assert out[1]["worlds"] == [2, 2]
    # ring average of [1,1..] and [2,2..] = 1.5 (int8 quant exact here)
    assert out[0]["val"] == pytest.approx([1.5] * 4, abs=0.02)
    assert out[0]["val"] == out[1]["val"]


def _early_worker(port, q_state):
    import torch

    from prime_amd.parallel.elastic import ElasticWorker

    state = {"theta_outer": torch.arange(8, dtype=torch.float32),
             "outer_buf": torch.zeros(8), "inner_step": 6, "outer_step": 3}
    w = ElasticWorker(port=port, host_store=True, worker_name="early",
                      heartbeat_interval=0.3, heartbeat_timeout=5.0,
                      ckpt_provider=lambda: state)
    v1 = w.sync()  # alone
    # signal the joiner it can start, then wait for it
    w.store.set("test/early_done_first", "1")
    while len([m for m in w.members() if m != w.wid]) < 1:
        time.sleep(0.05)
    w.store.wait(["test/joiner_bootstrapped"])
    v2 = w.sync()  # now two members
    w.close()
    return {"w1": v1.world, "w2": v2.world}


def _join_worker(port):
    from prime_amd.parallel.elastic import ElasticWorker

    w = ElasticWorker(port=port, host_store=False, worker_name="late",
                      heartbeat_interval=0.3, heartbeat_timeout=5.0,
                      ckpt_provider=lambda: {})
    w.store.wait(["test/early_done_first"])
    payload = w.bootstrap_from_peer()
    w.store.set("test/joiner_bootstrapped", "1")
    v = w.sync()
    w.close()
    return {"outer_step": payload["outer_step"],
            "theta": payload["theta_outer"].tolist(), "w": v.world}


def test_join_with_live_recovery():
    port = free_port()
    out = _spawn([
        (_early_worker, {}, (port, None)),
        (_join_worker, {}, (port,)),
    ])
    assert out[0]["w1"] == 1 and out[0]["w2"] == 2
    assert out[1]["outer_step"] == 3
    assert out[1]["theta"] == [float(i) for i in range(8)]
    assert out[1]["w"] == 2


def _survivor_worker(port):
    from prime_amd.parallel.elastic import ElasticWorker

    w = ElasticWorker(port=port, host_store=True, worker_name="surv",
                      heartbeat_interval=0.2, heartbeat_timeout=1.5,
                      ckpt_provider=lambda: {})
    while len(w.members()) < 2:
        time.sleep(0.05)
    v1 = w.sync()
    w.store.wait(["test/crasher_gone"])
    time.sleep(2.0)  # let the heartbeat go stale
    v2 = w.sync()    # must evict the crashed peer, not hang
    w.close()
    return {"w1": v1.world, "w2": v2.world}


def _crash_worker(port):
    from prime_amd.parallel.elastic import ElasticWorker

    w = ElasticWorker(port=port, host_store=False, worker_name="crash",
                      heartbeat_interval=0.2, heartbeat_timeout=1.5,
                      ckpt_provider=lambda: {})
    while len(w.members()) < 2:
        time.sleep(0.05)
    w.sync()
    # die abruptly: stop heartbeating without deregistering
    w._stop.set()
    w.store.set("test/crasher_gone", "1")
    return {"ok": True}


def test_dead_peer_eviction():
    port = free_port()
    out = _spawn([
        (_survivor_worker, {}, (port,)),
        (_crash_worker, {}, (port,)),
    ], timeout=120)
    assert out[0]["w1"] == 2
    assert out[0]["w2"] == 1  # crashed peer evicted


# ------------------------------------------------ end-to-end via Trainer
def _elastic_trainer(port, host, steps):
    from prime_amd.train import Trainer
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )

    cfg = TrainConfig(
        run_name=f"el_{host}",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=64),
        diloco=DilocoConfig(H=2),
        parallel=ParallelConfig(elastic=True, heartbeat_interval=0.3,
                                heartbeat_timeout=10.0),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/elastic_{host}")
    # both workers must be registered before the first boundary so the
    # first view has world=2 (otherwise the test's equality check is moot)
    while len(tr.elastic.members()) < 2:
        time.sleep(0.05)
    res = tr.run()
    head = tr.flat.flat_w[:16].tolist()
    outer = tr.diloco.theta_outer[:16].tolist()
    tr.close()
    return {"head": head, "outer": outer, "outer_steps": res["outer_steps"]}


def test_elastic_trainer_two_workers():
    port = free_port()
    env = {"PRIME_GLOBAL_PORT": port, "WORLD_SIZE": 1, "RANK": 0,
           "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": free_port()}
    out = _spawn([
        (_elastic_trainer, {**env, "PRIME_GLOBAL_HOST": 1}, (port, True, 4)),
        (_elastic_trainer, {**env, "PRIME_GLOBAL_HOST": 0,
                            "MASTER_PORT": free_port()}, (port, False, 4)),
    ], timeout=300)
    assert out[0]["outer_steps"] == 2
    assert out[0]["head"] == out[1]["head"]
    assert out[0]["outer"] == out[1]["outer"]


def _stall_worker(port):
    """Worker that stalls past the heartbeat timeout mid-run, gets evicted,
    and must auto-rejoin at its next boundary."""
    import torch

    from prime_amd.parallel.elastic import ElasticWorker
    from prime_amd.parallel.diloco import DilocoOptimizer
    from prime_amd.parallel.flat import FlatParamSpace, FusedAdamW
    from prime_amd.parallel.mesh import ElasticDeviceMesh, MeshConfig
    from prime_amd.models import build_model

    torch.manual_seed(0)
    mesh = ElasticDeviceMesh(MeshConfig())
    model = build_model("llama_test")
    flat = FlatParamSpace(model)
    el = ElasticWorker(port=port, host_store=False, worker_name="staller",
                       heartbeat_interval=0.2, heartbeat_timeout=1.5,
                       ckpt_provider=lambda: dl.live_state())
    dl = DilocoOptimizer(flat, mesh, FusedAdamW(flat), H=1, elastic=el)
    while len(el.members()) < 2:
        time.sleep(0.05)
    dl.outer_step()           # boundary 1: both present
    # stall: stop heartbeating long enough to be evicted
    el._stop.set()
    el._hb_thread.join()
    el.store.set("test/stalling", "1")
    el.store.wait(["test/evicted_me"])
    el._stop.clear()
    import threading

    el._hb_thread = threading.Thread(target=el._hb_loop, daemon=True)
    el._hb_thread.start()
    dl.outer_step()           # detects eviction -> rejoin + bootstrap
    assert getattr(dl, "rejoined", 0) == 1
    el.store.set("test/rejoined", "1")
    dl.outer_step()           # participates again (world 2)
    el.store.set("test/staller_done", "1")
    el.close()
    return {"rejoined": dl.rejoined, "outer": dl.outer_step_count}


def _survivor2_worker(port):
    import torch

    from prime_amd.parallel.elastic import ElasticWorker
    from prime_amd.parallel.diloco import DilocoOptimizer
    from prime_amd.parallel.flat import FlatParamSpace, FusedAdamW
    from prime_amd.parallel.mesh import ElasticDeviceMesh, MeshConfig
    from prime_amd.models import build_model

    torch.manual_seed(0)
    mesh = ElasticDeviceMesh(MeshConfig())
    model = build_model("llama_test")
    flat = FlatParamSpace(model)
    el = ElasticWorker(port=port, host_store=True, worker_name="surv2",
                       heartbeat_interval=0.2, heartbeat_timeout=1.5,
                       ckpt_provider=lambda: dl.live_state())
    dl = DilocoOptimizer(flat, mesh, FusedAdamW(flat), H=1, elastic=el)
    while len(el.members()) < 2:
        time.sleep(0.05)
    dl.outer_step()           # boundary 1
    el.store.wait(["test/stalling"])
    time.sleep(2.0)           # staller's heartbeat goes stale
    dl.outer_step()           # evicts the staller (world 1)
    el.store.set("test/evicted_me", "1")
    worlds = [dl._view.world]
    el.store.wait(["test/rejoined"])  # deterministic: staller re-registered
    dl.outer_step()           # staller rejoined -> world 2 again
    worlds.append(dl._view.world)
    el.store.wait(["test/staller_done"])
    el.close()
    return {"worlds": worlds}


def test_evicted_worker_auto_rejoins():
    port = free_port()
    out = _spawn([
        (_survivor2_worker, {}, (port,)),
        (_stall_worker, {}, (port,)),
    ], timeout=240)
    assert out[1]["rejoined"] == 1
    assert out[0]["worlds"][0] == 1   # after eviction
    assert out[0]["worlds"][1] == 2   # after rejoin


def _weighted_worker(port, host, contribution):
    import torch

    from prime_amd.parallel.elastic import ElasticWorker, ring_allreduce_int8_pg
    from prime_amd.ops import QBLK

    w = ElasticWorker(port=port, host_store=host, worker_name="wt",
                      heartbeat_interval=0.3, heartbeat_timeout=5.0,
                      ckpt_provider=lambda: {})
    while len(w.members()) < 2:
        time.sleep(0.05)
    view = w.sync(contribution=contribution)
    delta = torch.full((view.world * QBLK,), 1.0 if host else 3.0)
    delta.mul_(view.my_scale())
    ring_allreduce_int8_pg(delta, view.pg, view.my_index, view.world)
    w.close()
    return {"weights": view.weights, "avg": float(delta[0])}


def test_weighted_outer_average():
    port = free_port()
    # worker A contributed a full round (w=1.0), worker B half (w=0.5):
    # weighted mean of (1.0, 3.0) = (1*1 + 0.5*3)/1.5 = 5/3
    out = _spawn([
        (_weighted_worker, {}, (port, True, 1.0)),
        (_weighted_worker, {}, (port, False, 0.5)),
    ])
    assert sorted(out[0]["weights"]) == [0.5, 1.0]
    assert abs(out[0]["avg"] - 5.0 / 3.0) < 0.05
    assert abs(out[1]["avg"] - 5.0 / 3.0) < 0.05


def _departing_trainer(port, host, steps):
    from prime_amd.train import Trainer
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )

    cfg = TrainConfig(
        run_name=f"dep_{host}",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=64),
        diloco=DilocoConfig(H=2),
        parallel=ParallelConfig(elastic=True, heartbeat_interval=0.3,
                                heartbeat_timeout=8.0),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/depart_{host}_{steps}")
    while len(tr.elastic.members()) < 2:
        time.sleep(0.05)
    res = tr.run()
    last_world = tr.diloco._view.world if tr.diloco._view else 1
    tr.close()  # graceful leave: sets leaving/<wid> + deregisters
    return {"outer_steps": res["outer_steps"], "last_world": last_world}


def test_graceful_departure_mid_run():
    """Worker A runs 4 steps (2 boundaries) then leaves; worker B runs 8
    steps — its later boundaries must proceed at world=1, not hang."""
    port = free_port()
    env = {"PRIME_GLOBAL_PORT": port, "WORLD_SIZE": 1, "RANK": 0,
           "MASTER_ADDR": "127.0.0.1"}
    out = _spawn([
        (_departing_trainer, {**env, "PRIME_GLOBAL_HOST": 1,
                              "MASTER_PORT": free_port()}, (port, True, 8)),
        (_departing_trainer, {**env, "PRIME_GLOBAL_HOST": 0,
                              "MASTER_PORT": free_port()}, (port, False, 4)),
    ], timeout=300)
    assert out[1]["outer_steps"] == 2          # departed after 2 boundaries
    assert out[0]["outer_steps"] == 4          # survivor completed all
    assert out[0]["last_world"] == 1           # finished alone
