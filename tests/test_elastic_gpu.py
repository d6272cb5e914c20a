"""Elastic fabric on GPU: two single-GPU DiLoCo workers co-located on one
device, syncing over the TCPStore + gloo ring with CUDA<->CPU staging."""
import multiprocessing as mp
import os
import time

import pytest
import torch

from tests.conftest import free_port

pytestmark = pytest.mark.gpu


def _worker(idx, port, host, q):
    os.environ.update({
        "PRIME_GLOBAL_PORT": str(port), "PRIME_GLOBAL_HOST": "1" if host else "0",
        "WORLD_SIZE": "1", "RANK": "0", "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(free_port()),
    })
    try:
        from prime_amd.train import Trainer
        from prime_amd.utils.config import (
            DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
        )

        cfg = TrainConfig(
            run_name=f"el_gpu_{idx}", steps=4,
            model=ModelConfig(name="llama_150m", seq_len=128),
            diloco=DilocoConfig(H=2),
            parallel=ParallelConfig(elastic=True, heartbeat_interval=0.3,
                                    heartbeat_timeout=15.0),
            metrics=MetricsConfig(log_interval=100),
        )
        cfg.data.micro_batch_size = 1
        tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/elgpu_{idx}")
        assert tr.device.type == "cuda"
        while len(tr.elastic.members()) < 2:
            time.sleep(0.05)
        res = tr.run()
        head = tr.flat.flat_w[:16].float().cpu().tolist()
        tr.close()
        q.put(("ok", idx, {"outer": res["outer_steps"], "head": head}))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", idx, f"{e}\n{traceback.format_exc()}"))


def test_elastic_two_gpu_workers_one_device():
    port = free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(i, port, i == 0, q)) for i in range(2)]
    for p in procs:
        p.start()
    out, errs = {}, []
    for _ in range(2):
        kind, idx, payload = q.get(timeout=300)
        (out if kind == "ok" else errs.__class__)
        if kind == "ok":
            out[idx] = payload
        else:
            errs.append(payload)
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    assert not errs, errs
    assert out[0]["outer"] == 2
    assert out[0]["head"] == out[1]["head"]
