"""DiLoCo plumbing on CPU/gloo (BASELINE.json config 1: runs without GPUs)."""
import torch

from tests.conftest import run_distributed


def _diloco_worker(rank, world, steps, H, quant):
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, TrainConfig,
    )
    from prime_amd.train import Trainer

    cfg = TrainConfig(
        run_name=f"diloco_test_w{world}",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=64),
        diloco=DilocoConfig(H=H, quant_int8=quant),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/diloco_r{rank}")
    res = tr.run()
    # after a final outer step, params equal across workers
    flat_after = tr.flat.flat_w.clone()
    outer_after = tr.diloco.theta_outer.clone()
    tr.close()
    return {
        "loss": res["loss"],
        "outer_steps": res["outer_steps"],
        "flat_sum": float(flat_after.double().sum()),
        "flat_head": flat_after[:16].tolist(),
        "outer_head": outer_after[:16].tolist(),
    }


def test_diloco_two_workers_sync():
    # 4 steps, H=2 -> 2 outer steps; last step IS an outer boundary so
    # params must be bit-identical across workers afterwards
    outs = run_distributed(_diloco_worker, 2, args=(4, 2, True), timeout=300)
    assert outs[0]["outer_steps"] == 2
    assert outs[0]["flat_head"] == outs[1]["flat_head"]
    assert outs[0]["outer_head"] == outs[1]["outer_head"]
    assert all(torch.isfinite(torch.tensor(o["loss"])) for o in outs)


def test_diloco_fp32_outer():
    outs = run_distributed(_diloco_worker, 2, args=(2, 2, False), timeout=300)
    assert outs[0]["outer_steps"] == 1
    assert outs[0]["flat_head"] == outs[1]["flat_head"]


def test_diloco_single_worker_no_comm():
    # world=1: outer step still applies (self-average is identity)
    outs = run_distributed(_diloco_worker, 1, args=(2, 2, True), timeout=300)
    assert outs[0]["outer_steps"] == 1


def _mixed_worker(rank, world):
    """2 workers x 2 DP ranks: inner grad all-reduce + outer int8 ring."""
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )
    from prime_amd.train import Trainer

    cfg = TrainConfig(
        run_name="mixed",
        steps=4,
        model=ModelConfig(name="llama_test", seq_len=64),
        diloco=DilocoConfig(H=2),
        parallel=ParallelConfig(worker_size=2),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/mixed_r{rank}")
    res = tr.run()
    out = {
        "outer": res["outer_steps"],
        "head": tr.flat.flat_w[:16].tolist(),
        "worker": tr.mesh.worker_id,
        "n_workers": tr.mesh.n_workers,
    }
    tr.close()
    return out


def test_mixed_topology_2x2():
    outs = run_distributed(_mixed_worker, 4, timeout=300)
    assert outs[0]["n_workers"] == 2
    assert outs[0]["outer"] == 2
    # after the final outer boundary every rank holds identical params
    assert outs[0]["head"] == outs[1]["head"] == outs[2]["head"] == outs[3]["head"]


def _soak_worker(rank, world, steps, h):
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )
    from prime_amd.train import Trainer

    cfg = TrainConfig(
        run_name="soak8",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=32),
        diloco=DilocoConfig(H=h),
        parallel=ParallelConfig(worker_size=1),
        metrics=MetricsConfig(log_interval=1000),
    )
    cfg.data.micro_batch_size = 1
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/soak8_r{rank}")
    for _ in range(steps):
        tr.train_step()
    head = tr.diloco.theta_outer[:16].tolist()
    w = tr.flat.flat_w[:16].tolist()
    outer = tr.diloco.outer_step_count
    tr.close()
    return {"head": head, "w": w, "outer": outer}


def test_eight_worker_outer_soak():
    """8 DiLoCo workers on gloo through 3 outer boundaries with the
    multi-ring int8 all-reduce (default-on; W=8 engages 4 coprime
    offsets). All workers must hold identical outer state afterwards."""
    outs = run_distributed(_soak_worker, 8, args=(6, 2), timeout=600)
    assert outs[0]["outer"] == 3
    for r in range(1, 8):
        assert outs[r]["head"] == outs[0]["head"], r
        assert outs[r]["w"] == outs[0]["w"], r


def test_host_outer_buffer_tiers(monkeypatch):
    """Low-RAM hosts fall back to a disk-file-backed mapping instead of
    risking the OOM killer at N ranks x 84 GB of outer state."""
    from prime_amd.parallel.diloco import host_outer_buffer

    t = host_outer_buffer(1024, 1)
    assert t.numel() == 1024 and float(t.abs().sum()) == 0.0
    monkeypatch.setenv("PRIME_AMD_OUTER_FILEBACKED", "1")
    t2 = host_outer_buffer(1024, 8)
    assert t2.numel() == 1024 and float(t2.abs().sum()) == 0.0
    t2.fill_(3.0)  # mapping is writable
    assert float(t2[123]) == 3.0
