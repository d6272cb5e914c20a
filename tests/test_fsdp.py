"""FSDP sharding tests on CPU/gloo: 2-rank sharded training must match
2-rank data-parallel training (same seed, same data shards)."""
import torch

from tests.conftest import run_distributed


def _train_worker(rank, world, fsdp, steps):
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )
    from prime_amd.train import Trainer

    cfg = TrainConfig(
        run_name=f"fsdp_{fsdp}",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=64,
                          activation_checkpointing=True),
        diloco=DilocoConfig(H=10**6),  # pure inner loop for this test
        parallel=ParallelConfig(worker_size=world, fsdp=fsdp),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/fsdp{fsdp}_r{rank}")
    losses = []
    for _ in range(steps):
        losses.append(float(tr.train_step()))
    # materialize full params for comparison
    if fsdp:
        full = {}
        for u in tr.flat.units:
            tr.flat.gather_unit(u)
            for n, p in u.params:
                full[n] = p.detach().clone()
            tr.flat.release_unit(u)
        for n, p in tr.flat.repl_params:
            full[n] = p.detach().clone()
    else:
        full = {n: p.detach().clone() for n, p in tr.model.named_parameters()}
    tr.close()
    return {"losses": losses, "w": {k: v.sum().item() for k, v in full.items()},
            "head": full["layers.0.attn.wqkv.weight"].flatten()[:8].tolist()}


def test_fsdp_matches_dp_two_ranks():
    dp = run_distributed(_train_worker, 2, args=(False, 3), timeout=300)
    sh = run_distributed(_train_worker, 2, args=(True, 3), timeout=300)
    # same data, same seed => same loss trajectory (bf16-free CPU fp32 math)
    for a, b in zip(dp[0]["losses"], sh[0]["losses"]):
        assert abs(a - b) < 1e-3, (dp[0]["losses"], sh[0]["losses"])
    # parameters after training match between the two modes
    for k in dp[0]["w"]:
        assert abs(dp[0]["w"][k] - sh[0]["w"][k]) < 2e-2, k
    assert dp[0]["head"] == dp[1]["head"]  # DP replicas in sync
    assert sh[0]["head"] == sh[1]["head"]  # gathered shards identical


def _fsdp_diloco_worker(rank, world):
    from prime_amd.utils.config import (
        DilocoConfig, MetricsConfig, ModelConfig, ParallelConfig, TrainConfig,
    )
    from prime_amd.train import Trainer

    cfg = TrainConfig(
        run_name="fsdp_diloco",
        steps=4,
        model=ModelConfig(name="llama_test", seq_len=64,
                          activation_checkpointing=True),
        diloco=DilocoConfig(H=2),
        parallel=ParallelConfig(worker_size=world, fsdp=True),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/fsdp_diloco_r{rank}")
    res = tr.run()
    out = (res["outer_steps"], float(tr.flat.flat_w.float().sum()))
    tr.close()
    return out


def test_fsdp_with_diloco_outer():
    """FSDP inner sharding + DiLoCo outer step run together (single worker
    of 2 ranks: outer step is identity-average but exercises the path)."""
    outs = run_distributed(_fsdp_diloco_worker, 2, timeout=300)
    assert outs[0][0] == 2
    assert all(torch.isfinite(torch.tensor([o[1] for o in outs])))


def test_fsdp_requires_checkpointing():
    import pytest

    from prime_amd.models import build_model
    from prime_amd.parallel.fsdp import ShardedParamSpace

    class FakeMesh:
        local_group = object()

    m = build_model("llama_test")  # no activation checkpointing
    with pytest.raises(ValueError, match="activation_checkpointing"):
        ShardedParamSpace(m, FakeMesh())


def _fsdp_ckpt_worker(rank, world, tmpdir):
    from prime_amd.utils.config import (
        CheckpointConfig, DilocoConfig, MetricsConfig, ModelConfig,
        ParallelConfig, TrainConfig,
    )
    from prime_amd.train import Trainer

    def cfg(steps, resume=None):
        c = TrainConfig(
            run_name="fsdp_ck", steps=steps,
            model=ModelConfig(name="llama_test", seq_len=64,
                              activation_checkpointing=True),
            diloco=DilocoConfig(H=2),
            parallel=ParallelConfig(worker_size=world, fsdp=True),
            checkpoint=CheckpointConfig(interval=1, path=f"{tmpdir}/ck",
                                        async_save=False, resume=resume),
            metrics=MetricsConfig(log_interval=100),
        )
        c.data.micro_batch_size = 2
        return c

    tr = Trainer(cfg(2), run_dir=f"{tmpdir}/run_r{rank}")
    tr.run()
    shard_sum = float(tr.flat.master32.double().sum())
    tr.close(destroy_pg=False)  # second trainer reuses the process group

    tr2 = Trainer(cfg(2, resume="latest"), run_dir=f"{tmpdir}/run2_r{rank}")
    resumed_sum = float(tr2.flat.master32.double().sum())
    tr2.close()
    return {"saved": shard_sum, "resumed": resumed_sum}


def test_fsdp_per_shard_checkpoint(tmp_path):
    outs = run_distributed(_fsdp_ckpt_worker, 2, args=(str(tmp_path),),
                           timeout=300)
    for o in outs:
        assert abs(o["saved"] - o["resumed"]) < 1e-6
    # the two shards are genuinely different state
    assert outs[0]["saved"] != outs[1]["saved"]


# -------------------- async ordering stress (fake delayed collectives)
def _train_worker_async_fake(rank, world, steps):
    """Run the FSDP space with DELAYED fake collectives that emulate RCCL
    async semantics on CPU: outputs are poisoned with NaN until .wait(),
    and inputs are snapshotted at issue time — any read-before-wait
    propagates NaN into the loss, and any in-flight input mutation trips
    an assertion. This exercises the async reduce-scatter, 2-buffer
    parity reuse and prefetch drain ordering (fsdp.py) that the gloo
    fallback never runs."""
    import torch

    from prime_amd.parallel import fsdp as fsdp_mod

    class _DelayedWork:
        def __init__(self, fn, snap_src, src_live):
            self._fn = fn
            self._snap = snap_src
            self._live = src_live
            self._done = False

        def wait(self):
            if self._done:
                return
            self._done = True
            torch.testing.assert_close(
                self._live, self._snap, rtol=0, atol=0,
                msg="collective input mutated while in flight",
            )
            self._fn(self._snap)

    import torch.distributed as dist

    def fake_all_gather(full, shard, group, async_op=False):
        W = dist.get_world_size(group)
        snap = shard.detach().clone()
        live = shard

        def complete(src):
            chunks = list(full.chunk(W))
            dist.all_gather(chunks, src.contiguous(), group=group)

        full.fill_(float("nan"))  # poison until wait()
        w = _DelayedWork(complete, snap, live)
        if async_op:
            return w
        w.wait()
        return None

    def fake_reduce_scatter(out_shard, grad_full, group, async_op=False):
        W = dist.get_world_size(group)
        r = dist.get_rank(group)
        snap = grad_full.detach().clone()
        live = grad_full

        def complete(src):
            buf = src.clone()
            dist.all_reduce(buf, group=group)
            n = buf.numel() // W
            out_shard.copy_(buf[r * n : (r + 1) * n])

        out_shard.fill_(float("nan"))
        w = _DelayedWork(complete, snap, live)
        if async_op:
            return w
        w.wait()
        return None

    fsdp_mod._all_gather_flat = fake_all_gather
    fsdp_mod._reduce_scatter_flat = fake_reduce_scatter

    out = _train_worker(rank, world, True, steps)
    for ls in out["losses"]:
        assert ls == ls, "NaN loss: an output buffer was read before wait()"
    return out


def test_fsdp_async_ordering_stress():
    dp = run_distributed(_train_worker, 2, args=(False, 3), timeout=300)
    sh = run_distributed(_train_worker_async_fake, 2, args=(3,), timeout=300)
    for a, b in zip(dp[0]["losses"], sh[0]["losses"]):
        assert abs(a - b) < 1e-3, (dp[0]["losses"], sh[0]["losses"])
    for k in dp[0]["w"]:
        assert abs(dp[0]["w"][k] - sh[0]["w"][k]) < 2e-2, k


def _consensus_stop_worker(rank, world):
    """Only rank 1 receives the 'signal'; both ranks must stop together at
    the same boundary (the consensus all-reduce) with a final checkpoint."""
    from prime_amd.train import Trainer
    from prime_amd.utils.config import (
        CheckpointConfig, DilocoConfig, MetricsConfig, ModelConfig,
        ParallelConfig, TrainConfig,
    )

    cfg = TrainConfig(
        run_name="stopc",
        steps=50,
        model=ModelConfig(name="llama_test", seq_len=32),
        diloco=DilocoConfig(H=10**6),
        parallel=ParallelConfig(worker_size=world),
        checkpoint=CheckpointConfig(interval=1, async_save=False),
        metrics=MetricsConfig(log_interval=1000),
    )
    cfg.data.micro_batch_size = 1
    tr = Trainer(cfg, run_dir=f"/tmp/prime_amd_test/stopc_r{rank}")
    steps_done = 0
    orig = tr.train_step

    def step_and_maybe_stop():
        nonlocal steps_done
        out = orig()
        steps_done += 1
        if rank == 1 and steps_done == 3:
            tr.stop_requested = True  # simulated SIGTERM on ONE rank
        return out

    tr.train_step = step_and_maybe_stop
    res = tr.run()
    tr.close()
    return {"steps": res["steps"], "stopped": tr.stop_requested}


def test_consensus_stop_two_ranks():
    outs = run_distributed(_consensus_stop_worker, 2, args=(), timeout=300)
    # both ranks agreed to stop at the first boundary after rank 1's
    # step-3 request (stop is detected at the next loop top)
    assert outs[0]["steps"] == outs[1]["steps"] == 3
    assert outs[0]["stopped"] and outs[1]["stopped"]
