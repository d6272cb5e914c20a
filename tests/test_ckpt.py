import torch

from prime_amd.ckpt import CheckpointManager
from prime_amd.utils.config import (
    CheckpointConfig, DilocoConfig, MetricsConfig, ModelConfig, TrainConfig,
)
from prime_amd.train import Trainer


def test_save_load_roundtrip(tmp_path):
    mgr = CheckpointManager(tmp_path / "ck", keep=2, async_save=False)
    t = {"a": torch.arange(10, dtype=torch.float32), "b": torch.ones(4)}
    mgr.save(1, t, {"step": 1})
    out = mgr.load()
    assert out is not None
    torch.testing.assert_close(out["tensors"]["a"], t["a"])
    assert out["meta"]["step"] == 1


def test_retention(tmp_path):
    mgr = CheckpointManager(tmp_path / "ck", keep=2, async_save=False)
    for s in (1, 2, 3, 4):
        mgr.save(s, {"x": torch.tensor([float(s)])}, {})
    dirs = sorted(
        d.name for d in (tmp_path / "ck").iterdir()
        if d.is_dir() and not d.is_symlink()
    )
    assert dirs == ["step_3", "step_4"]
    assert (tmp_path / "ck" / "latest").resolve().name == "step_4"


def _train_cfg(steps, tmp_path, resume=None):
    cfg = TrainConfig(
        run_name="ckpt_test",
        steps=steps,
        model=ModelConfig(name="llama_test", seq_len=64),
        diloco=DilocoConfig(H=2),
        checkpoint=CheckpointConfig(
            interval=1, path=str(tmp_path / "ck"), async_save=False, resume=resume
        ),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 2
    return cfg


def test_trainer_resume(tmp_path):
    tr = Trainer(_train_cfg(4, tmp_path), run_dir=tmp_path / "runA")
    tr.run()
    w_end = tr.flat.flat_w.clone()
    inner_end = tr.diloco.inner_step_count
    tr.close()

    tr2 = Trainer(_train_cfg(6, tmp_path, resume="latest"), run_dir=tmp_path / "runB")
    assert tr2.diloco.inner_step_count == inner_end
    torch.testing.assert_close(tr2.flat.flat_w, w_end)
    res = tr2.run()  # continues training up to the TOTAL step target
    assert res["steps"] == 6
    tr2.close()


def test_export_safetensors(tmp_path):
    from safetensors.torch import load_file

    tr = Trainer(_train_cfg(2, tmp_path), run_dir=tmp_path / "runE")
    tr.run()
    tr.close()
    from prime_amd.ckpt.manager import export_safetensors

    out = tmp_path / "m.safetensors"
    n = export_safetensors(tmp_path / "ck", "llama_test", out)
    assert n > 0 and out.exists()
    tensors = load_file(str(out))
    assert "tok_embeddings.weight" in tensors
    assert tensors["layers.0.attn.wqkv.weight"].shape[1] == 64


def test_remote_copy(tmp_path):
    mgr = CheckpointManager(tmp_path / "ck", keep=2, async_save=False,
                            remote_path=tmp_path / "remote")
    mgr.save(1, {"x": torch.arange(4, dtype=torch.float32)}, {"step": 1})
    assert (tmp_path / "remote" / "step_1" / "worker0.pt").exists()
    assert (tmp_path / "remote" / "step_1" / "meta.json").exists()


def test_cpu_tensors_staged_not_live(tmp_path):
    """A later outer step mutating host-resident theta_outer in place must
    not corrupt the snapshot a background save is serializing."""
    mgr = CheckpointManager(tmp_path / "ck", async_save=False)
    t = torch.arange(8, dtype=torch.float32)
    staged = mgr._stager.stage_cpu({"theta_outer": t})
    t.mul_(100.0)  # in-place mutation after staging
    assert staged["theta_outer"][3].item() == 3.0


def test_import_export_roundtrip(tmp_path):
    """export -> import -> the reloaded flat master matches, and the
    imported checkpoint trains/serves (resume path)."""
    from prime_amd.ckpt.manager import export_safetensors, import_safetensors
    from prime_amd.models import build_model
    from prime_amd.parallel.flat import FlatParamSpace

    tr = Trainer(_train_cfg(2, tmp_path), run_dir=tmp_path / "runX")
    tr.run()
    tr.close()
    out = tmp_path / "w.safetensors"
    export_safetensors(tmp_path / "ck", "llama_test", out)
    n = import_safetensors(out, "llama_test", tmp_path / "ck2")
    assert n > 0
    m = build_model("llama_test")
    flat = FlatParamSpace(m)
    from prime_amd.ckpt import CheckpointManager

    payload = CheckpointManager(tmp_path / "ck2").load()
    flat.load_flat_(payload["tensors"]["master32"])
    # bf16 quantization is the only loss in the roundtrip
    ref = CheckpointManager(tmp_path / "ck").load()["tensors"]["master32"]
    err = (flat.master32 - ref).abs().max()
    assert float(err) < 2e-2

    # strict mode catches missing tensors
    import pytest
    from safetensors.torch import load_file, save_file

    t = load_file(str(out))
    t.pop("layers.0.attn.wqkv.weight")
    partial = tmp_path / "partial.safetensors"
    save_file(t, str(partial))
    with pytest.raises(ValueError, match="missing"):
        import_safetensors(partial, "llama_test", tmp_path / "ck3")


def test_capacity_guard_refuses_oversized(tmp_path, monkeypatch):
    """A checkpoint bigger than the target filesystem must refuse loudly
    (a tmpfs target otherwise eats host RAM until the OOM killer takes
    the node - observed on a GPU soak run)."""
    import shutil as _shutil

    import pytest

    mgr = CheckpointManager(tmp_path / "ck", async_save=False)

    class FakeUsage:
        free = 1 << 20  # 1 MiB free

    monkeypatch.setattr(_shutil, "disk_usage", lambda p: FakeUsage)
    with pytest.raises(RuntimeError, match="refusing"):
        mgr.save(1, {"x": torch.zeros(1 << 22)}, {})  # 16 MiB > 1 MiB free


def test_resume_trajectory_matches_uninterrupted(tmp_path):
    """Stopping at an outer boundary and resuming must reproduce the
    uninterrupted run EXACTLY: same data order (loader batch_idx), same
    LR schedule (step_count), same AdamW moments — so the final weights
    of 2+2 steps equal those of 4 straight steps."""
    torch.manual_seed(0)
    tr = Trainer(_train_cfg(4, tmp_path / "a"), run_dir=tmp_path / "runA")
    tr.run()
    w_straight = tr.flat.flat_w.clone()
    m_straight = tr.inner.m.clone()
    tr.close()

    torch.manual_seed(0)
    tr1 = Trainer(_train_cfg(2, tmp_path / "b"), run_dir=tmp_path / "runB1")
    tr1.run()
    tr1.close()
    tr2 = Trainer(_train_cfg(4, tmp_path / "b", resume="latest"),
                  run_dir=tmp_path / "runB2")
    res = tr2.run()
    assert res["steps"] == 4
    torch.testing.assert_close(tr2.flat.flat_w, w_straight)
    torch.testing.assert_close(tr2.inner.m, m_straight)
    tr2.close()
