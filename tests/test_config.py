import pytest

from prime_amd.utils.config import (
    ConfigError,
    TrainConfig,
    default_config_toml,
    load_config,
)


def test_template_roundtrip(tmp_path):
    p = tmp_path / "cfg.toml"
    p.write_text(default_config_toml("llama_150m"))
    cfg = load_config(p)
    assert cfg.model.name == "llama_150m"
    assert cfg.diloco.H == 100
    assert cfg.diloco.quant_int8 is True


def test_unknown_key_rejected(tmp_path):
    p = tmp_path / "bad.toml"
    p.write_text('steps = 10\n[model]\nnam = "typo"\n')
    with pytest.raises(ConfigError) as ei:
        load_config(p)
    assert "model.nam" in str(ei.value)


def test_toml_syntax_error(tmp_path):
    p = tmp_path / "broken.toml"
    p.write_text("steps = [unclosed")
    with pytest.raises(ConfigError):
        load_config(p)


def test_missing_file():
    with pytest.raises(ConfigError):
        load_config("/nonexistent/cfg.toml")


def test_defaults():
    cfg = TrainConfig()
    assert cfg.parallel.worker_size == 1
    assert cfg.optim.lr == pytest.approx(3e-4)
    assert cfg.checkpoint.interval == 0


def test_shipped_configs_load():
    from pathlib import Path

    cfg_dir = Path(__file__).resolve().parent.parent / "configs"
    files = sorted(cfg_dir.glob("*.toml"))
    assert len(files) >= 5
    for f in files:
        cfg = load_config(f)
        assert cfg.steps > 0, f


def test_fp8_fsdp_rejected(tmp_path):
    from prime_amd.train import Trainer
    from prime_amd.utils.config import (ModelConfig, ParallelConfig,
                                        TrainConfig)

    import pytest

    cfg = TrainConfig(
        run_name="bad",
        model=ModelConfig(name="llama_test", fp8=True,
                          activation_checkpointing=True),
        parallel=ParallelConfig(fsdp=True, worker_size=2),
    )
    with pytest.raises(ValueError, match="fp8.*fsdp|fsdp.*fp8"):
        Trainer(cfg, run_dir=tmp_path)


def test_all_shipped_configs_parse():
    """Every TOML under configs/ must validate against the strict schema
    (the 8-GPU-node configs are otherwise only exercised by the driver)."""
    from pathlib import Path

    from prime_amd.utils.config import load_config

    root = Path(__file__).resolve().parent.parent / "configs"
    tomls = sorted(root.glob("*.toml"))
    assert len(tomls) >= 6
    for t in tomls:
        cfg = load_config(t)
        assert cfg.model.name


def test_fp8_flag_inert_on_cpu(tmp_path):
    """fp8 config on a CPU run must be a no-op (the plumbing configs keep
    working on fp8-flagged TOMLs)."""
    from prime_amd.train import Trainer
    from prime_amd.utils.config import (DilocoConfig, MetricsConfig,
                                        ModelConfig, TrainConfig)

    cfg = TrainConfig(
        run_name="fp8cpu", steps=2,
        model=ModelConfig(name="llama_test", seq_len=32, fp8=True,
                          fp8_dgrad=True, fp8_wgrad=True),
        diloco=DilocoConfig(H=10**6),
        metrics=MetricsConfig(log_interval=100),
    )
    cfg.data.micro_batch_size = 1
    tr = Trainer(cfg, run_dir=tmp_path)
    loss = tr.train_step()
    assert float(loss) > 0
    tr.close()


def test_numeric_bounds_rejected(tmp_path):
    """Out-of-range numerics fail at load time with a friendly ConfigError
    (H=0 or log_interval=0 would otherwise surface as ZeroDivisionError
    deep inside the step loop)."""
    import pytest

    from prime_amd.utils.config import ConfigError, load_config

    for body in (
        "[diloco]\nH = 0\n",
        "[metrics]\nlog_interval = 0\n",
        "[data]\nmicro_batch_size = 0\n",
        "[model]\nseq_len = -1\n",
        "[parallel]\nworker_size = 0\n",
        "steps = -5\n",
    ):
        f = tmp_path / "bad.toml"
        f.write_text('run_name = "x"\n' + body)
        with pytest.raises(ConfigError):
            load_config(str(f))
