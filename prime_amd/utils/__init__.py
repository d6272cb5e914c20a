from .config import ConfigError, TrainConfig, default_config_toml, load_config
from .logging import JsonLogger, render_log_line
from .metrics import MetricsWriter, mfu, model_flops_per_token, read_metrics

__all__ = [
    "ConfigError",
    "TrainConfig",
    "default_config_toml",
    "load_config",
    "JsonLogger",
    "render_log_line",
    "MetricsWriter",
    "mfu",
    "model_flops_per_token",
    "read_metrics",
]
