from .configs import CONFIGS, LlamaConfig, get_config
from .llama import Llama, build_model

__all__ = ["CONFIGS", "LlamaConfig", "get_config", "Llama", "build_model"]
