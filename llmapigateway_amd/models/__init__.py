from .configs import ModelConfig, get_model_config, MODEL_PRESETS
from .llama import LlamaModel

__all__ = ["ModelConfig", "get_model_config", "MODEL_PRESETS", "LlamaModel"]
