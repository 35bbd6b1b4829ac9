from .jsonc import loads as jsonc_loads, load as jsonc_load, JsoncError
from .settings import Settings, settings
from .loader import (
    ConfigError,
    ConfigLoader,
    EngineSpec,
    FallbackModelRule,
    ModelFallbackConfig,
    ProviderDetails,
)

__all__ = [
    "jsonc_loads",
    "jsonc_load",
    "JsoncError",
    "Settings",
    "settings",
    "ConfigError",
    "ConfigLoader",
    "EngineSpec",
    "FallbackModelRule",
    "ModelFallbackConfig",
    "ProviderDetails",
]
