"""Gateway settings from environment / .env.

Mirrors the reference's Settings surface
(/root/reference/llm_gateway_core/config/settings.py:16-35): fallback
provider, gateway API key, log-file limit, port, chat-message logging flag
and CORS origins — without the pydantic-settings/python-dotenv dependencies
(not available offline). A tiny .env parser covers the same use.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from pathlib import Path
from typing import List, Optional


def load_dotenv(path: str | os.PathLike = ".env", override: bool = False) -> bool:
    """Load KEY=VALUE lines from a .env file into os.environ."""
    p = Path(path)
    if not p.exists():
        return False
    for raw in p.read_text(encoding="utf-8").splitlines():
        line = raw.strip()
        if not line or line.startswith("#") or "=" not in line:
            continue
        key, _, value = line.partition("=")
        key, value = key.strip(), value.strip()
        if len(value) >= 2 and value[0] == value[-1] and value[0] in "\"'":
            value = value[1:-1]
        if override or key not in os.environ:
            os.environ[key] = value
    return True


def _as_bool(v: Optional[str], default: bool = False) -> bool:
    if v is None:
        return default
    return v.strip().lower() in ("1", "true", "yes", "on")


@dataclass
class Settings:
    fallback_provider: str = "openrouter"
    gateway_api_key: Optional[str] = None
    log_file_limit: int = 15
    gateway_host: str = "0.0.0.0"
    gateway_port: int = 9100
    log_chat_messages: bool = False
    cors_origins_raw: str = "*"
    # MI355X engine knobs (no reference equivalent — local-engine additions)
    engine_kv_block_size: int = 64
    engine_max_batch_size: int = 256
    engine_hbm_fraction: float = 0.90  # fraction of free HBM given to KV cache
    engine_max_queue: int = 2048       # admission cap: overload fails fast to fallback
    engine_prefix_caching: bool = True # content-addressed KV reuse of shared prompt prefixes
    engine_use_hipgraph: bool = True
    extra: dict = field(default_factory=dict)

    @property
    def cors_allow_origins(self) -> List[str]:
        raw = self.cors_origins_raw.strip()
        if not raw or raw == "*":
            return ["*"]
        return [o.strip() for o in raw.split(",") if o.strip()]

    @classmethod
    def from_env(cls, dotenv_path: str | os.PathLike = ".env") -> "Settings":
        load_dotenv(dotenv_path)
        env = os.environ
        return cls(
            fallback_provider=env.get("FALLBACK_PROVIDER", "openrouter"),
            gateway_api_key=env.get("GATEWAY_API_KEY") or None,
            log_file_limit=int(env.get("LOG_FILE_LIMIT", "15")),
            gateway_host=env.get("GATEWAY_HOST", "0.0.0.0"),
            gateway_port=int(env.get("GATEWAY_PORT", "9100")),
            log_chat_messages=_as_bool(env.get("LOG_CHAT_MESSAGES"), False),
            cors_origins_raw=env.get("CORS_ALLOW_ORIGINS", "*"),
            engine_kv_block_size=int(env.get("ENGINE_KV_BLOCK_SIZE", "64")),
            engine_max_batch_size=int(env.get("ENGINE_MAX_BATCH_SIZE", "256")),
            engine_hbm_fraction=float(env.get("ENGINE_HBM_FRACTION", "0.90")),
            engine_max_queue=int(env.get("ENGINE_MAX_QUEUE", "2048")),
            engine_prefix_caching=_as_bool(env.get("ENGINE_PREFIX_CACHING"), True),
            engine_use_hipgraph=_as_bool(env.get("ENGINE_USE_HIPGRAPH"), True),
        )


# Module-level singleton, as the reference exposes. Tests construct their own.
settings = Settings.from_env()
