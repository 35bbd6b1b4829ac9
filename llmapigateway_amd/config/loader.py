"""Provider / fallback-rule config loading, validation and live reload.

Schema parity with the reference (providers.json is a list of one-key dicts,
models_fallback_rules.json is a list of rule objects — see
/root/reference/llm_gateway_core/config/loader.py:14-56 for the fields we
keep verbatim), with one extension point: a provider whose baseUrl uses the
``local://`` scheme (or that carries an ``engine`` object) resolves to a
GPU-resident MI355X inference engine instead of an HTTP upstream.

Differences from the reference, by design:
- initial-load failures raise ConfigError instead of sys.exit(1) — the app
  entrypoint decides process policy; reloads stay non-fatal
  (reference: loader.py:74,100,164 exits; loader.py:166-282 reloads).
- one ConfigLoader instance is shared by every router (the reference's
  models.py builds a second, stale instance at import time — models.py:14-16;
  we fix that quirk).
"""

from __future__ import annotations

import logging
import os
from pathlib import Path
from typing import Any, Dict, List, Optional, Tuple
from urllib.parse import parse_qs, urlparse

from pydantic import BaseModel, Field, ValidationError, field_validator

from . import jsonc

logger = logging.getLogger(__name__)


class ConfigError(RuntimeError):
    """Raised when a config file is missing or fails validation at load."""


class EngineSpec(BaseModel):
    """Local-engine resolution of a provider entry (MI355X extension).

    Parsed either from an explicit ``engine`` object on the provider entry or
    from a ``local://<model>?device=N&...`` baseUrl.
    """

    model: str = "llama-3-8b"           # architecture preset name (models/configs.py)
    device: int = 0                      # GPU ordinal (cuda:N == MI355X N)
    tp: int = 1                          # tensor-parallel degree over xGMI
    dtype: str = "bfloat16"
    max_batch_size: Optional[int] = None
    kv_block_size: Optional[int] = None
    kv_dtype: str = "auto"               # "auto" (= compute dtype) | "fp8"
    fail_rate: float = 0.0               # failure injection: P(request fails)
    fail_requests: Optional[int] = None  # failure injection: fail first N requests


class ProviderDetails(BaseModel):
    baseUrl: str
    apikey: str = ""
    engine: Optional[EngineSpec] = None

    model_config = {"extra": "allow"}  # reference tolerates extra fields (e.g. multiple_models)

    @property
    def is_local(self) -> bool:
        return self.baseUrl.startswith("local://") or self.engine is not None

    def engine_spec(self) -> EngineSpec:
        """Resolve the engine spec for a local provider entry."""
        spec = self.engine or EngineSpec()
        if self.baseUrl.startswith("local://"):
            u = urlparse(self.baseUrl)
            updates: Dict[str, Any] = {}
            if u.netloc:
                updates["model"] = u.netloc
            q = parse_qs(u.query)
            for key in ("device", "tp", "max_batch_size", "kv_block_size", "fail_requests"):
                if key in q:
                    updates[key] = int(q[key][0])
            if "dtype" in q:
                updates["dtype"] = q["dtype"][0]
            if "kv_dtype" in q:
                updates["kv_dtype"] = q["kv_dtype"][0]
            if "fail_rate" in q:
                updates["fail_rate"] = float(q["fail_rate"][0])
            if updates:
                spec = spec.model_copy(update=updates)
        return spec


class FallbackModelRule(BaseModel):
    provider: str
    model: str
    use_provider_order_as_fallback: bool = False
    providers_order: Optional[List[str]] = None
    retry_delay: Optional[int] = None
    retry_count: Optional[int] = None
    custom_body_params: Dict[str, Any] = Field(default_factory=dict)
    custom_headers: Dict[str, Any] = Field(default_factory=dict)

    model_config = {"extra": "allow"}


class ModelFallbackConfig(BaseModel):
    gateway_model_name: str
    fallback_models: List[FallbackModelRule]
    rotate_models: bool = False

    model_config = {"extra": "allow"}

    @field_validator("rotate_models", mode="before")
    @classmethod
    def _coerce_bool(cls, v: Any) -> Any:
        # reference accepts the string "true"/"false" (loader.py:52-56)
        if isinstance(v, str):
            return v.strip().lower() == "true"
        return v


def parse_providers(raw: Any) -> Dict[str, ProviderDetails]:
    """Validate the raw providers.json value (list of one-key dicts)."""
    if not isinstance(raw, list):
        raise ValueError("providers.json must contain a JSON list")
    out: Dict[str, ProviderDetails] = {}
    for item in raw:
        if not isinstance(item, dict) or len(item) != 1:
            raise ValueError(
                "Each provider entry must be a dictionary with a single key (the provider name)."
            )
        name, details = next(iter(item.items()))
        out[name] = ProviderDetails(**details)
    return out


def parse_fallback_rules(raw: Any) -> Dict[str, ModelFallbackConfig]:
    """Validate the raw models_fallback_rules.json value (list of rules).

    Later duplicates of a gateway_model_name override earlier ones, matching
    the reference's dict-build order.
    """
    if not isinstance(raw, list):
        raise ValueError("models_fallback_rules.json must contain a JSON list")
    out: Dict[str, ModelFallbackConfig] = {}
    for item in raw:
        rule = ModelFallbackConfig(**item)
        out[rule.gateway_model_name] = rule
    return out


def semantic_errors(
    providers: Dict[str, ProviderDetails],
    rules: Dict[str, ModelFallbackConfig],
    fallback_provider: Optional[str] = None,
) -> List[str]:
    """Cross-checks mirroring the reference (loader.py:102-122,284-314)."""
    errs: List[str] = []
    if fallback_provider and fallback_provider not in providers:
        errs.append(
            f"Fallback provider '{fallback_provider}' from settings is not defined in providers.json"
        )
    for gw_name, rule in rules.items():
        if not rule.fallback_models:
            errs.append(f"Rule '{gw_name}' must define at least one fallback model")
        for fm in rule.fallback_models:
            if fm.provider not in providers:
                errs.append(
                    f"Rule '{gw_name}' references provider '{fm.provider}' which is not defined in providers.json"
                )
    return errs


class ConfigLoader:
    """Loads, validates and hot-reloads the two config files."""

    def __init__(
        self,
        providers_path: str | os.PathLike = "providers.json",
        fallback_rules_path: str | os.PathLike = "models_fallback_rules.json",
        fallback_provider: Optional[str] = None,
        check_fallback_provider: bool = True,
    ):
        self.providers_path = Path(providers_path)
        self.fallback_rules_path = Path(fallback_rules_path)
        self.fallback_provider = fallback_provider
        self.check_fallback_provider = check_fallback_provider
        self.providers_config: Dict[str, ProviderDetails] = {}
        self.fallback_rules: Dict[str, ModelFallbackConfig] = {}

    # ---- initial (fatal) load ----
    def load(self) -> "ConfigLoader":
        self.providers_config = self._load_providers()
        self.fallback_rules = self._load_rules()
        errs = semantic_errors(
            self.providers_config,
            self.fallback_rules,
            self.fallback_provider if self.check_fallback_provider else None,
        )
        if errs:
            raise ConfigError("; ".join(errs))
        self._warn_missing_keys()
        logger.info(
            "Loaded %d providers, %d fallback rules",
            len(self.providers_config),
            len(self.fallback_rules),
        )
        return self

    def _load_providers(self) -> Dict[str, ProviderDetails]:
        if not self.providers_path.exists():
            raise ConfigError(f"Provider configuration file not found at {self.providers_path}")
        try:
            raw = jsonc.loads(self.providers_path.read_text(encoding="utf-8"))
            return parse_providers(raw)
        except (jsonc.JsoncError, ValueError, ValidationError) as e:
            raise ConfigError(f"Failed to load '{self.providers_path.name}': {e}") from e

    def _load_rules(self) -> Dict[str, ModelFallbackConfig]:
        if not self.fallback_rules_path.exists():
            raise ConfigError(
                f"Fallback rules configuration file not found at {self.fallback_rules_path}"
            )
        try:
            raw = jsonc.loads(self.fallback_rules_path.read_text(encoding="utf-8"))
            return parse_fallback_rules(raw)
        except (jsonc.JsoncError, ValueError, ValidationError) as e:
            raise ConfigError(f"Failed to load '{self.fallback_rules_path.name}': {e}") from e

    def _warn_missing_keys(self) -> None:
        for name, cfg in self.providers_config.items():
            if cfg.is_local or not cfg.apikey:
                continue
            if not os.getenv(cfg.apikey):
                logger.warning(
                    "Environment variable '%s' for provider '%s' is not set "
                    "(the literal value will be used as the API key).",
                    cfg.apikey,
                    name,
                )

    # ---- non-fatal reloads (editor API; reference loader.py:166-282) ----
    def reload_fallback_rules(self) -> Tuple[bool, List[str]]:
        try:
            raw = jsonc.loads(self.fallback_rules_path.read_text(encoding="utf-8"))
            new_rules = parse_fallback_rules(raw)
        except (OSError, jsonc.JsoncError, ValueError, ValidationError) as e:
            return False, [str(e)]
        errs = semantic_errors(self.providers_config, new_rules)
        if errs:
            return False, errs
        self.fallback_rules = new_rules
        logger.info("Reloaded %d fallback rules", len(new_rules))
        return True, []

    def reload_providers_config(self) -> Tuple[bool, List[str]]:
        try:
            raw = jsonc.loads(self.providers_path.read_text(encoding="utf-8"))
            new_providers = parse_providers(raw)
        except (OSError, jsonc.JsoncError, ValueError, ValidationError) as e:
            return False, [str(e)]
        errs = semantic_errors(
            new_providers,
            self.fallback_rules,
            self.fallback_provider if self.check_fallback_provider else None,
        )
        if errs:
            return False, errs
        self.providers_config = new_providers
        logger.info("Reloaded %d providers", len(new_providers))
        return True, []
