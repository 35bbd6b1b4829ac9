"""GET /v1/models (+ OpenCode / GitHub-Copilot exporters).

Parity: /root/reference/llm_gateway_core/api/v1/models.py:224-312 (model
list: gateway-rule models first, owned_by "llmgateway", then models fetched
live from the fallback provider) and the AsOpenCodeFormat /
AsGitHubCopilotFormat reshapes (models.py:89-222). Local engine providers
contribute their engine models instead of a remote /models fetch. Uses the
app's single ConfigLoader — the reference's import-time second loader
(models.py:14-16) made this endpoint see stale config after editor reloads.
"""

from __future__ import annotations

import logging
from typing import Any, Dict, List

import httpx
from fastapi import APIRouter, Request

logger = logging.getLogger(__name__)

router = APIRouter()

REASONING_VARIANTS = {
    "none": {"reasoningEffort": "none"},
    "minimal": {"reasoningEffort": "minimal"},
    "low": {"reasoningEffort": "low"},
    "medium": {"reasoningEffort": "medium"},
    "high": {"reasoningEffort": "high"},
    "xhigh": {"reasoningEffort": "xhigh"},
}


def _extract_modalities(model_info: Dict[str, Any]) -> Dict[str, List[str]]:
    arch = model_info.get("architecture")
    if isinstance(arch, dict):
        inp, out = arch.get("input_modalities"), arch.get("output_modalities")
        if isinstance(inp, list) and isinstance(out, list):
            seen, remapped = set(), []
            for m in inp:
                m = "pdf" if m == "file" else m  # OpenCode accepts pdf, not file
                if m not in seen:
                    seen.add(m)
                    remapped.append(m)
            return {"input": remapped, "output": out}
    return {"input": ["text", "image", "pdf"], "output": ["text"]}


def _extract_variants(model_info: Dict[str, Any]) -> Dict[str, Any]:
    sp = model_info.get("supported_parameters")
    if isinstance(sp, list):
        return dict(REASONING_VARIANTS) if "reasoning" in sp else {}
    return dict(REASONING_VARIANTS)


async def _fetch_provider_models(request: Request) -> List[Dict[str, Any]]:
    """Fetch live models from the fallback provider; degrade gracefully."""
    config_loader = request.app.state.config_loader
    settings = request.app.state.settings
    provider_cfg = config_loader.providers_config.get(settings.fallback_provider)
    if provider_cfg is None:
        return []
    if provider_cfg.is_local:
        registry = getattr(request.app.state.dispatcher, "engine_registry", None)
        if registry is None:
            return []
        return registry.list_models(provider_cfg.engine_spec())
    url = f"{provider_cfg.baseUrl.rstrip('/')}/models"
    try:
        # reuse the dispatcher's shared client; tighter timeout for this
        # endpoint (reference models.py:19 uses 60 s / 10 s)
        client = request.app.state.dispatcher.http._client
        resp = await client.get(url, timeout=httpx.Timeout(60.0, connect=10.0))
        resp.raise_for_status()
        data = resp.json().get("data", [])
        return data if isinstance(data, list) else []
    except Exception as e:
        logger.warning("Could not fetch models from %s: %s", url, e)
        return []


@router.get("")
async def get_models(request: Request):
    config_loader = request.app.state.config_loader
    settings = request.app.state.settings

    gateway_models: Dict[str, Dict[str, Any]] = {}
    for gw_name in config_loader.fallback_rules:
        gateway_models[gw_name] = {
            "id": gw_name,
            "object": "model",
            "owned_by": "llmgateway",
        }

    provider_models: List[Dict[str, Any]] = []
    for m in await _fetch_provider_models(request):
        mid = m.get("id")
        if not mid or mid in gateway_models:
            continue
        entry = dict(m)
        entry.setdefault("object", "model")
        entry["source_provider"] = settings.fallback_provider
        provider_models.append(entry)
    provider_models.sort(key=lambda m: m.get("id", ""))

    return {"object": "list", "data": list(gateway_models.values()) + provider_models}


@router.get("/AsOpenCodeFormat")
async def get_models_as_opencode(request: Request, includefallback: bool = False):
    config_loader = request.app.state.config_loader
    settings = request.app.state.settings
    models_data = await get_models(request)

    opencode_models: Dict[str, Any] = {}
    for info in models_data.get("data", []):
        mid = info.get("id")
        if not mid:
            continue
        if not includefallback and mid not in config_loader.fallback_rules:
            continue
        context_length, max_out = 200000, 32000
        top = info.get("top_provider", {}) or {}
        if top.get("context_length") is not None:
            context_length = top["context_length"]
        if top.get("max_completion_tokens") is not None:
            max_out = top["max_completion_tokens"]
        opencode_models[mid] = {
            "name": info.get("name", mid),
            "limit": {"context": context_length, "output": max_out},
            "modalities": _extract_modalities(info),
            "variants": _extract_variants(info),
        }

    api_key = settings.gateway_api_key or "12345678"
    return {
        "provider": {
            "llm-gateway-local": {
                "npm": "@ai-sdk/openai-compatible",
                "name": "LLM Gateway (local)",
                "options": {
                    "baseURL": f"http://localhost:{settings.gateway_port}/v1",
                    "apiKey": api_key,
                    "headers": {"Authorization": f"Bearer {api_key}"},
                },
                "models": opencode_models,
            }
        }
    }


@router.get("/AsGitHubCopilotFormat")
async def get_models_as_github_copilot(request: Request, includefallback: bool = False):
    config_loader = request.app.state.config_loader
    settings = request.app.state.settings
    models_data = await get_models(request)

    copilot_models: List[Dict[str, Any]] = []
    for info in models_data.get("data", []):
        mid = info.get("id")
        if not mid:
            continue
        is_gateway_model = mid in config_loader.fallback_rules
        if not includefallback and not is_gateway_model:
            continue

        arch = info.get("architecture", {}) or {}
        input_mods = arch.get("input_modalities", []) if isinstance(arch, dict) else []
        vision = isinstance(input_mods, list) and "image" in input_mods
        sp = info.get("supported_parameters", []) or []
        supports_reasoning = isinstance(sp, list) and "reasoning" in sp
        if is_gateway_model:
            vision = True
            supports_reasoning = True

        max_in, max_out = 400000, 60000
        top = info.get("top_provider", {}) or {}
        if top.get("context_length") is not None:
            max_in = top["context_length"]
        elif info.get("context_length") is not None:
            max_in = info["context_length"]
        if top.get("max_completion_tokens") is not None:
            max_out = top["max_completion_tokens"]

        entry = {
            "id": mid,
            "name": info.get("name", mid),
            "url": f"http://localhost:{settings.gateway_port}/v1/chat/completions",
            "toolCalling": True,
            "vision": vision,
            "maxInputTokens": max_in,
            "maxOutputTokens": max_out,
        }
        if supports_reasoning:
            entry["supportsReasoningEffort"] = ["none", "minimal", "low", "medium", "high", "xhigh"]
        copilot_models.append(entry)

    api_key = settings.gateway_api_key or "12345678"
    return {
        "name": "LLMGateway",
        "vendor": "customendpoint",
        "apiKey": api_key,
        "apiType": "chat-completions",
        "models": copilot_models,
    }
