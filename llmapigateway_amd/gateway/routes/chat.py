"""POST /v1/chat/completions — the fallback / rotation / retry orchestrator.

Control-plane parity with the reference's chat router
(/root/reference/llm_gateway_core/api/v1/chat.py:21-198):

- rule lookup by requested model; no rule → a synthesized single-entry
  sequence on the settings fallback provider (chat.py:48-59);
- rotation keyed by (api_key, gateway_model), advanced per request at
  request start regardless of outcome (chat.py:64-78);
- per-entry retries: ``while retry_count >= 0`` with a sleep only when
  0 < retry_delay < 120 and retries remain (chat.py:127,191-194);
- OpenRouter sub-provider ordering, either as a request hint
  (payload.provider = {order: [...]}, allow_fallbacks=False) or — with
  use_provider_order_as_fallback — one attempt per sub-provider
  (chat.py:129-189);
- custom_body_params / custom_headers injection (chat.py:116-123);
- terminal HTTP 503 carrying the last error detail (chat.py:196-198).

The difference from the reference is *what an attempt is*: Dispatcher routes
a local provider to a GPU-resident engine instead of an HTTP upstream.
"""

from __future__ import annotations

import asyncio
import copy
import logging
import os
import time
from typing import Any, Dict

from fastapi import APIRouter, HTTPException, Request

from ...config import jsonc
from ...config.loader import FallbackModelRule

logger = logging.getLogger(__name__)

router = APIRouter()

GATEWAY_REFERER = "https://github.com/llmapigateway-amd"
GATEWAY_TITLE = "LLMGateway-AMD"


def _resolve_api_key(apikey_field: str) -> str:
    """Env-var name first; the literal value as fallback (chat.py:93-101)."""
    if not apikey_field:
        return ""
    return os.getenv(apikey_field) or apikey_field


@router.post("/completions")
async def chat_completions(request: Request):
    app = request.app
    config_loader = app.state.config_loader
    dispatcher = app.state.dispatcher
    rotation_db = app.state.rotation_db
    settings = app.state.settings

    try:
        body_bytes = await request.body()
        body = jsonc.loads(body_bytes.decode("utf-8"))
        if not isinstance(body, dict):
            raise ValueError("request body must be a JSON object")
    except Exception as e:
        raise HTTPException(status_code=400, detail=f"Error reading request body: {e}")

    requested_model = body.get("model")
    if not requested_model:
        raise HTTPException(status_code=400, detail="Missing 'model' in request body")
    is_streaming = bool(body.get("stream", False))

    providers_config = config_loader.providers_config
    rule = config_loader.fallback_rules.get(requested_model)
    if rule is None:
        logger.warning(
            "No fallback rule for model '%s'; using fallback provider '%s'",
            requested_model,
            settings.fallback_provider,
        )
        sequence = [FallbackModelRule(provider=settings.fallback_provider, model=requested_model)]
        rotate = False
    else:
        sequence = list(rule.fallback_models)
        rotate = rule.rotate_models

    api_key = request.headers.get("Authorization", "").replace("Bearer ", "")

    # rotation advances per request at request start (chat.py:64-78)
    if rotate and len(sequence) > 1:
        start = rotation_db.get_next_model_index(
            api_key=api_key, gateway_model=requested_model, total_models=len(sequence)
        )
        sequence = sequence[start:] + sequence[:start]
        logger.info("Rotation: starting at index %d for '%s'", start, requested_model)

    metrics = getattr(request.app.state, "metrics", None)
    t_request = time.monotonic()
    had_fallback = False
    last_error = "No providers were attempted."
    for entry in sequence:
        provider_name = entry.provider
        provider_model = entry.model
        provider_cfg = providers_config.get(provider_name)
        if provider_cfg is None:
            last_error = f"Provider '{provider_name}' is not configured"
            logger.warning(last_error)
            continue

        key = _resolve_api_key(provider_cfg.apikey)
        headers = {
            "Content-Type": "application/json",
            "HTTP-Referer": GATEWAY_REFERER,
            "X-Title": GATEWAY_TITLE,
            **({"Authorization": f"Bearer {key}"} if key else {}),
            **{str(k): str(v) for k, v in (entry.custom_headers or {}).items()},
        }

        payload: Dict[str, Any] = copy.deepcopy(body)
        payload["model"] = provider_model
        if provider_name == "openrouter" and "usage" not in payload:
            payload["usage"] = {"include": True}
        for k, v in (entry.custom_body_params or {}).items():
            payload[k] = v

        sub_order = entry.providers_order or []
        retry_count = entry.retry_count or 0
        retry_delay = entry.retry_delay or 0

        while retry_count >= 0:
            attempts: list[Dict[str, Any]]
            if sub_order and entry.use_provider_order_as_fallback:
                # one attempt per sub-provider (chat.py:159-189)
                attempts = [
                    {**payload, "provider": {"order": [sub]}, "allow_fallbacks": False}
                    for sub in sub_order
                ]
            elif sub_order:
                attempts = [
                    {**payload, "provider": {"order": list(sub_order)}, "allow_fallbacks": False}
                ]
            else:
                attempts = [payload]

            for attempt_payload in attempts:
                result, error = await dispatcher.make_request(
                    provider_name, provider_cfg, headers, attempt_payload, is_streaming
                )
                if result is not None and error is None:
                    state = request.scope.setdefault("state", {})
                    state["served_provider"] = provider_name
                    state["served_model"] = provider_model
                    if metrics is not None:
                        metrics.requests_total.labels(
                            status="fallback_success" if had_fallback else "success"
                        ).inc()
                        # for local engines make_request returns after the
                        # first token; for proxies after stream priming —
                        # either way this is time-to-first-byte
                        metrics.ttft_seconds.observe(time.monotonic() - t_request)
                    logger.info(
                        "Success: model '%s' on provider '%s'%s",
                        provider_model,
                        provider_name,
                        " (streaming)" if is_streaming else "",
                    )
                    return result
                last_error = (
                    f"Model {provider_model} failed with provider '{provider_name}': {error}"
                )
                had_fallback = True
                if metrics is not None:
                    metrics.fallback_attempts_total.labels(provider=provider_name).inc()
                logger.warning(last_error)

            if retry_count > 0 and 0 < retry_delay < 120:
                logger.info(
                    "RETRYING %s in %ds... %d attempts left",
                    provider_model,
                    retry_delay,
                    retry_count - 1,
                )
                await asyncio.sleep(retry_delay)
            retry_count -= 1

    if metrics is not None:
        metrics.requests_total.labels(status="error").inc()
    logger.error("All providers failed for '%s'. Last error: %s", requested_model, last_error)
    raise HTTPException(
        status_code=503,
        detail=f"All configured providers failed for model '{requested_model}'. Last error: {last_error}",
    )
