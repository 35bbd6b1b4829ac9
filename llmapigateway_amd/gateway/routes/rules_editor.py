"""Config editor API + rules-editor UI page.

Parity: /root/reference/llm_gateway_core/api/v1/rules_editor.py:27-163 —
GET/POST of the raw config text (comments preserved on disk), validation
before write, live reload after write, 400 with structured errors on
validation failure, 500 if saved but reload failed.
"""

from __future__ import annotations

import logging
from pathlib import Path

from fastapi import APIRouter, HTTPException, Request
from fastapi.responses import FileResponse, JSONResponse, PlainTextResponse
from pydantic import ValidationError

from ...config import jsonc
from ...config.loader import parse_fallback_rules, parse_providers, semantic_errors

logger = logging.getLogger(__name__)

router = APIRouter()

STATIC_DIR = Path(__file__).resolve().parents[3] / "static"


@router.get("/ui/rules-editor")
async def rules_editor_page():
    page = STATIC_DIR / "rules-editor.html"
    if not page.exists():
        raise HTTPException(status_code=404, detail="rules-editor.html not found")
    return FileResponse(page, media_type="text/html")


def _validation_error_payload(e: Exception):
    if isinstance(e, ValidationError):
        return [
            {"loc": list(err.get("loc", [])), "msg": err.get("msg"), "type": err.get("type")}
            for err in e.errors()
        ]
    return [{"loc": [], "msg": str(e), "type": "value_error"}]


@router.get("/config/models-rules", response_class=PlainTextResponse)
async def get_models_rules(request: Request):
    path = request.app.state.config_loader.fallback_rules_path
    if not path.exists():
        raise HTTPException(status_code=404, detail=f"{path} not found")
    return path.read_text(encoding="utf-8")


@router.post("/config/models-rules")
async def save_models_rules(request: Request):
    config_loader = request.app.state.config_loader
    raw_text = (await request.body()).decode("utf-8")
    try:
        raw = jsonc.loads(raw_text)
        if not isinstance(raw, list):
            raise ValueError("Rules config must be a JSON list")
        rules = parse_fallback_rules(raw)
    except (jsonc.JsoncError, ValueError, ValidationError) as e:
        return JSONResponse(status_code=400, content={"detail": _validation_error_payload(e)})
    errs = semantic_errors(config_loader.providers_config, rules)
    if errs:
        return JSONResponse(
            status_code=400,
            content={"detail": [{"loc": [], "msg": m, "type": "semantic"} for m in errs]},
        )
    config_loader.fallback_rules_path.write_text(raw_text, encoding="utf-8")
    ok, reload_errs = config_loader.reload_fallback_rules()
    if not ok:
        raise HTTPException(
            status_code=500, detail=f"Saved but reload failed: {'; '.join(reload_errs)}"
        )
    return {"status": "ok", "rules": len(config_loader.fallback_rules)}


@router.get("/config/providers", response_class=PlainTextResponse)
async def get_providers(request: Request):
    path = request.app.state.config_loader.providers_path
    if not path.exists():
        raise HTTPException(status_code=404, detail=f"{path} not found")
    return path.read_text(encoding="utf-8")


@router.post("/config/providers")
async def save_providers(request: Request):
    config_loader = request.app.state.config_loader
    raw_text = (await request.body()).decode("utf-8")
    try:
        raw = jsonc.loads(raw_text)
        if not isinstance(raw, list):
            raise ValueError("Providers config must be a JSON list")
        providers = parse_providers(raw)
    except (jsonc.JsoncError, ValueError, ValidationError) as e:
        return JSONResponse(status_code=400, content={"detail": _validation_error_payload(e)})
    errs = semantic_errors(
        providers,
        config_loader.fallback_rules,
        config_loader.fallback_provider if config_loader.check_fallback_provider else None,
    )
    if errs:
        return JSONResponse(
            status_code=400,
            content={"detail": [{"loc": [], "msg": m, "type": "semantic"} for m in errs]},
        )
    config_loader.providers_path.write_text(raw_text, encoding="utf-8")
    ok, reload_errs = config_loader.reload_providers_config()
    if not ok:
        raise HTTPException(
            status_code=500, detail=f"Saved but reload failed: {'; '.join(reload_errs)}"
        )
    # release engines whose provider entry no longer references them — a
    # removed/changed local provider would otherwise keep its GPU-resident
    # weights (up to 141 GB) alive forever
    registry = getattr(request.app.state.dispatcher, "engine_registry", None)
    if registry is not None:
        registry.prune(
            [p.engine_spec() for p in config_loader.providers_config.values() if p.is_local]
        )
    return {"status": "ok", "providers": len(config_loader.providers_config)}
