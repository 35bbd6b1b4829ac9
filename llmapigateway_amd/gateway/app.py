"""FastAPI app factory and wiring.

Parity: /root/reference/main.py — lifespan-managed config loader + usage DB
on app.state, middleware order CORS → RequestLogging → auth → chat
accounting (main.py:69-90; note ASGI middleware added last runs first, so
add order here is the reverse), /v1 routers, /static mount, / redirect to
the rules editor, /health.
"""

from __future__ import annotations

import logging
from contextlib import asynccontextmanager
from pathlib import Path
from typing import Optional

from fastapi import FastAPI, Response
from fastapi.middleware.cors import CORSMiddleware
from fastapi.responses import RedirectResponse
from fastapi.staticfiles import StaticFiles

from ..config.loader import ConfigLoader
from ..config.settings import Settings
from ..db.rotation import ModelRotationDB
from ..db.usage import TokensUsageDB
from .dispatch import Dispatcher
from .middleware import AuthMiddleware, ChatAccountingMiddleware, RequestLoggingMiddleware
from .routes import v1_router

logger = logging.getLogger(__name__)

STATIC_DIR = Path(__file__).resolve().parents[2] / "static"


def _needs_engines(config_loader: ConfigLoader) -> bool:
    return any(p.is_local for p in config_loader.providers_config.values())


def create_app(
    settings: Optional[Settings] = None,
    providers_path: str = "providers.json",
    fallback_rules_path: str = "models_fallback_rules.json",
    db_dir: str = "db",
    log_dir: str = "logs",
    config_loader: Optional[ConfigLoader] = None,
    engine_registry: Optional[object] = None,
) -> FastAPI:
    settings = settings or Settings.from_env()

    if config_loader is None:
        config_loader = ConfigLoader(
            providers_path,
            fallback_rules_path,
            fallback_provider=settings.fallback_provider,
            check_fallback_provider=False,  # a pure-local setup needn't define the HTTP fallback provider
        ).load()

    rotation_db = ModelRotationDB(Path(db_dir) / "llmgateway_rotation.db")
    usage_db = TokensUsageDB(Path(db_dir) / "tokens_usage.db")
    usage_db.cleanup_old_records(days=180)

    if engine_registry is None and _needs_engines(config_loader):
        from ..engine.registry import EngineRegistry

        engine_registry = EngineRegistry(settings=settings)

    dispatcher = Dispatcher(engine_registry=engine_registry)

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        logger.info("Gateway starting (providers=%s)", list(config_loader.providers_config))
        yield
        await dispatcher.aclose()
        rotation_db.close()
        usage_db.close()

    app = FastAPI(title="LLM API Gateway (MI355X)", lifespan=lifespan)
    app.state.settings = settings
    app.state.config_loader = config_loader
    app.state.rotation_db = rotation_db
    app.state.usage_db = usage_db
    app.state.dispatcher = dispatcher

    # middleware: outermost CORS, then request logging, auth, chat accounting
    # (add_middleware prepends, so add in reverse execution order)
    app.add_middleware(ChatAccountingMiddleware, settings=settings, usage_db=usage_db, log_dir=log_dir)
    app.add_middleware(AuthMiddleware, settings=settings)
    app.add_middleware(RequestLoggingMiddleware)
    app.add_middleware(
        CORSMiddleware,
        allow_origins=settings.cors_allow_origins,
        allow_credentials=True,
        allow_methods=["*"],
        allow_headers=["*"],
    )

    app.include_router(v1_router, prefix="/v1")

    if STATIC_DIR.exists():
        app.mount("/static", StaticFiles(directory=str(STATIC_DIR)), name="static")

    @app.get("/")
    async def root():
        return RedirectResponse(url="/v1/ui/rules-editor", status_code=307)

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    from .metrics import CONTENT_TYPE_LATEST, GatewayMetrics

    app.state.metrics = GatewayMetrics()

    @app.get("/metrics")
    async def metrics():
        m = app.state.metrics
        registry = getattr(getattr(app.state, "dispatcher", None), "engine_registry", None)
        m.refresh_engine_gauges(registry)
        if registry is not None:
            try:
                # counters must be monotonic: advance by the delta since
                # the last scrape, tracked on our side (no reliance on
                # prometheus-client private internals)
                last = getattr(app.state, "_metrics_last", None)
                if last is None:
                    last = app.state._metrics_last = {}
                gen = sum(r.get("decode_tokens", 0) for r in registry.stats())
                pre = sum(r.get("prefill_tokens", 0) for r in registry.stats())
                pc = sum(r.get("prefix_cached_tokens", 0) for r in registry.stats())
                for name, counter, total in (
                    ("gen", m.generated_tokens_total, gen),
                    ("pre", m.prefill_tokens_total, pre),
                    ("pc", m.prefix_cached_tokens_total, pc),
                ):
                    delta = total - last.get(name, 0)
                    if delta > 0:
                        counter.inc(delta)
                        last[name] = total
            except Exception:  # pragma: no cover
                pass
        return Response(content=m.render(), media_type=CONTENT_TYPE_LATEST)

    return app
