"""Usage-stats API + dashboard page.

Parity: /root/reference/llm_gateway_core/api/v1/stats.py:15-83 — the
period-aggregated stats endpoint (hour/day/week/month with 24 h / 2 w /
15 w / 365 d lookbacks) and the paginated raw-records endpoint. Adds
/v1/api/engine-stats exposing local-engine counters (no reference
equivalent; the engines replace remote providers here).
"""

from __future__ import annotations

import logging
from datetime import datetime, timedelta
from pathlib import Path

from fastapi import APIRouter, HTTPException, Request
from fastapi.responses import FileResponse, JSONResponse

logger = logging.getLogger(__name__)

router = APIRouter()

STATIC_DIR = Path(__file__).resolve().parents[3] / "static"

_LOOKBACK = {
    "hour": timedelta(hours=24),
    "day": timedelta(weeks=2),
    "week": timedelta(weeks=15),
    "month": timedelta(days=365),
}


@router.get("/ui/usage-stats")
async def usage_stats_page():
    page = STATIC_DIR / "usage-stats.html"
    if not page.exists():
        raise HTTPException(status_code=404, detail="usage-stats.html not found")
    return FileResponse(page, media_type="text/html")


@router.get("/api/usage-stats/{period}")
async def get_aggregated_stats(request: Request, period: str):
    db = request.app.state.usage_db
    if period not in _LOOKBACK:
        raise HTTPException(
            status_code=400, detail="Invalid period. Must be 'hour', 'day', 'week', or 'month'."
        )
    end = datetime.now()
    rows = db.get_aggregated_usage(period, start_date=end - _LOOKBACK[period], end_date=end)
    return JSONResponse(content=rows)


@router.get("/api/usage-records")
async def get_usage_records(request: Request, limit: int = 25, offset: int = 0):
    db = request.app.state.usage_db
    return {
        "records": db.get_latest_usage_records(limit=limit, offset=offset),
        "total": db.get_total_records_count(),
        "limit": limit,
        "offset": offset,
    }


@router.get("/api/engine-stats")
async def get_engine_stats(request: Request):
    registry = getattr(request.app.state.dispatcher, "engine_registry", None)
    if registry is None:
        return {"engines": []}
    return {"engines": registry.stats()}
