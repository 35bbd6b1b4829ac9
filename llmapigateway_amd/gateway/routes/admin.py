"""Admin API: runtime failure injection on local engines.

No reference equivalent (the reference author toggled failures by editing
commented-out lines, chat.py:143-144); here fault injection for fallback
testing (BASELINE configs[2]) is endpoint-driven.
"""

from __future__ import annotations

from typing import Optional

from fastapi import APIRouter, HTTPException, Request
from pydantic import BaseModel

router = APIRouter()


class FailureSpec(BaseModel):
    fail_rate: float = 0.0
    fail_requests: Optional[int] = None


@router.post("/admin/engines/{provider}/failures")
async def set_engine_failures(request: Request, provider: str, spec: FailureSpec):
    registry = getattr(request.app.state.dispatcher, "engine_registry", None)
    if registry is None:
        raise HTTPException(status_code=400, detail="No local engines are configured")
    registry.set_failure(provider, fail_rate=spec.fail_rate, fail_requests=spec.fail_requests)
    return {"status": "ok", "provider": provider, **spec.model_dump()}
