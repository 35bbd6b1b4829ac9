from fastapi import APIRouter

from .admin import router as admin_router
from .chat import router as chat_router
from .models import router as models_router
from .rules_editor import router as rules_editor_router
from .stats import router as stats_router

# aggregation mirrors the reference (api/v1/__init__.py:9-11): chat under
# /chat, models under /models, editor+stats mounted at the v1 root
v1_router = APIRouter()
v1_router.include_router(chat_router, prefix="/chat", tags=["Chat"])
v1_router.include_router(models_router, prefix="/models", tags=["Models"])
v1_router.include_router(rules_editor_router, tags=["Config Editor"])
v1_router.include_router(stats_router, tags=["Usage Stats"])
v1_router.include_router(admin_router, tags=["Admin"])

__all__ = ["v1_router"]
