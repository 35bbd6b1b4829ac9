"""ASGI middleware stack: auth, request logging, chat accounting.

Parity targets (all pure-ASGI here; the reference mixes styles):
- auth: Bearer check of Authorization vs the configured gateway key
  (/root/reference/llm_gateway_core/middleware/auth.py:29-42). The
  reference's skip condition tests ``endswith("/chat/completion")`` —
  singular — so auth is never actually enforced (auth.py:17); we implement
  the documented intent and enforce it on /v1/chat/completions.
- request logging: UUID request id, /health skipped, sensitive headers
  masked, x-request-id response header, duration log
  (/root/reference/llm_gateway_core/middleware/request_logging.py:17-90).
- chat accounting: tee the /v1/chat/completions response (streamed or not),
  accumulate content and usage, write a transcript file and a usage-DB row
  (/root/reference/llm_gateway_core/middleware/chat_logging.py:69-272 —
  there a queue-fed thread; here an asyncio background task).
"""

from __future__ import annotations

import asyncio
import json
import logging
import os
import time
import uuid
from typing import Any, Dict, List, Optional

from ..config.settings import Settings
from ..db.usage import TokensUsageDB
from . import sse

logger = logging.getLogger(__name__)

SENSITIVE_HEADERS = {"authorization", "x-api-key", "api-key", "cookie"}
CHAT_PATH_SUFFIX = "/chat/completions"


def _mask_headers(headers: List[tuple]) -> Dict[str, str]:
    out = {}
    for k, v in headers:
        key = k.decode("latin-1") if isinstance(k, bytes) else k
        val = v.decode("latin-1") if isinstance(v, bytes) else v
        if key.lower() in SENSITIVE_HEADERS:
            val = val[:12] + "***" if len(val) > 12 else "***"
        out[key] = val
    return out


class AuthMiddleware:
    """Enforce Bearer auth on /v1/chat/completions (documented intent of the
    reference, README.md:108-110; fixes auth.py:17)."""

    def __init__(self, app, settings: Settings):
        self.app = app
        self.settings = settings

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http":
            return await self.app(scope, receive, send)
        path = scope.get("path", "")
        api_key = self.settings.gateway_api_key
        if api_key and path.endswith(CHAT_PATH_SUFFIX):
            auth = ""
            for k, v in scope.get("headers", []):
                if k == b"authorization":
                    auth = v.decode("latin-1")
                    break
            token = auth[7:] if auth.startswith("Bearer ") else auth
            if token != api_key:
                body = json.dumps({"detail": "Invalid or missing API key"}).encode()
                await send(
                    {
                        "type": "http.response.start",
                        "status": 401,
                        "headers": [
                            (b"content-type", b"application/json"),
                            (b"content-length", str(len(body)).encode()),
                        ],
                    }
                )
                await send({"type": "http.response.body", "body": body})
                return
        await self.app(scope, receive, send)


class RequestLoggingMiddleware:
    def __init__(self, app):
        self.app = app

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http" or scope.get("path") == "/health":
            return await self.app(scope, receive, send)

        request_id = str(uuid.uuid4())
        scope.setdefault("state", {})["request_id"] = request_id
        start = time.perf_counter()
        method, path = scope.get("method", ""), scope.get("path", "")

        # POST chat payload summary with messages/tools excluded (the
        # reference logs the same shape: request_logging.py:49-61). The
        # body is buffered and replayed through a wrapped `receive`.
        summary = None
        if method == "POST" and path.endswith("/chat/completions"):
            body = b""
            more = True
            while more:
                message = await receive()
                body += message.get("body", b"")
                more = message.get("more_body", False)
            try:
                payload = json.loads(body or b"{}")
                if isinstance(payload, dict):
                    summary = {
                        k: v for k, v in payload.items()
                        if k not in ("messages", "tools")
                    }
            except ValueError:
                summary = "<unparseable>"
            replayed = {"done": False}
            orig_receive = receive

            async def receive():  # noqa: F811 — replay the buffered body
                if not replayed["done"]:
                    replayed["done"] = True
                    return {"type": "http.request", "body": body, "more_body": False}
                # afterwards delegate (disconnect detection during SSE)
                return await orig_receive()

        logger.info(
            "rid=%s --> %s %s headers=%s%s",
            request_id,
            method,
            path,
            _mask_headers(scope.get("headers", [])),
            f" payload={summary}" if summary is not None else "",
        )

        status_holder = {"status": 0}

        async def send_wrapper(message):
            if message["type"] == "http.response.start":
                status_holder["status"] = message["status"]
                headers = list(message.get("headers", []))
                headers.append((b"x-request-id", request_id.encode()))
                message = {**message, "headers": headers}
            await send(message)

        try:
            await self.app(scope, receive, send_wrapper)
        finally:
            dur_ms = (time.perf_counter() - start) * 1000.0
            logger.info(
                "rid=%s <-- %s %s %d (%.1f ms)",
                request_id,
                method,
                path,
                status_holder["status"],
                dur_ms,
            )


class ChatAccountingMiddleware:
    """Tee /v1/chat/completions responses for usage accounting + transcripts."""

    def __init__(
        self,
        app,
        settings: Settings,
        usage_db: Optional[TokensUsageDB] = None,
        log_dir: str = "logs",
    ):
        self.app = app
        self.settings = settings
        self.usage_db = usage_db
        self.log_dir = log_dir

    async def __call__(self, scope, receive, send):
        if (
            scope["type"] != "http"
            or scope.get("method") != "POST"
            or not scope.get("path", "").endswith(CHAT_PATH_SUFFIX)
        ):
            return await self.app(scope, receive, send)

        state = scope.setdefault("state", {})
        req_chunks: List[bytes] = []
        request_done = {"flag": False}

        async def receive_wrapper():
            message = await receive()
            if message["type"] == "http.request":
                req_chunks.append(message.get("body", b""))
                if not message.get("more_body", False):
                    request_done["flag"] = True
            return message

        resp_info: Dict[str, Any] = {"status": 0, "streaming": False}
        parser = sse.SSEParser()
        sniffer = sse.StreamSniffer()
        nonstream_body: List[bytes] = []

        async def send_wrapper(message):
            if message["type"] == "http.response.start":
                resp_info["status"] = message["status"]
                for k, v in message.get("headers", []):
                    if k == b"content-type" and b"text/event-stream" in v:
                        resp_info["streaming"] = True
            elif message["type"] == "http.response.body":
                body = message.get("body", b"")
                if body:
                    if resp_info["streaming"]:
                        for frame in parser.feed(body):
                            sniffer.observe(frame)
                    else:
                        nonstream_body.append(body)
                if not message.get("more_body", False):
                    self._finalize(state, req_chunks, resp_info, sniffer, nonstream_body)
            await send(message)

        await self.app(scope, receive_wrapper, send_wrapper)

    def _finalize(self, state, req_chunks, resp_info, sniffer, nonstream_body) -> None:
        try:
            if resp_info["status"] >= 400:
                return
            if not resp_info["streaming"]:
                try:
                    obj = json.loads(b"".join(nonstream_body))
                except Exception:
                    return
                if isinstance(obj.get("usage"), dict):
                    sniffer.usage = obj["usage"]
                if obj.get("model"):
                    sniffer.model = obj["model"]
                for choice in obj.get("choices") or []:
                    msg = (choice or {}).get("message") or {}
                    if msg.get("content"):
                        sniffer.content.append(msg["content"])

            provider = state.get("served_provider")
            model = state.get("served_model") or sniffer.model
            if sniffer.error_detail:  # mid-stream error chunk: log-only,
                # bytes already reached the client (request_handler.py:125-133)
                logger.warning(
                    "Error chunk mid-stream from provider %s: %s",
                    provider, str(sniffer.error_detail)[:500],
                )
            if sniffer.usage and self.usage_db is not None:
                fields = sse.token_usage_fields(sniffer.usage)
                self.usage_db.insert_usage(model=model, provider=provider, **fields)

            if self.settings.log_chat_messages:
                self._write_transcript(req_chunks, sniffer, provider, model)
        except Exception:
            logger.exception("chat accounting failed")

    def _write_transcript(self, req_chunks, sniffer, provider, model) -> None:
        os.makedirs(self.log_dir, exist_ok=True)
        ts = time.strftime("%Y%m%d_%H%M%S") + f"_{int((time.time() % 1) * 1e6):06d}"
        path = os.path.join(self.log_dir, f"{ts}.txt")
        try:
            req_text = b"".join(req_chunks).decode("utf-8", errors="replace")
        except Exception:
            req_text = "<unreadable>"
        with open(path, "w", encoding="utf-8") as f:
            f.write(f"provider: {provider}\nmodel: {model}\n\n=== REQUEST ===\n{req_text}\n")
            f.write(f"\n=== RESPONSE ===\n{sniffer.full_content()}\n")
            if sniffer.usage:
                f.write(f"\n=== USAGE ===\n{json.dumps(sniffer.usage, indent=2)}\n")
        self._prune_logs()

    def _prune_logs(self) -> None:
        limit = self.settings.log_file_limit
        try:
            files = sorted(
                f for f in os.listdir(self.log_dir) if f.endswith(".txt")
            )
            for old in files[: max(0, len(files) - limit)]:
                os.remove(os.path.join(self.log_dir, old))
        except OSError:
            pass
