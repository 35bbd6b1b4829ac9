"""Downstream request dispatch: HTTP upstreams and local MI355X engines.

This is the rebuild of the reference's data-plane service
(/root/reference/llm_gateway_core/services/request_handler.py:8-189): turn a
(provider, model, payload) tuple into an (SSE stream | JSON dict | None,
error_detail) pair, where *failure before any client byte* lets the caller
fall back to the next model. Two backends share those semantics:

- HTTP proxy (httpx) for remote providers — same 300 s/60 s timeouts,
  status>=400 / error-body / first-SSE-chunk error detection;
- the local engine registry (engine/registry.py) for ``local://`` providers,
  where "upstream failure" is an injected fault, OOM, or engine error raised
  before the first token is yielded.
"""

from __future__ import annotations

import logging
from typing import Any, AsyncIterator, Dict, Optional, Tuple, Union

import httpx
from fastapi.responses import StreamingResponse

from ..config.loader import ProviderDetails
from . import sse

logger = logging.getLogger(__name__)

RequestResult = Tuple[Optional[Union[Dict[str, Any], StreamingResponse]], Optional[str]]

STREAM_HEADERS = {"X-Accel-Buffering": "no"}


class HttpDispatcher:
    """Proxy-mode backend: forwards to an OpenAI-compatible HTTP upstream."""

    def __init__(self, timeout_total: float = 300.0, timeout_connect: float = 60.0):
        self._client = httpx.AsyncClient(
            timeout=httpx.Timeout(timeout_total, connect=timeout_connect)
        )

    async def aclose(self) -> None:
        await self._client.aclose()

    async def make_request(
        self,
        target_url: str,
        headers: Dict[str, str],
        payload: Dict[str, Any],
        is_streaming: bool,
    ) -> RequestResult:
        try:
            if is_streaming:
                return await self._make_streaming(target_url, headers, payload)
            return await self._make_plain(target_url, headers, payload)
        except httpx.RequestError as e:
            detail = f"RequestError connecting to {target_url}: {e}"
            logger.error(detail)
            return None, detail
        except Exception as e:  # defensive: never crash the fallback loop
            detail = f"Unexpected error during request to {target_url}: {e}"
            logger.exception(detail)
            return None, detail

    async def _make_plain(
        self, url: str, headers: Dict[str, str], payload: Dict[str, Any]
    ) -> RequestResult:
        resp = await self._client.post(url, headers=headers, json=payload)
        if resp.status_code >= 400:
            return None, resp.text
        try:
            body = resp.json()
        except Exception:
            return None, f"Invalid JSON response from {url}: {resp.text[:1000]}"
        if isinstance(body, dict) and sse.frame_is_error(body):
            return None, sse.extract_error_detail(body)
        return body, None

    async def _make_streaming(
        self, url: str, headers: Dict[str, str], payload: Dict[str, Any]
    ) -> RequestResult:
        """Open the upstream stream and *prime* it: consume until the first
        real ``data: {`` frame. If that frame is an error, abort with zero
        bytes surfaced so the caller can fall back."""
        req = self._client.build_request("POST", url, headers=headers, json=payload)
        resp = await self._client.send(req, stream=True)

        if resp.status_code >= 400:
            body = await resp.aread()
            await resp.aclose()
            return None, body.decode("utf-8", errors="replace")

        parser = sse.SSEParser()
        prologue: list[bytes] = []
        first_error: Optional[str] = None
        found_real = False
        aiter = resp.aiter_bytes()

        try:
            async for chunk in aiter:
                if not chunk:
                    continue
                prologue.append(chunk)
                for frame in parser.feed(chunk):
                    obj = sse.parse_data_frame(frame)
                    if obj is None:
                        continue
                    found_real = True
                    if sse.frame_is_error(obj):
                        first_error = frame
                    break
                if found_real:
                    break
        except Exception as e:
            await resp.aclose()
            return None, f"Stream error from {url} before first chunk: {e}"

        if first_error is not None:
            await resp.aclose()
            logger.warning("Error in first stream chunk from %s: %s", url, first_error[:500])
            return None, first_error
        if not found_real and not prologue:
            await resp.aclose()
            return None, f"Empty stream from {url}"

        async def passthrough() -> AsyncIterator[bytes]:
            try:
                for chunk in prologue:
                    yield chunk
                async for chunk in aiter:
                    if chunk:
                        yield chunk
            finally:
                await resp.aclose()

        return (
            StreamingResponse(
                passthrough(), media_type="text/event-stream", headers=dict(STREAM_HEADERS)
            ),
            None,
        )


class Dispatcher:
    """Routes a provider entry to the right backend (HTTP vs local engine)."""

    def __init__(self, engine_registry: Optional[Any] = None):
        self.http = HttpDispatcher()
        self.engine_registry = engine_registry

    async def aclose(self) -> None:
        await self.http.aclose()
        if self.engine_registry is not None:
            await self.engine_registry.aclose()

    async def make_request(
        self,
        provider_name: str,
        provider_cfg: ProviderDetails,
        headers: Dict[str, str],
        payload: Dict[str, Any],
        is_streaming: bool,
    ) -> RequestResult:
        if provider_cfg.is_local:
            if self.engine_registry is None:
                return None, f"Provider '{provider_name}' is local but no engine registry is running"
            return await self.engine_registry.make_request(
                provider_name, provider_cfg.engine_spec(), payload, is_streaming
            )
        target_url = f"{provider_cfg.baseUrl.rstrip('/')}/chat/completions"
        return await self.http.make_request(target_url, headers, payload, is_streaming)
