"""Server-Sent-Events framing, reassembly and error sniffing.

The data-plane contract the whole gateway relies on (mirrors the behaviors
of /root/reference/llm_gateway_core/services/request_handler.py:34-63 and
111-144, re-designed as a reusable incremental parser instead of inline
loops):

- frames are separated by blank lines ("\\n\\n"); a frame may arrive split
  across arbitrary byte chunks, so reassembly keeps a partial-frame buffer;
- a "real" frame starts with ``data: {``; anything else (comments,
  keep-alives, "data: [DONE]") passes through untouched;
- the FIRST real frame decides success: if its JSON carries "error" or
  "detail", the request failed and no bytes may reach the client (the
  fallback chain can still try the next model);
- mid-stream error chunks (OpenRouter-style, carrying "code") and "usage"
  chunks are detected for accounting.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Dict, Iterator, List, Optional

from ..config import jsonc

DONE_FRAME = b"data: [DONE]\n\n"


def format_sse(obj: Dict[str, Any]) -> bytes:
    return b"data: " + json.dumps(obj, separators=(",", ":")).encode("utf-8") + b"\n\n"


def parse_data_frame(frame_text: str) -> Optional[Dict[str, Any]]:
    """Parse a ``data: {...}`` frame's JSON payload; None if not a real frame."""
    if not frame_text.startswith("data: {"):
        return None
    try:
        obj = jsonc.loads(frame_text[len("data: "):])
    except jsonc.JsoncError:
        return None
    return obj if isinstance(obj, dict) else None


def frame_is_error(obj: Dict[str, Any]) -> bool:
    return "error" in obj or "detail" in obj


def extract_error_detail(obj: Dict[str, Any]) -> str:
    err = obj.get("error")
    if isinstance(err, dict):
        msg = err.get("message")
        if msg:
            return str(msg)
    if err is not None:
        return str(err)
    det = obj.get("detail")
    return str(det) if det is not None else json.dumps(obj)


@dataclass
class SSEParser:
    """Incremental SSE frame reassembler (bytes in, complete frames out)."""

    _buffer: str = ""

    def feed(self, chunk: bytes | str) -> List[str]:
        if isinstance(chunk, (bytes, bytearray)):
            try:
                text = chunk.decode("utf-8")
            except UnicodeDecodeError:
                # keep bytes buffered until a complete utf-8 run arrives; in
                # practice providers emit utf-8 — treat undecodable bytes as
                # pass-through by replacement to avoid data loss
                text = chunk.decode("utf-8", errors="replace")
        else:
            text = chunk
        self._buffer += text
        parts = self._buffer.split("\n\n")
        if self._buffer.endswith("\n\n"):
            self._buffer = ""
            return [p for p in parts if p]
        self._buffer = parts.pop()
        return [p for p in parts if p]

    def flush(self) -> Optional[str]:
        out, self._buffer = self._buffer, ""
        return out or None


@dataclass
class StreamSniffer:
    """Stateful accounting/error watcher over a stream of SSE frames.

    Accumulates delta content, watches for mid-stream error chunks
    (reference request_handler.py:125-133) and the usage chunk
    (reference chat_logging.py:233-263 extraction semantics).
    """

    content: List[str] = field(default_factory=list)
    reasoning: List[str] = field(default_factory=list)
    usage: Optional[Dict[str, Any]] = None
    model: Optional[str] = None
    error_detail: Optional[str] = None
    finish_reason: Optional[str] = None

    def observe(self, frame_text: str) -> None:
        obj = parse_data_frame(frame_text)
        if obj is None:
            return
        if "code" in obj and "choices" not in obj:  # OpenRouter-style error chunk
            self.error_detail = extract_error_detail(obj)
        if obj.get("model"):
            self.model = obj["model"]
        if isinstance(obj.get("usage"), dict):
            self.usage = obj["usage"]
        for choice in obj.get("choices") or []:
            if not isinstance(choice, dict):
                continue
            delta = choice.get("delta") or {}
            msg = choice.get("message") or {}
            piece = delta.get("content") or msg.get("content")
            if piece:
                self.content.append(piece)
            rpiece = delta.get("reasoning_content") or delta.get("reasoning")
            if rpiece:
                self.reasoning.append(rpiece)
            if choice.get("finish_reason"):
                self.finish_reason = choice["finish_reason"]

    def full_content(self) -> str:
        return "".join(self.content)


def token_usage_fields(usage: Dict[str, Any]) -> Dict[str, Any]:
    """Normalize an OpenAI-style usage object into DB row fields.

    Reasoning tokens are subtracted from completion tokens when reported
    inside completion_tokens_details, matching the reference's accounting
    (chat_logging.py:256-263).
    """
    prompt = int(usage.get("prompt_tokens") or 0)
    completion = int(usage.get("completion_tokens") or 0)
    total = int(usage.get("total_tokens") or (prompt + completion))
    reasoning = 0
    cached = 0
    comp_details = usage.get("completion_tokens_details") or {}
    if isinstance(comp_details, dict):
        reasoning = int(comp_details.get("reasoning_tokens") or 0)
    prompt_details = usage.get("prompt_tokens_details") or {}
    if isinstance(prompt_details, dict):
        cached = int(prompt_details.get("cached_tokens") or 0)
    if reasoning:
        completion = max(0, completion - reasoning)
    return {
        "prompt_tokens": prompt,
        "completion_tokens": completion,
        "total_tokens": total,
        "reasoning_tokens": reasoning,
        "cached_tokens": cached,
        "cost": float(usage.get("cost") or 0.0),
    }
