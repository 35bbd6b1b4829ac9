"""A local mock OpenAI-compatible upstream used by gateway integration tests.

Behavior is selected by the requested model name:
- "ok"            -> non-streaming JSON or well-formed SSE stream with usage
- "http500"       -> HTTP 500
- "errbody"       -> 200 with {"error": ...} JSON body
- "errchunk"      -> SSE whose first real data frame carries an error
- "slowsplit"     -> SSE frames split across odd byte boundaries
- "flaky:<n>"     -> fails (500) the first n requests per model string, then ok
- "subonly:<p>"   -> succeeds only when payload provider.order == [p]
                     (OpenRouter sub-provider ordering); records orders seen
- "echo"          -> non-streaming; response content carries the request
                     body + selected headers (param-injection assertions)
"""

from __future__ import annotations

import json
from collections import defaultdict

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

FLAKY_COUNTS: defaultdict[str, int] = defaultdict(int)
SUB_ORDERS_SEEN: list = []


def make_mock_upstream() -> FastAPI:
    app = FastAPI()

    def sse(obj) -> bytes:
        return b"data: " + json.dumps(obj).encode() + b"\n\n"

    def ok_chunks(model: str):
        for i, piece in enumerate(["Hello", " from", " mock"]):
            yield sse(
                {
                    "id": "cmpl-1",
                    "object": "chat.completion.chunk",
                    "model": model,
                    "choices": [{"index": 0, "delta": {"content": piece}, "finish_reason": None}],
                }
            )
        yield sse(
            {
                "id": "cmpl-1",
                "object": "chat.completion.chunk",
                "model": model,
                "choices": [{"index": 0, "delta": {}, "finish_reason": "stop"}],
                "usage": {"prompt_tokens": 7, "completion_tokens": 3, "total_tokens": 10, "cost": 0.002},
            }
        )
        yield b"data: [DONE]\n\n"

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        body = await request.json()
        model = body.get("model", "ok")
        stream = bool(body.get("stream", False))

        if model == "echo":
            return JSONResponse(content={
                "id": "cmpl-echo", "object": "chat.completion", "model": model,
                "choices": [{"index": 0, "finish_reason": "stop",
                             "message": {"role": "assistant", "content": json.dumps({
                                 "body": body,
                                 "x_demo": request.headers.get("x-demo"),
                                 "auth": request.headers.get("authorization"),
                             })}}],
                "usage": {"prompt_tokens": 1, "completion_tokens": 1, "total_tokens": 2},
            })

        if model.startswith("subonly:"):
            want = model.split(":")[1]
            order = (body.get("provider") or {}).get("order")
            SUB_ORDERS_SEEN.append(order)
            if order != [want]:
                return JSONResponse(
                    status_code=502, content={"error": f"sub-provider {order} unavailable"}
                )
            model = "ok"

        if model.startswith("flaky:"):
            n = int(model.split(":")[1])
            FLAKY_COUNTS[model] += 1
            if FLAKY_COUNTS[model] <= n:
                return JSONResponse(status_code=500, content={"error": "flaky failure"})
            model = "ok"

        if model == "http500":
            return JSONResponse(status_code=500, content={"error": {"message": "upstream broke"}})
        if model == "errbody":
            return JSONResponse(content={"error": {"message": "bad model"}})

        if stream:
            if model == "errchunk":
                def err_gen():
                    yield b": keepalive\n\n"
                    yield sse({"error": {"message": "no capacity"}, "code": 429})

                return StreamingResponse(err_gen(), media_type="text/event-stream")
            if model == "slowsplit":
                whole = b"".join(ok_chunks(model))

                def split_gen():
                    # odd-sized chunks to exercise partial-frame reassembly
                    for i in range(0, len(whole), 7):
                        yield whole[i : i + 7]

                return StreamingResponse(split_gen(), media_type="text/event-stream")
            return StreamingResponse(ok_chunks(model), media_type="text/event-stream")

        return {
            "id": "cmpl-1",
            "object": "chat.completion",
            "model": model,
            "choices": [
                {"index": 0, "message": {"role": "assistant", "content": "Hello from mock"},
                 "finish_reason": "stop"}
            ],
            "usage": {"prompt_tokens": 7, "completion_tokens": 3, "total_tokens": 10},
        }

    @app.get("/v1/models")
    async def models():
        return {
            "object": "list",
            "data": [
                {"id": "mock-model-b", "object": "model"},
                {"id": "mock-model-a", "object": "model"},
            ],
        }

    return app
