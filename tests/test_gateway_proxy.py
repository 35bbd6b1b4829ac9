"""Gateway integration tests in HTTP-proxy mode (BASELINE configs[0]):
2-model fallback, retries, rotation, auth, usage accounting — against the
local mock OpenAI upstream in tests/mock_upstream.py."""

import json
import time

import httpx
import pytest
from fastapi.testclient import TestClient

from llmapigateway_amd.config.settings import Settings
from llmapigateway_amd.gateway.app import create_app
from tests.mock_upstream import FLAKY_COUNTS, make_mock_upstream

PROVIDERS = """
[
    { "mock": { "baseUrl": "http://mock.test/v1", "apikey": "MOCK_KEY" } },
    { "mock2": { "baseUrl": "http://mock.test/v1", "apikey": "" } }
]
"""

RULES = """
[
    { "gateway_model_name": "gw/simple",
      "fallback_models": [ { "provider": "mock", "model": "ok" } ] },
    { "gateway_model_name": "gw/fallback",
      "fallback_models": [
          { "provider": "mock", "model": "http500" },
          { "provider": "mock2", "model": "ok" } ] },
    { "gateway_model_name": "gw/allfail",
      "fallback_models": [
          { "provider": "mock", "model": "http500" },
          { "provider": "mock", "model": "errbody" } ] },
    { "gateway_model_name": "gw/errchunk",
      "fallback_models": [
          { "provider": "mock", "model": "errchunk" },
          { "provider": "mock", "model": "ok" } ] },
    { "gateway_model_name": "gw/rotate", "rotate_models": true,
      "fallback_models": [
          { "provider": "mock", "model": "ok" },
          { "provider": "mock2", "model": "ok" } ] },
    { "gateway_model_name": "gw/retry",
      "fallback_models": [
          { "provider": "mock", "model": "flaky:2", "retry_count": 3, "retry_delay": 1 } ] },
    { "gateway_model_name": "gw/split",
      "fallback_models": [ { "provider": "mock", "model": "slowsplit" } ] },
    { "gateway_model_name": "gw/suborder",
      "fallback_models": [
          { "provider": "mock", "model": "subonly:good",
            "use_provider_order_as_fallback": true,
            "providers_order": ["bad", "good"] } ] },
    { "gateway_model_name": "gw/inject",
      "fallback_models": [
          { "provider": "mock", "model": "echo",
            "custom_body_params": {"reasoning_effort": "high", "service_tier": "flex"},
            "custom_headers": {"x-demo": "42"} } ] },
    { "gateway_model_name": "gw/suborder-pinned",
      "fallback_models": [
          { "provider": "mock", "model": "subonly:good",
            "providers_order": ["bad", "good"] } ] }
]
"""


def build_app(tmp_path, api_key=None, log_chat=False):
    (tmp_path / "providers.json").write_text(PROVIDERS)
    (tmp_path / "models_fallback_rules.json").write_text(RULES)
    settings = Settings(
        fallback_provider="mock",
        gateway_api_key=api_key,
        log_chat_messages=log_chat,
        log_file_limit=3,
    )
    app = create_app(
        settings=settings,
        providers_path=str(tmp_path / "providers.json"),
        fallback_rules_path=str(tmp_path / "models_fallback_rules.json"),
        db_dir=str(tmp_path / "db"),
        log_dir=str(tmp_path / "logs"),
    )
    # point the dispatcher's HTTP client at the in-process mock upstream
    mock = make_mock_upstream()
    app.state.dispatcher.http._client = httpx.AsyncClient(
        transport=httpx.ASGITransport(app=mock), base_url="http://mock.test"
    )
    return app


@pytest.fixture
def client(tmp_path):
    FLAKY_COUNTS.clear()
    app = build_app(tmp_path)
    with TestClient(app) as c:
        c.tmp_path = tmp_path
        yield c


def chat(client, model, stream=False, **kw):
    return client.post(
        "/v1/chat/completions",
        json={"model": model, "messages": [{"role": "user", "content": "hi"}], "stream": stream, **kw},
    )


def read_sse_content(resp) -> str:
    text = resp.content.decode()
    out = []
    for frame in text.split("\n\n"):
        if frame.startswith("data: {"):
            obj = json.loads(frame[6:])
            for ch in obj.get("choices", []):
                if ch.get("delta", {}).get("content"):
                    out.append(ch["delta"]["content"])
    return "".join(out)


def test_health_and_root(client):
    assert client.get("/health").json() == {"status": "ok"}
    r = client.get("/", follow_redirects=False)
    assert r.status_code == 307 and "/v1/ui/rules-editor" in r.headers["location"]


def test_simple_completion(client):
    r = chat(client, "gw/simple")
    assert r.status_code == 200
    body = r.json()
    assert body["choices"][0]["message"]["content"] == "Hello from mock"
    assert "x-request-id" in r.headers


def test_streaming_completion(client):
    r = chat(client, "gw/simple", stream=True)
    assert r.status_code == 200
    assert r.headers["content-type"].startswith("text/event-stream")
    assert read_sse_content(r) == "Hello from mock"
    assert "data: [DONE]" in r.content.decode()


def test_fallback_on_http_error(client):
    r = chat(client, "gw/fallback")
    assert r.status_code == 200
    assert r.json()["choices"][0]["message"]["content"] == "Hello from mock"


def test_fallback_on_first_chunk_error_streaming(client):
    # primary model emits an error in the first SSE chunk -> zero client bytes
    # from it; fallback must serve the stream (request_handler.py:67-100 parity)
    r = chat(client, "gw/errchunk", stream=True)
    assert r.status_code == 200
    content = r.content.decode()
    assert "no capacity" not in content
    assert read_sse_content(r) == "Hello from mock"


def test_all_fail_503(client):
    r = chat(client, "gw/allfail")
    assert r.status_code == 503
    assert "failed" in r.json()["detail"]


def test_unknown_model_uses_fallback_provider(client):
    r = chat(client, "ok")  # no rule, FALLBACK_PROVIDER=mock, model passthrough
    assert r.status_code == 200


def test_rotation_round_robin(client):
    seq = []
    for _ in range(4):
        r = chat(client, "gw/rotate")
        assert r.status_code == 200
        seq.append(r.status_code)
    # rotation state advanced in the DB: 4 requests → indices 0,1,0,1
    rows = client.app.state.rotation_db._conn.execute(
        "SELECT last_model_index FROM model_rotation"
    ).fetchall()
    assert rows and rows[0][0] in (0, 1)


def test_retries_until_success(client):
    t0 = time.monotonic()
    r = chat(client, "gw/retry")
    elapsed = time.monotonic() - t0
    assert r.status_code == 200
    assert elapsed >= 2.0  # two 1 s retry delays before success


def test_split_frames_reassembled(client):
    r = chat(client, "gw/split", stream=True)
    assert r.status_code == 200
    assert read_sse_content(r) == "Hello from mock"


def test_usage_accounting_nonstreaming(client):
    chat(client, "gw/simple")
    db = client.app.state.usage_db
    assert db.get_total_records_count() == 1
    rec = db.get_latest_usage_records(1)[0]
    assert rec["prompt_tokens"] == 7 and rec["completion_tokens"] == 3
    assert rec["provider"] == "mock"


def test_usage_accounting_streaming(client):
    chat(client, "gw/simple", stream=True)
    db = client.app.state.usage_db
    assert db.get_total_records_count() == 1
    rec = db.get_latest_usage_records(1)[0]
    assert rec["total_tokens"] == 10
    assert rec["cost"] == pytest.approx(0.002)


def test_missing_model_400(client):
    r = client.post("/v1/chat/completions", json={"messages": []})
    assert r.status_code == 400


def test_auth_enforced(tmp_path):
    app = build_app(tmp_path, api_key="secret123")
    with TestClient(app) as c:
        r = chat(c, "gw/simple")
        assert r.status_code == 401
        r = c.post(
            "/v1/chat/completions",
            json={"model": "gw/simple", "messages": []},
            headers={"Authorization": "Bearer wrong"},
        )
        assert r.status_code == 401
        r = c.post(
            "/v1/chat/completions",
            json={"model": "gw/simple", "messages": []},
            headers={"Authorization": "Bearer secret123"},
        )
        assert r.status_code == 200
        # non-chat endpoints stay open
        assert c.get("/health").status_code == 200
        assert c.get("/v1/models").status_code == 200


def test_models_endpoint(client):
    r = client.get("/v1/models")
    assert r.status_code == 200
    data = r.json()["data"]
    ids = [m["id"] for m in data]
    # gateway models first, then provider models sorted
    assert ids[0].startswith("gw/")
    assert data[0]["owned_by"] == "llmgateway"
    assert "mock-model-a" in ids and "mock-model-b" in ids
    assert ids.index("mock-model-a") < ids.index("mock-model-b")


def test_models_opencode_format(client):
    r = client.get("/v1/models/AsOpenCodeFormat")
    assert r.status_code == 200
    block = r.json()["provider"]["llm-gateway-local"]
    assert block["npm"] == "@ai-sdk/openai-compatible"
    models = block["models"]
    assert "gw/simple" in models
    assert models["gw/simple"]["limit"] == {"context": 200000, "output": 32000}
    assert "variants" in models["gw/simple"]


def test_models_copilot_format(client):
    r = client.get("/v1/models/AsGitHubCopilotFormat")
    assert r.status_code == 200
    body = r.json()
    assert body["apiType"] == "chat-completions"
    entry = [m for m in body["models"] if m["id"] == "gw/simple"][0]
    assert entry["toolCalling"] is True and entry["vision"] is True
    assert entry["supportsReasoningEffort"]


def test_chat_transcript_logging(tmp_path):
    app = build_app(tmp_path, log_chat=True)
    with TestClient(app) as c:
        for _ in range(5):
            chat(c, "gw/simple", stream=True)
    logs = list((tmp_path / "logs").glob("*.txt"))
    assert 1 <= len(logs) <= 3  # pruned to log_file_limit=3
    text = logs[-1].read_text()
    assert "Hello from mock" in text


def test_rules_editor_roundtrip(client):
    r = client.get("/v1/config/models-rules")
    assert r.status_code == 200 and "gw/simple" in r.text

    # invalid save -> 400 + old config kept
    r = client.post("/v1/config/models-rules", content=b"[{\"bad\": 1}]")
    assert r.status_code == 400
    assert "gw/simple" in client.app.state.config_loader.fallback_rules

    # valid save -> reloaded
    new_rules = '[{"gateway_model_name": "gw/new", "fallback_models": [{"provider": "mock", "model": "ok"}]}] // comment'
    r = client.post("/v1/config/models-rules", content=new_rules.encode())
    assert r.status_code == 200
    assert set(client.app.state.config_loader.fallback_rules) == {"gw/new"}
    # and /v1/models reflects the reload immediately (reference quirk fixed)
    ids = [m["id"] for m in client.get("/v1/models").json()["data"]]
    assert "gw/new" in ids and "gw/simple" not in ids


def test_providers_editor_semantic_guard(client):
    # removing a provider still referenced by rules must 400
    r = client.post(
        "/v1/config/providers",
        content=b'[{"mock": {"baseUrl": "http://mock.test/v1", "apikey": ""}}]',
    )
    assert r.status_code == 400


def test_stats_endpoints(client):
    chat(client, "gw/simple")
    r = client.get("/v1/api/usage-stats/day")
    assert r.status_code == 200
    rows = r.json()
    assert rows and rows[0]["count"] == 1
    assert client.get("/v1/api/usage-stats/bogus").status_code == 400
    r = client.get("/v1/api/usage-records?limit=10")
    assert r.json()["total"] == 1


def test_custom_params_injected(client, monkeypatch):
    seen = {}
    dispatcher = client.app.state.dispatcher
    orig = dispatcher.http.make_request

    async def spy(url, headers, payload, is_streaming):
        seen["payload"] = payload
        seen["headers"] = headers
        return await orig(url, headers, payload, is_streaming)

    monkeypatch.setattr(dispatcher.http, "make_request", spy)
    # add a rule with custom params via the editor
    rules = (
        '[{"gateway_model_name": "gw/custom", "fallback_models": '
        '[{"provider": "mock", "model": "ok", '
        '"custom_body_params": {"reasoning_effort": "high"}, '
        '"custom_headers": {"x-demo": "1"}}]}]'
    )
    assert client.post("/v1/config/models-rules", content=rules.encode()).status_code == 200
    r = chat(client, "gw/custom")
    assert r.status_code == 200
    assert seen["payload"]["reasoning_effort"] == "high"
    assert seen["headers"]["x-demo"] == "1"
    assert seen["payload"]["model"] == "ok"


def test_sub_provider_order_as_fallback(tmp_path):
    """OpenRouter sub-provider ordering (reference chat.py:159-189): with
    use_provider_order_as_fallback, one attempt per sub-provider in order,
    each pinned with allow_fallbacks semantics."""
    from tests.mock_upstream import SUB_ORDERS_SEEN

    client = TestClient(build_app(tmp_path))
    with client:
        SUB_ORDERS_SEEN.clear()
        r = chat(client, "gw/suborder")
        assert r.status_code == 200
        assert SUB_ORDERS_SEEN == [["bad"], ["good"]]


def test_sub_provider_order_pinned_list(tmp_path):
    """Without the fallback flag the whole order list is sent in ONE
    attempt (chat.py:150-156) — the mock only accepts ["good"], so the
    pinned ["bad", "good"] attempt fails and the request 503s."""
    from tests.mock_upstream import SUB_ORDERS_SEEN

    client = TestClient(build_app(tmp_path))
    with client:
        SUB_ORDERS_SEEN.clear()
        r = chat(client, "gw/suborder-pinned")
        assert r.status_code == 503
        assert SUB_ORDERS_SEEN == [["bad", "good"]]


def test_custom_params_and_headers_injected(tmp_path):
    """custom_body_params / custom_headers reach the upstream verbatim
    (reference chat.py:116-123), plus the provider Bearer key."""
    client = TestClient(build_app(tmp_path))
    with client:
        r = client.post(
            "/v1/chat/completions",
            json={"model": "gw/inject", "messages": [{"role": "user", "content": "hi"}],
                  "stream": False},
        )
        assert r.status_code == 200
        echoed = json.loads(r.json()["choices"][0]["message"]["content"])
        assert echoed["body"]["reasoning_effort"] == "high"
        assert echoed["body"]["service_tier"] == "flex"
        assert echoed["body"]["model"] == "echo"  # per-entry model override
        assert echoed["x_demo"] == "42"
        assert echoed["auth"] == "Bearer MOCK_KEY"


def test_models_degrades_when_upstream_unreachable(tmp_path):
    """Upstream /models failure degrades to the gateway-rule list
    (reference models.py:239-296 graceful path)."""
    (tmp_path / "providers.json").write_text(
        PROVIDERS.replace('"http://mock.test/v1", "apikey": "MOCK_KEY"',
                          '"http://unreachable.test/v1", "apikey": "MOCK_KEY"')
    )
    (tmp_path / "models_fallback_rules.json").write_text(RULES)
    settings = Settings(fallback_provider="mock", gateway_api_key=None)
    app = create_app(
        settings=settings,
        providers_path=str(tmp_path / "providers.json"),
        fallback_rules_path=str(tmp_path / "models_fallback_rules.json"),
        db_dir=str(tmp_path / "db"),
        log_dir=str(tmp_path / "logs"),
    )
    # no mock transport wired: the /models fetch truly fails
    with TestClient(app) as client:
        r = client.get("/v1/models")
        assert r.status_code == 200
        ids = [m["id"] for m in r.json()["data"]]
        assert ids and all(i.startswith("gw/") for i in ids)


def test_request_log_includes_payload_summary(caplog):
    """POST chat payloads are logged with messages/tools excluded
    (reference request_logging.py:49-61 parity)."""
    import logging as _logging

    from llmapigateway_amd.gateway.middleware import RequestLoggingMiddleware

    captured = {}

    async def app(scope, receive, send):
        msg = await receive()
        captured["body"] = msg.get("body")
        await send({"type": "http.response.start", "status": 200, "headers": []})
        await send({"type": "http.response.body", "body": b"{}"})

    mw = RequestLoggingMiddleware(app)
    body = (
        b'{"model": "m", "temperature": 0.5, '
        b'"messages": [{"role": "user", "content": "secret"}], "tools": []}'
    )
    sent = []

    async def receive():
        return {"type": "http.request", "body": body, "more_body": False}

    async def send(m):
        sent.append(m)

    scope = {
        "type": "http",
        "method": "POST",
        "path": "/v1/chat/completions",
        "headers": [],
    }
    import asyncio

    with caplog.at_level(_logging.INFO):
        asyncio.run(mw(scope, receive, send))
    assert captured["body"] == body  # replayed intact to the app
    line = next(r.message for r in caplog.records if "payload=" in r.message)
    assert "temperature" in line and "secret" not in line and "messages" not in line
