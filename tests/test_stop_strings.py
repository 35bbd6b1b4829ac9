"""Stop-string semantics on local engines: generation halts at the stop
string and the stop text never reaches the client (streaming or not)."""

import json

import pytest
import torch
from fastapi.testclient import TestClient

from llmapigateway_amd.config.settings import Settings
from llmapigateway_amd.engine import LLMEngine, EngineRequest, SamplingParams
from llmapigateway_amd.engine.tokenizer import ByteTokenizer
from llmapigateway_amd.gateway.app import create_app

PROVIDERS = '[{"eng": {"baseUrl": "local://tiny-llama?device=0", "apikey": ""}}]'
RULES = (
    '[{"gateway_model_name": "local/m", "fallback_models":'
    ' [{"provider": "eng", "model": "tiny-llama"}]}]'
)


@pytest.fixture
def client(tmp_path):
    (tmp_path / "providers.json").write_text(PROVIDERS)
    (tmp_path / "models_fallback_rules.json").write_text(RULES)
    app = create_app(
        settings=Settings(fallback_provider="eng"),
        providers_path=str(tmp_path / "providers.json"),
        fallback_rules_path=str(tmp_path / "models_fallback_rules.json"),
        db_dir=str(tmp_path / "db"),
        log_dir=str(tmp_path / "logs"),
    )
    with TestClient(app) as c:
        yield c


def _find_stop_char(client):
    """Generate some text and pick a character from it to use as stop."""
    r = client.post(
        "/v1/chat/completions",
        json={"model": "local/m", "messages": [{"role": "user", "content": "hi"}],
              "max_tokens": 20, "ignore_eos": True},
    )
    text = r.json()["choices"][0]["message"]["content"]
    assert text
    return text, text[max(0, len(text) // 2)]


def test_engine_level_stop():
    eng = LLMEngine(model="tiny-llama", device="cpu", dtype=torch.float32,
                    block_size=16, num_blocks=64, seed=0)
    full = eng.generate([1, 5, 9], SamplingParams(max_tokens=12, ignore_eos=True))
    text = ByteTokenizer().decode(full.out_ids)
    stop_char = text[5]
    stopped = eng.generate(
        [1, 5, 9], SamplingParams(max_tokens=12, ignore_eos=True, stop=[stop_char])
    )
    assert stopped.finish_reason == "stop"
    assert len(stopped.out_ids) <= len(full.out_ids)


def test_nonstream_stop_truncates(client):
    full_text, stop_char = _find_stop_char(client)
    r = client.post(
        "/v1/chat/completions",
        json={"model": "local/m", "messages": [{"role": "user", "content": "hi"}],
              "max_tokens": 20, "ignore_eos": True, "stop": stop_char},
    )
    body = r.json()
    out = body["choices"][0]["message"]["content"]
    assert stop_char not in out
    assert full_text.startswith(out)


def test_stream_stop_truncates(client):
    full_text, stop_char = _find_stop_char(client)
    r = client.post(
        "/v1/chat/completions",
        json={"model": "local/m", "messages": [{"role": "user", "content": "hi"}],
              "max_tokens": 20, "ignore_eos": True, "stop": [stop_char], "stream": True},
    )
    assert r.status_code == 200
    content = "".join(
        c["delta"].get("content", "")
        for f in r.content.decode().split("\n\n")
        if f.startswith("data: {")
        for c in json.loads(f[6:])["choices"]
    )
    assert stop_char not in content
    assert full_text.startswith(content)
