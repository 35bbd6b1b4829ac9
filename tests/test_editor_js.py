"""The self-hosted JSONC editor component (static/jsonc-editor.js):
validation/highlight logic unit-tested under node (same dialect as the
server's config/jsonc.py)."""

import json
import shutil
import subprocess

import pytest

NODE = shutil.which("node")
pytestmark = pytest.mark.skipif(NODE is None, reason="node not available")


def run_js(expr: str) -> dict:
    script = (
        "const E = require('./static/jsonc-editor.js');"
        f"console.log(JSON.stringify({expr}));"
    )
    out = subprocess.run(
        [NODE, "-e", script], capture_output=True, text=True, cwd="."
    )
    assert out.returncode == 0, out.stderr
    return json.loads(out.stdout.strip())


def test_valid_jsonc_with_comments_and_trailing_commas():
    text = '{\n // hi\n "a": [1, 2,], /* block */ "b": "x",\n}'
    assert run_js(f"E.validateJsonc({json.dumps(text)})")["ok"] is True


def test_invalid_jsonc_reports_line():
    text = '{\n "a": 1,\n "b": oops\n}'
    v = run_js(f"E.validateJsonc({json.dumps(text)})")
    assert v["ok"] is False
    assert v.get("line") == 3


def test_comment_inside_string_preserved():
    text = '{"url": "http://x/y"}'
    assert run_js(f"E.validateJsonc({json.dumps(text)})")["ok"] is True


def test_highlight_classes():
    text = '{"k": "v", "n": 3, "t": true} // c'
    html = run_js(f"E.highlight({json.dumps(text)})")
    for cls in ("tk-key", "tk-s", "tk-n", "tk-k", "tk-c", "tk-p"):
        assert cls in html, f"missing {cls} in {html}"


def test_dialect_matches_server():
    """Client- and server-side validators must agree on the dialect."""
    from llmapigateway_amd.config import jsonc

    cases = [
        ('{"a": 1, // c\n "b": [2,],}', True),
        ('{"a": }', False),
        ('[1, 2, /* x */ 3]', True),
        ('{"s": "a//b"}', True),
    ]
    for text, ok in cases:
        server_ok = True
        try:
            jsonc.loads(text)
        except Exception:
            server_ok = False
        client_ok = run_js(f"E.validateJsonc({json.dumps(text)})")["ok"]
        assert server_ok == ok and client_ok == ok, (text, server_ok, client_ok)
