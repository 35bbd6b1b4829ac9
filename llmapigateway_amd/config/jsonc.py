"""Lenient JSON ("JSON with comments") parser.

The reference gateway parses its two config files with the json5 package so
users can annotate them with // and /* */ comments (see
/root/reference/llm_gateway_core/config/loader.py:60-65 for the behavior we
reproduce). json5 is not available in this environment, and we only need the
subset the reference's config files actually exercise:

- // line comments and /* block */ comments,
- trailing commas in objects and arrays,
- otherwise standard JSON.

We strip comments and trailing commas with a small string-aware scanner and
delegate the rest to the stdlib json module, which keeps the error messages
precise (we map positions back to the original text by never changing
offsets: stripped characters are replaced with spaces, newlines preserved).
"""

from __future__ import annotations

import json
from typing import Any, IO


class JsoncError(ValueError):
    """Raised when the text is not valid JSON-with-comments."""


def _strip(text: str) -> str:
    out = list(text)
    i, n = 0, len(text)
    in_string = False
    while i < n:
        c = text[i]
        if in_string:
            if c == "\\":
                i += 2
                continue
            if c == '"':
                in_string = False
            i += 1
            continue
        if c == '"':
            in_string = True
            i += 1
            continue
        if c == "/" and i + 1 < n and text[i + 1] == "/":
            while i < n and text[i] != "\n":
                out[i] = " "
                i += 1
            continue
        if c == "/" and i + 1 < n and text[i + 1] == "*":
            start = i
            i += 2
            while i + 1 < n and not (text[i] == "*" and text[i + 1] == "/"):
                i += 1
            if i + 1 >= n:
                raise JsoncError(f"Unterminated block comment starting at offset {start}")
            i += 2
            for j in range(start, i):
                if out[j] != "\n":
                    out[j] = " "
            continue
        i += 1
    if in_string:
        raise JsoncError("Unterminated string")
    return "".join(out)


def _strip_trailing_commas(text: str) -> str:
    # After comment stripping, remove commas whose next non-space char is } or ].
    out = list(text)
    i, n = 0, len(text)
    in_string = False
    while i < n:
        c = text[i]
        if in_string:
            if c == "\\":
                i += 2
                continue
            if c == '"':
                in_string = False
            i += 1
            continue
        if c == '"':
            in_string = True
        elif c == ",":
            j = i + 1
            while j < n and text[j] in " \t\r\n":
                j += 1
            if j < n and text[j] in "}]":
                out[i] = " "
        i += 1
    return "".join(out)


def loads(text: str | bytes) -> Any:
    if isinstance(text, (bytes, bytearray)):
        text = text.decode("utf-8")
    cleaned = _strip_trailing_commas(_strip(text))
    try:
        return json.loads(cleaned)
    except json.JSONDecodeError as e:
        raise JsoncError(str(e)) from e


def load(fp: IO) -> Any:
    return loads(fp.read())


def dumps(obj: Any, **kw: Any) -> str:
    return json.dumps(obj, **kw)
