import pytest

from llmapigateway_amd.config import jsonc


def test_plain_json():
    assert jsonc.loads('{"a": 1, "b": [2, 3]}') == {"a": 1, "b": [2, 3]}


def test_line_comments():
    text = """
    {
        // a comment
        "a": 1, // trailing comment
        "url": "http://x//y"  // slashes inside strings stay
    }
    """
    assert jsonc.loads(text) == {"a": 1, "url": "http://x//y"}


def test_block_comments():
    text = '{"a": /* inline */ 1, /* multi\nline */ "b": 2}'
    assert jsonc.loads(text) == {"a": 1, "b": 2}


def test_trailing_commas():
    assert jsonc.loads('{"a": [1, 2,], }') == {"a": [1, 2]}


def test_comment_markers_inside_strings():
    assert jsonc.loads('{"a": "// not a comment /* neither */"}') == {
        "a": "// not a comment /* neither */"
    }


def test_unterminated_block_comment():
    with pytest.raises(jsonc.JsoncError):
        jsonc.loads('{"a": 1} /* oops')


def test_invalid_json_raises():
    with pytest.raises(jsonc.JsoncError):
        jsonc.loads("{a: 1}")


def test_bytes_input():
    assert jsonc.loads(b'{"a": 1}') == {"a": 1}


def test_reference_example_configs_parse():
    # the real reference example files use comments + nesting; our parser must accept them
    for p in (
        "/root/reference/providers.json.example",
        "/root/reference/models_fallback_rules.json.example",
    ):
        with open(p) as f:
            data = jsonc.load(f)
        assert isinstance(data, list) and data
