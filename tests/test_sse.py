from llmapigateway_amd.gateway import sse


def test_parser_reassembles_split_frames():
    p = sse.SSEParser()
    frames = []
    whole = b'data: {"a": 1}\n\ndata: {"b": 2}\n\n'
    for i in range(0, len(whole), 5):
        frames.extend(p.feed(whole[i : i + 5]))
    assert frames == ['data: {"a": 1}', 'data: {"b": 2}']


def test_parser_keeps_partial():
    p = sse.SSEParser()
    assert p.feed(b'data: {"a":') == []
    assert p.feed(b" 1}\n\n") == ['data: {"a": 1}']
    assert p.flush() is None


def test_parse_data_frame():
    assert sse.parse_data_frame('data: {"x": 1}') == {"x": 1}
    assert sse.parse_data_frame("data: [DONE]") is None
    assert sse.parse_data_frame(": comment") is None
    assert sse.parse_data_frame("data: {broken") is None


def test_error_detection():
    assert sse.frame_is_error({"error": {"message": "boom"}})
    assert sse.frame_is_error({"detail": "nope"})
    assert not sse.frame_is_error({"choices": []})
    assert sse.extract_error_detail({"error": {"message": "boom"}}) == "boom"
    assert sse.extract_error_detail({"detail": "nope"}) == "nope"


def test_sniffer_accumulates_content_and_usage():
    s = sse.StreamSniffer()
    s.observe('data: {"model": "m", "choices": [{"delta": {"content": "He"}}]}')
    s.observe('data: {"choices": [{"delta": {"content": "y"}, "finish_reason": "stop"}]}')
    s.observe('data: {"usage": {"prompt_tokens": 2, "completion_tokens": 1, "total_tokens": 3}}')
    assert s.full_content() == "Hey"
    assert s.model == "m"
    assert s.finish_reason == "stop"
    assert s.usage["total_tokens"] == 3


def test_sniffer_detects_midstream_error_chunk():
    s = sse.StreamSniffer()
    s.observe('data: {"code": 429, "error": {"message": "rate limited"}}')
    assert s.error_detail == "rate limited"


def test_token_usage_fields_reasoning_subtraction():
    fields = sse.token_usage_fields(
        {
            "prompt_tokens": 10,
            "completion_tokens": 20,
            "total_tokens": 30,
            "completion_tokens_details": {"reasoning_tokens": 5},
            "prompt_tokens_details": {"cached_tokens": 4},
            "cost": 0.5,
        }
    )
    assert fields["completion_tokens"] == 15  # reasoning subtracted
    assert fields["reasoning_tokens"] == 5
    assert fields["cached_tokens"] == 4
    assert fields["cost"] == 0.5


def test_format_sse_roundtrip():
    frame = sse.format_sse({"a": 1})
    assert frame.startswith(b"data: ") and frame.endswith(b"\n\n")
    p = sse.SSEParser()
    [text] = p.feed(frame)
    assert sse.parse_data_frame(text) == {"a": 1}
