"""In-tree byte-level BPE tokenizer (engine/tokenizer.py BPETokenizer,
vocab assets/bpe32k.json — trained offline by tools/train_tokenizer.py).

VERDICT r1 item 5 'Done' criteria: stop-string and usage-token tests pass
with the BPE, render_chat produces a real llama-style template.
"""

import pytest
import torch

from llmapigateway_amd.engine import EngineRequest, LLMEngine, SamplingParams
from llmapigateway_amd.engine.tokenizer import BPETokenizer, ByteTokenizer, get_tokenizer


@pytest.fixture(scope="module")
def tok():
    return BPETokenizer.shared()


def test_round_trip_lossless(tok):
    for text in [
        "Hello world!",
        "def f(x):\n    return x ** 2  # comment",
        "tabs\tand\nnewlines and  double  spaces",
        "unicode: café → 中文 \U0001f600",
    ]:
        assert tok.decode(tok.encode(text)) == text


def test_real_bpe_merges(tok):
    # common words are single tokens, rare words split into several
    assert len(tok.encode("the", add_bos=False)) == 1
    assert len(tok.encode(" return", add_bos=False)) == 1
    long_ids = tok.encode("internationalization", add_bos=False)
    assert 1 < len(long_ids) < 21  # merged pieces, not bytes
    # whitespace-prefixed word pieces (byte-level BPE property; the
    # corpus is code, so "value"/" value" merged but "hello" did not)
    a = tok.encode("value value", add_bos=False)
    assert len(a) == 2  # "value" + " value" both single tokens


def test_llama_style_template(tok):
    msgs = [
        {"role": "system", "content": "Be terse."},
        {"role": "user", "content": "Hi there"},
    ]
    text = tok.render_chat(msgs)
    assert "<|start_header_id|>system<|end_header_id|>" in text
    assert "<|eot_id|>" in text
    assert text.endswith("<|start_header_id|>assistant<|end_header_id|>\n\n")
    ids = tok.encode(text)
    # specials encode to single ids
    assert ids.count(tok._hdr_s) == 3
    assert ids.count(tok.EOT) == 2


def test_specials_and_oov_decode(tok):
    # special ids vanish from decoded text; OOV ids fold to printable
    s = tok.decode([tok.BOS, 500, tok.EOT, 120000])
    assert "<|" not in s and all(32 <= ord(c) < 127 or c in "\n\t" for c in s)


def test_preset_selection():
    assert isinstance(get_tokenizer("auto", 512), ByteTokenizer)
    assert isinstance(get_tokenizer("auto", 128256), BPETokenizer)
    assert isinstance(get_tokenizer("byte", 128256), ByteTokenizer)


def test_engine_stop_string_with_bpe():
    """Stop strings must work through the engine with BPE decode: feed a
    prompt, force the output tokens (greedy over tiny random weights is
    arbitrary), and check the finish logic uses BPE text."""
    tok = BPETokenizer.shared()
    eng = LLMEngine(
        model="tiny-llama", device="cpu", dtype=torch.float32, num_blocks=64,
        tokenizer=tok,
    )
    stop_ids = tok.encode("STOP", add_bos=False)
    req = EngineRequest(
        [1, 2, 3],
        SamplingParams(max_tokens=50, stop=["STOP"], ignore_eos=True),
    )
    eng.add_request(req)
    eng.step()  # prefill + first token
    # inject the stop string as if sampled (deliver path runs under lock)
    with eng._lock:
        eng._deliver([req], [stop_ids[0]])
        for t in stop_ids[1:]:
            if req.state != "running":
                break
            eng._deliver([req], [t])
    assert req.state == "finished"
    assert req.finish_reason == "stop"
    assert "STOP" in req.text


def test_usage_token_counts_with_bpe():
    """prompt_tokens reported by the gateway path must equal the BPE
    encoding length of the rendered template (reference usage semantics:
    chat_logging.py:233-263)."""
    tok = BPETokenizer.shared()
    msgs = [{"role": "user", "content": "Count my tokens please."}]
    prompt_ids = tok.encode(tok.render_chat(msgs))
    assert len(prompt_ids) > 8  # template + content, single-id specials
    # engine enforces context from the same count
    eng = LLMEngine(
        model="tiny-llama", device="cpu", dtype=torch.float32, num_blocks=64,
        max_model_len=16, tokenizer=tok,
    )
    too_long = tok.encode("word " * 40)
    with pytest.raises(ValueError):
        eng.add_request(EngineRequest(too_long, SamplingParams(max_tokens=4)))
