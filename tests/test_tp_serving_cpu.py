"""TP worker-group serving path (engine/tp_group.py) on CPU over gloo.

Covers BASELINE configs[4] plumbing without GPUs: a TPEngineClient with
tp=2 must serve requests through the lockstep multi-process loop and emit
exactly the tokens a single-process TP=1 engine produces (greedy), because
TP=N is a pure sharding of the TP=1 model (parallel/tp.py docstring).
"""

import sys

import pytest
import torch

from llmapigateway_amd.engine import EngineRequest, LLMEngine, SamplingParams
from llmapigateway_amd.engine.tp_group import TPEngineClient

pytestmark = pytest.mark.skipif(
    sys.platform != "linux", reason="multiprocessing spawn test is linux-only"
)

PROMPT = list(range(5, 37))


def _reference_tokens(n=8):
    engine = LLMEngine(
        model="tiny-llama", device="cpu", dtype=torch.float32, num_blocks=64, seed=0
    )
    req = engine.generate(PROMPT, SamplingParams(max_tokens=n, ignore_eos=True))
    assert req.state == "finished"
    return req.out_ids


@pytest.fixture(scope="module")
def tp_client():
    client = TPEngineClient(
        model="tiny-llama", tp=2, max_batch_size=8, kv_block_size=16, num_blocks=64,
        start_timeout=180.0,
    )
    yield client
    client.stop()


def test_tp2_greedy_matches_tp1(tp_client):
    import threading

    done = threading.Event()
    req = EngineRequest(
        PROMPT,
        SamplingParams(max_tokens=8, ignore_eos=True),
        on_finish=lambda r: done.set(),
    )
    tp_client.add_request(req)
    assert done.wait(timeout=120.0), "TP group did not finish the request"
    assert req.state == "finished"
    assert req.out_ids == _reference_tokens(8)


def test_tp2_concurrent_requests(tp_client):
    import threading

    n = 4
    events = [threading.Event() for _ in range(n)]
    reqs = []
    for i in range(n):
        req = EngineRequest(
            PROMPT[: 8 + i],
            SamplingParams(max_tokens=5, ignore_eos=True),
            on_finish=lambda r, e=events[i]: e.set(),
        )
        tp_client.add_request(req)
        reqs.append(req)
    for e in events:
        assert e.wait(timeout=120.0)
    for req in reqs:
        assert req.state == "finished"
        assert len(req.out_ids) == 5


def test_tp2_abort(tp_client):
    req = EngineRequest(PROMPT, SamplingParams(max_tokens=10_000, ignore_eos=True))
    tp_client.add_request(req)
    tp_client.abort_request(req)
    assert req.state in ("finished", "failed")
    assert req.finish_reason == "aborted"


def test_stop_fails_inflight_requests_immediately():
    """Regression: registry.prune() stops a TP group while
    clients are streaming; stop() must fail pending requests right away
    instead of leaving them to the 300 s first-token timeout."""
    client = TPEngineClient(
        model="tiny-llama", tp=2, max_batch_size=8, kv_block_size=16, num_blocks=64,
        start_timeout=180.0,
    )
    import threading
    import time

    done = threading.Event()
    req = EngineRequest(
        PROMPT,
        SamplingParams(max_tokens=100_000, ignore_eos=True),
        on_finish=lambda r: done.set(),
    )
    client.add_request(req)
    t0 = time.monotonic()
    client.stop()
    assert done.wait(timeout=15.0), "stop() left the in-flight request hanging"
    assert time.monotonic() - t0 < 15.0
    assert req.state == "failed" and req.finish_reason == "error"


def test_registry_tp_spec_creates_group():
    from llmapigateway_amd.config.loader import EngineSpec
    from llmapigateway_amd.engine.registry import EngineRegistry, _TPGroupHandle

    reg = EngineRegistry()
    spec = EngineSpec(model="tiny-llama", tp=2, max_batch_size=4, kv_block_size=16)
    # CPU path: TPEngineClient defaults num_blocks=256 per rank
    handle = reg.get_engine(spec)
    try:
        assert isinstance(handle, _TPGroupHandle)
        assert handle.engine.tp == 2
        # second lookup returns the cached group
        assert reg.get_engine(spec) is handle
    finally:
        handle.stop()
        reg._engines.clear()


@pytest.mark.timeout(300)
def test_tp4_lockstep_broadcast_overhead_under_5pct():
    """World-4 lockstep group under a busy stream: the command channel
    (fixed-size tensor broadcast every CMD_STRIDE steps) must cost <5% of
    busy step time (VERDICT r1 item 2 'Done' criterion)."""
    import threading

    client = TPEngineClient(
        model="tiny-llama-tp4", tp=4, max_batch_size=16, kv_block_size=16,
        num_blocks=128, start_timeout=240.0,
    )
    try:
        n = 12
        events = [threading.Event() for _ in range(n)]
        for i in range(n):
            req = EngineRequest(
                PROMPT[: 8 + (i % 7)],
                SamplingParams(max_tokens=24, ignore_eos=True),
                on_finish=lambda r, e=events[i]: e.set(),
            )
            client.add_request(req)
        for e in events:
            assert e.wait(timeout=240.0)
    finally:
        client.stop()
    stats = client.lockstep_stats
    assert stats is not None, "worker did not report lockstep stats"
    assert stats["step_s"] > 0
    ratio = stats["bcast_s"] / (stats["bcast_s"] + stats["step_s"])
    assert ratio < 0.05, f"command-channel overhead {ratio:.1%} (bcast {stats})"
