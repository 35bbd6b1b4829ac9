"""GPU engine integration tests (small d128 model, all native kernels)."""

import pytest
import torch

from llmapigateway_amd import ops
from llmapigateway_amd.engine import LLMEngine, EngineRequest, SamplingParams
from llmapigateway_amd.models.configs import ModelConfig

pytestmark = pytest.mark.gpu

GPU_TINY = ModelConfig(
    name="gpu-tiny",
    hidden_size=512,
    intermediate_size=1024,
    num_layers=2,
    num_heads=4,
    num_kv_heads=2,
    vocab_size=2048,
    head_dim=128,
    rope_theta=10000.0,
    max_positions=1024,
)


@pytest.fixture(scope="module")
def engine():
    assert torch.cuda.is_available() and ops.have_native()
    return LLMEngine(
        model=GPU_TINY,
        device="cuda:0",
        dtype=torch.bfloat16,
        block_size=16,
        num_blocks=256,
        seed=0,
    )


def test_greedy_deterministic(engine):
    p = [1, 9, 17, 33, 200]
    r1 = engine.generate(p, SamplingParams(max_tokens=8, ignore_eos=True))
    r2 = engine.generate(p, SamplingParams(max_tokens=8, ignore_eos=True))
    assert r1.out_ids == r2.out_ids
    assert len(r1.out_ids) == 8


def test_decode_consistent_with_recompute(engine):
    """Incremental decode (paged KV + decode kernel) tokens must match a full
    prefill recompute of the grown sequence at every step."""
    prompt = [1, 7, 42, 99, 500, 3]
    req = engine.generate(prompt, SamplingParams(max_tokens=6, ignore_eos=True))
    gen = req.out_ids
    for i in range(1, len(gen) + 1):
        # re-prefill prompt + gen[:i-1]; next token must equal gen[i-1]
        r = engine.generate(
            prompt + gen[: i - 1], SamplingParams(max_tokens=1, ignore_eos=True)
        )
        assert r.out_ids[0] == gen[i - 1], f"divergence at step {i}"


def test_batched_matches_solo(engine):
    prompts = [[1, 5, 9], [2, 4, 8, 16], [3, 6, 9, 12, 15]]
    solo = [
        engine.generate(p, SamplingParams(max_tokens=5, ignore_eos=True)).out_ids
        for p in prompts
    ]
    reqs = [
        EngineRequest(p, SamplingParams(max_tokens=5, ignore_eos=True)) for p in prompts
    ]
    for r in reqs:
        engine.add_request(r)
    while any(r.state in ("waiting", "running") for r in reqs):
        engine.step()
    assert [r.out_ids for r in reqs] == solo


def test_long_generation_crosses_blocks(engine):
    # generation spans multiple 16-token KV blocks
    req = engine.generate(
        list(range(3, 20)), SamplingParams(max_tokens=60, ignore_eos=True)
    )
    assert len(req.out_ids) == 60
    assert req.state == "finished"


def test_kv_blocks_freed(engine):
    free0 = engine.kv.manager.num_free_blocks
    engine.generate([1, 2, 3, 4], SamplingParams(max_tokens=4, ignore_eos=True))
    assert engine.kv.manager.num_free_blocks == free0
