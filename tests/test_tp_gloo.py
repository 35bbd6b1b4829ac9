"""Tensor-parallel correctness on CPU: TP=2 over gloo must produce the same
logits as TP=1 (the shards are slices of the same full random weights).
Runs as a spawned 2-process group on 127.0.0.1 (no GPU needed)."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _tp_worker(rank, world, port, result_queue):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from llmapigateway_amd.engine import LLMEngine, SamplingParams
        from llmapigateway_amd.models.configs import get_model_config

        eng = LLMEngine(
            model="tiny-llama",
            device="cpu",
            dtype=torch.float32,
            block_size=16,
            num_blocks=64,
            seed=0,
            tp_group=dist.group.WORLD,
            tp_rank=rank,
            tp_size=world,
        )
        req = eng.generate([1, 5, 9, 13, 21], SamplingParams(max_tokens=6, ignore_eos=True))
        result_queue.put((rank, req.out_ids))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_tp2_matches_tp1():
    # TP=1 baseline
    from llmapigateway_amd.engine import LLMEngine, SamplingParams

    base = LLMEngine(
        model="tiny-llama", device="cpu", dtype=torch.float32, block_size=16,
        num_blocks=64, seed=0,
    )
    expected = base.generate(
        [1, 5, 9, 13, 21], SamplingParams(max_tokens=6, ignore_eos=True)
    ).out_ids

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29781
    procs = [
        ctx.Process(target=_tp_worker, args=(r, 2, port, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, out_ids = q.get(timeout=150)
        results[rank] = out_ids
    for p in procs:
        p.join(timeout=30)
    assert results[0] == results[1] == expected, (results, expected)


def test_shard_helpers_roundtrip():
    from llmapigateway_amd.parallel.tp import (
        shard_column,
        shard_gate_up,
        shard_qkv,
        shard_row,
    )

    full = torch.arange(48, dtype=torch.float32).reshape(8, 6)
    cols = [shard_column(full, r, 2) for r in range(2)]
    assert torch.equal(torch.cat(cols, dim=0), full)
    rows = [shard_row(full, r, 2) for r in range(2)]
    assert torch.equal(torch.cat(rows, dim=1), full)

    # qkv: q=4 rows, k=2, v=2
    qkv = [shard_qkv(full, r, 2, 4, 2) for r in range(2)]
    assert torch.equal(qkv[0], torch.cat([full[0:2], full[4:5], full[6:7]]))
    assert torch.equal(qkv[1], torch.cat([full[2:4], full[5:6], full[7:8]]))

    gu = [shard_gate_up(full, r, 2, 4) for r in range(2)]
    assert torch.equal(gu[0], torch.cat([full[0:2], full[4:6]]))
