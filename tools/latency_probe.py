"""Decompose decode-step latency on the GPU box: full engine.step() vs bare
hipGraph replay vs sampling vs Python bookkeeping. Run via gpurun."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from llmapigateway_amd.engine import LLMEngine, EngineRequest, SamplingParams


def main():
    torch.cuda.init()
    eng = LLMEngine(
        model="llama-3-8b",
        device="cuda:0",
        dtype=torch.bfloat16,
        num_blocks=2048,
        max_batch_size=64,
        max_model_len=1024,
    )
    B = 64
    reqs = [
        EngineRequest(
            list(range(3, 3 + 128)), SamplingParams(max_tokens=500, ignore_eos=True)
        )
        for _ in range(B)
    ]
    for r in reqs:
        eng.add_request(r)
    eng.step()  # prefill
    for _ in range(5):
        eng.step()  # warm decode + graph capture
    torch.cuda.synchronize()

    # (1) full step
    t0 = time.monotonic()
    N = 50
    for _ in range(N):
        eng.step()
    torch.cuda.synchronize()
    full_ms = (time.monotonic() - t0) * 1000 / N
    print(f"full engine.step():       {full_ms:.3f} ms")

    # (2) graph replay only (reuse current state, fixed inputs)
    gr = eng.graph_runner
    tokens = [r.out_ids[-1] for r in reqs]
    pos = [r.num_tokens - 1 for r in reqs]
    bs = eng.kv.block_size
    slots = [r.block_table[p // bs] * bs + p % bs for r, p in zip(reqs, pos)]
    tables = [r.block_table for r in reqs]
    ctx = [p + 1 for p in pos]

    t0 = time.monotonic()
    for _ in range(N):
        gr.run(tokens, pos, slots, tables, ctx)
    torch.cuda.synchronize()
    print(f"graph_runner.run (sync'd): {(time.monotonic()-t0)*1000/N:.3f} ms")

    # (2b) replay only, inputs already staged
    t0 = time.monotonic()
    for _ in range(N):
        gr.graphs[64].replay()
    torch.cuda.synchronize()
    print(f"graph replay only:         {(time.monotonic()-t0)*1000/N:.3f} ms")

    # (3) host-side input staging alone
    t0 = time.monotonic()
    for _ in range(N):
        gr.h_token_ids[:B] = torch.tensor(tokens, dtype=torch.long)
        gr.h_positions[:B] = torch.tensor(pos, dtype=torch.long)
        gr.h_slot_mapping[:B] = torch.tensor(slots, dtype=torch.long)
        gr.h_context_lens[:B] = torch.tensor(ctx, dtype=torch.int32)
        gr.h_block_tables[:B].zero_()
        for i, bt in enumerate(tables):
            gr.h_block_tables[i, : len(bt)] = torch.tensor(bt, dtype=torch.int32)
    print(f"host staging only:         {(time.monotonic()-t0)*1000/N:.3f} ms")

    # (4) sampling + host sync
    logits = gr.graph_logits[64]
    temps = torch.zeros(B, dtype=torch.float32, device="cuda:0")
    from llmapigateway_amd import ops

    t0 = time.monotonic()
    for _ in range(N):
        out = ops.sample(logits, temps)
        out.tolist()
    torch.cuda.synchronize()
    print(f"sample + tolist:           {(time.monotonic()-t0)*1000/N:.3f} ms")

    # (5) eager forward for comparison
    eng2_graph, eng.graph_runner = eng.graph_runner, None
    t0 = time.monotonic()
    for _ in range(10):
        eng.step()
    torch.cuda.synchronize()
    print(f"eager engine.step():       {(time.monotonic()-t0)*1000/10:.3f} ms")
    eng.graph_runner = eng2_graph


if __name__ == "__main__":
    main()
