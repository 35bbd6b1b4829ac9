"""Serving benchmark: req/sec + p50 TTFT at fixed QPS for Llama-3-8B bf16
(BASELINE.json headline metric).

One rank per GPU (weak scaling: each rank runs an identical engine replica —
the gateway's rotation mode, BASELINE configs[3]). Default mode is the
OPEN-LOOP fixed-QPS benchmark the baseline names: each rank schedules
``--steps x --batch`` synthetic requests (random-init weights, random token
prompts) as a Poisson arrival process at ``--qps`` requests/s (``auto`` =
1.1x a capacity estimate from the warmup waves, so the engine is saturated
and the value measures capacity) and drives them through the
continuous-batching engine as they arrive; TTFT is measured from each
request's SCHEDULED arrival (queueing included). A *step* is one
batch-sized cohort of arrivals; the timed region is the whole continuous
run bracketed by a barrier + torch.cuda.synchronize on both sides; value
is the whole-job achieved req/s aggregated over all ranks (max rank
time). ``--mode wave`` keeps the round-1 closed-loop wave benchmark for
comparability.

Launch (the driver does this for N>1):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import random
import statistics
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from llmapigateway_amd.engine import LLMEngine, EngineRequest, SamplingParams  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--batch", type=int, default=256, help="requests per step per GPU")
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--gen-tokens", type=int, default=64)
    p.add_argument("--kv-blocks", type=int, default=None)
    p.add_argument("--kv-dtype", choices=["auto", "fp8"], default="auto",
                   help="fp8 = e4m3 KV cache with per-row scales (halves "
                   "decode KV traffic; bf16 stays the headline default)")
    p.add_argument("--temperature", type=float, default=0.0)
    p.add_argument("--top-p", type=float, default=1.0)
    p.add_argument("--top-k", type=int, default=0)
    p.add_argument(
        "--mode", choices=["qps", "wave"], default="qps",
        help="qps = open-loop Poisson arrivals at --qps (the BASELINE "
        "metric); wave = round-1 closed-loop waves",
    )
    p.add_argument("--debug-stats", action="store_true")
    p.add_argument(
        "--qps", default="auto",
        help="offered request rate per GPU for --mode qps; 'auto' runs an "
        "untimed open-loop calibration segment after the wave warmups and "
        "offers exactly the achieved open-loop capacity, so the timed "
        "region is saturated with a bounded queue (stable TTFT)",
    )
    return p.parse_args()


def make_prompts(batch: int, prompt_len: int, vocab: int, seed: int):
    g = torch.Generator().manual_seed(seed)
    toks = torch.randint(3, min(vocab, 4096), (batch, prompt_len), generator=g)
    return [t.tolist() for t in toks]


def run_wave(engine: LLMEngine, prompts, gen_tokens: int, sp: dict = {}):
    t0 = time.monotonic()
    reqs = [
        EngineRequest(p, SamplingParams(max_tokens=gen_tokens, ignore_eos=True, **sp))
        for p in prompts
    ]
    for r in reqs:
        engine.add_request(r)
    while any(r.state in ("waiting", "running") for r in reqs):
        engine.step()
    if engine.device.type == "cuda":
        torch.cuda.synchronize(engine.device)
    ttfts = [
        (r.first_token_time - t0) * 1000.0 for r in reqs if r.first_token_time
    ]
    assert all(r.state == "finished" for r in reqs)
    assert all(len(r.out_ids) == gen_tokens for r in reqs)
    return ttfts


def run_open_loop(engine: LLMEngine, prompts, gen_tokens: int, qps: float, seed: int,
                  sp: dict = {}):
    """Drive len(prompts) requests as a Poisson process at `qps` req/s.

    Returns (ttfts_ms, elapsed_s). TTFT is measured from each request's
    scheduled arrival time; requests arriving while the engine is busy
    queue exactly as gateway traffic would."""
    rng = random.Random(seed)
    t = 0.0
    arrivals = []
    for _ in prompts:
        arrivals.append(t)
        t += rng.expovariate(qps)
    reqs = [
        EngineRequest(p, SamplingParams(max_tokens=gen_tokens, ignore_eos=True, **sp))
        for p in prompts
    ]
    t0 = time.monotonic()
    nxt = 0
    n = len(reqs)
    while True:
        now = time.monotonic() - t0
        while nxt < n and arrivals[nxt] <= now:
            engine.add_request(reqs[nxt])
            nxt += 1
        if engine.has_work():
            engine.step()
        elif nxt < n:
            time.sleep(min(0.002, max(0.0, arrivals[nxt] - (time.monotonic() - t0))))
        else:
            break
    if engine.device.type == "cuda":
        torch.cuda.synchronize(engine.device)
    elapsed = time.monotonic() - t0
    ttfts = [
        (r.first_token_time - t0 - a) * 1000.0
        for r, a in zip(reqs, arrivals)
        if r.first_token_time
    ]
    assert all(r.state == "finished" for r in reqs)
    return ttfts, elapsed


def main():
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    have_cuda = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if have_cuda else "cpu"

    dist = None
    if world_size > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        backend = "nccl" if have_cuda else "gloo"
        dist.init_process_group(backend=backend)
        if have_cuda:
            torch.cuda.set_device(local_rank)

    dtype = torch.bfloat16 if have_cuda else torch.float32
    engine = LLMEngine(
        model=args.model,
        device=device,
        dtype=dtype,
        num_blocks=args.kv_blocks
        if args.kv_blocks
        else (None if have_cuda else 512),
        max_batch_size=args.batch,
        max_model_len=args.prompt_len + args.gen_tokens + 8,
        seed=1234 + rank,
        kv_dtype=args.kv_dtype,
    )
    vocab = engine.full_config.vocab_size

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if have_cuda:
            torch.cuda.synchronize(engine.device)

    sp = {}
    if args.temperature > 0:
        sp = {"temperature": args.temperature, "top_p": args.top_p, "top_k": args.top_k}

    # warmup (closed-loop waves; also the capacity estimate for --qps auto)
    wave_rates = []
    for w in range(args.warmup):
        tw = time.monotonic()
        run_wave(engine, make_prompts(args.batch, args.prompt_len, vocab, 1000 + w), args.gen_tokens, sp)
        wave_rates.append(args.batch / (time.monotonic() - tw))

    qps = None
    if args.mode == "qps":
        if args.qps == "auto":
            # offer 0.95x the closed-loop wave estimate: open-loop
            # capacity measures ~0.9x wave capacity (continuous
            # prefill/decode interleaving), so this sits right at
            # saturation — achieved req/s reads the open-loop capacity
            # while the queue stays near-critical instead of growing
            # linearly for the whole run (stable, meaningful TTFT)
            cap = max(wave_rates) if wave_rates else 50.0
            qps = cap * 0.95
        else:
            qps = float(args.qps)
        # every rank must offer the same load: agree on rank 0's value
        if dist is not None:
            tq = torch.tensor([qps], dtype=torch.float64)
            if have_cuda:
                tq = tq.to(engine.device)
            dist.broadcast(tq, src=0)
            qps = float(tq.item())

    barrier_sync()
    t_start = time.monotonic()
    ttfts = []
    if args.mode == "qps":
        prompts = []
        for s in range(args.steps):
            prompts += make_prompts(
                args.batch, args.prompt_len, vocab, 2000 + s + rank * 7919
            )
        ttfts, _ = run_open_loop(
            engine, prompts, args.gen_tokens, qps, seed=97 + rank, sp=sp
        )
    else:
        for s in range(args.steps):
            ttfts += run_wave(
                engine, make_prompts(args.batch, args.prompt_len, vocab, 2000 + s + rank * 7919),
                args.gen_tokens, sp,
            )
    barrier_sync()
    elapsed = time.monotonic() - t_start

    # max over ranks
    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if have_cuda:
            t = t.to(engine.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        total_reqs = args.batch * args.steps * world_size
        req_per_sec = total_reqs / elapsed
        ttfts.sort()
        result = {
            "metric": "req/sec @ fixed QPS" if args.mode == "qps" else "req/sec",
            "value": round(req_per_sec, 3),
            "unit": "req/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000.0 / args.steps, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if have_cuda else "fp32",
            "data": "synthetic (random token prompts, random-init weights)",
            "config": {
                "model": args.model,
                "global_batch": args.batch * world_size,
                "seq_len": args.prompt_len + args.gen_tokens,
                "prompt_len": args.prompt_len,
                "gen_tokens": args.gen_tokens,
                "parallelism": f"dp{world_size}",
                "mode": "open-loop-poisson" if args.mode == "qps" else "closed-loop-waves",
                "sampling": sp or "greedy",
                "kv_dtype": args.kv_dtype,
                "qps_offered_per_gpu": round(qps, 1) if qps else None,
                "p50_ttft_ms": round(statistics.median(ttfts), 2) if ttfts else None,
                "p95_ttft_ms": round(ttfts[int(len(ttfts) * 0.95)], 2) if ttfts else None,
                "tokens_per_sec": round(
                    total_reqs * (args.prompt_len + args.gen_tokens) / elapsed, 1
                ),
            },
        }
        print(json.dumps(result))
        if args.debug_stats:
            print("ENGINE_STATS", json.dumps(engine.stats), file=sys.stderr)

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
