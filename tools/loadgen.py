"""Open-loop load generator for the HTTP gateway.

Fires `--qps` chat-completion requests per second (uniform arrivals) at a
running gateway for `--duration` seconds and reports req/s, p50/p95 TTFT
(first SSE content chunk) and end-to-end latency through the FULL stack:
uvicorn -> middleware -> fallback loop -> engine -> SSE streaming.

Usage (gateway already running with a local provider):
    python tools/loadgen.py --url http://127.0.0.1:9100 \
        --model llmgateway/llama-8b --qps 50 --duration 30
"""

from __future__ import annotations

import argparse
import asyncio
import json
import time

import httpx


def pct(xs, p):
    if not xs:
        return None
    xs = sorted(xs)
    return round(xs[min(len(xs) - 1, int(p / 100 * len(xs)))], 2)


async def one_request(client, args, results, idx=0):
    # --vary appends a unique tail so only the shared head can prefix-hit
    # (a fully repeated prompt makes the prefix cache absorb ~all prefill)
    content = "x " * args.prompt_words
    if args.vary:
        content += f"request number {idx} " * 4
    payload = {
        "model": args.model,
        "messages": [{"role": "user", "content": content}],
        "max_tokens": args.max_tokens,
        "stream": True,
        "ignore_eos": True,
    }
    t0 = time.monotonic()
    ttft = None
    ntok = 0
    try:
        async with client.stream(
            "POST", f"{args.url}/v1/chat/completions", json=payload
        ) as resp:
            if resp.status_code != 200:
                results["errors"] += 1
                return
            async for line in resp.aiter_lines():
                if not line.startswith("data: "):
                    continue
                data = line[6:]
                if data == "[DONE]":
                    break
                obj = json.loads(data)
                delta = obj.get("choices", [{}])[0].get("delta", {})
                if delta.get("content"):
                    if ttft is None:
                        ttft = (time.monotonic() - t0) * 1e3
                    ntok += 1
    except Exception:
        results["errors"] += 1
        return
    results["done"] += 1
    results["tokens"] += ntok
    if ttft is not None:
        results["ttfts"].append(ttft)
    results["e2es"].append((time.monotonic() - t0) * 1e3)


async def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default="http://127.0.0.1:9100")
    ap.add_argument("--model", default="llmgateway/llama-8b")
    ap.add_argument("--api-key", default="")
    ap.add_argument("--qps", type=float, default=50.0)
    ap.add_argument("--duration", type=float, default=30.0)
    ap.add_argument("--prompt-words", type=int, default=64)
    ap.add_argument("--max-tokens", type=int, default=64)
    ap.add_argument("--vary", action="store_true",
                    help="unique prompt tail per request (limits prefix-cache hits)")
    args = ap.parse_args()

    headers = {"Authorization": f"Bearer {args.api_key}"} if args.api_key else {}
    limits = httpx.Limits(max_connections=2048, max_keepalive_connections=2048)
    results = {"done": 0, "errors": 0, "tokens": 0, "ttfts": [], "e2es": []}
    async with httpx.AsyncClient(headers=headers, limits=limits, timeout=300.0) as client:
        tasks = []
        n = int(args.qps * args.duration)
        interval = 1.0 / args.qps
        t_start = time.monotonic()
        for i in range(n):
            target = t_start + i * interval
            delay = target - time.monotonic()
            if delay > 0:
                await asyncio.sleep(delay)
            tasks.append(asyncio.create_task(one_request(client, args, results, i)))
        await asyncio.gather(*tasks)
        elapsed = time.monotonic() - t_start

    print(json.dumps({
        "offered_qps": args.qps,
        "duration_s": round(elapsed, 2),
        "completed": results["done"],
        "errors": results["errors"],
        "achieved_req_per_s": round(results["done"] / elapsed, 2),
        "p50_ttft_ms": pct(results["ttfts"], 50),
        "p95_ttft_ms": pct(results["ttfts"], 95),
        "p50_e2e_ms": pct(results["e2es"], 50),
        "p95_e2e_ms": pct(results["e2es"], 95),
        "gen_tokens_per_s": round(results["tokens"] / elapsed, 1),
    }))


if __name__ == "__main__":
    asyncio.run(main())
