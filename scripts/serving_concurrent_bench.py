"""Multi-stream serving benchmark against a running API node.

Fires N concurrent /v1/chat/completions streams and reports per-stream
TTFB + decode tok/s and the aggregate (a measured
multi-stream serving number). Pure client; run the api+shard pair first
(see scripts/gpu_serving_marathon.sh for the launch recipe).
"""
import argparse
import asyncio
import json
import statistics
import time

import httpx


async def one_stream(client, base, model, max_tokens, prompt, out):
    t0 = time.perf_counter()
    ttfb = None
    n = 0
    async with client.stream(
            "POST", f"{base}/v1/chat/completions",
            json={"model": model, "max_tokens": max_tokens, "stream": True,
                  "messages": [{"role": "user", "content": prompt}]},
            timeout=300) as r:
        async for line in r.aiter_lines():
            if not line.startswith("data: ") or line.endswith("[DONE]"):
                continue
            d = json.loads(line[6:])
            c = d.get("choices", [{}])[0]
            if c.get("delta", {}).get("content"):
                n += 1
                if ttfb is None:
                    ttfb = (time.perf_counter() - t0) * 1e3
    dt = time.perf_counter() - t0
    out.append({"ttfb_ms": ttfb or 0.0, "tokens": n,
                "tok_s": n / max(dt - (ttfb or 0) / 1e3, 1e-6),
                "wall_s": dt})


async def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--base", default="http://127.0.0.1:18080")
    ap.add_argument("--model", default="qwen-2.5-32b-int8-synthetic")
    ap.add_argument("--streams", type=int, default=4)
    ap.add_argument("--max-tokens", type=int, default=64)
    args = ap.parse_args()
    out: list = []
    async with httpx.AsyncClient() as client:
        t0 = time.perf_counter()
        await asyncio.gather(*[
            one_stream(client, args.base, args.model, args.max_tokens,
                       f"stream {i}: tell me a story", out)
            for i in range(args.streams)])
        wall = time.perf_counter() - t0
    ttfbs = sorted(s["ttfb_ms"] for s in out)
    total_tokens = sum(s["tokens"] for s in out)
    print(json.dumps({
        "streams": args.streams,
        "p50_ttfb_ms": round(ttfbs[len(ttfbs) // 2], 1),
        "max_ttfb_ms": round(ttfbs[-1], 1),
        "per_stream_tok_s": [round(s["tok_s"], 1) for s in out],
        "aggregate_tok_s": round(total_tokens / wall, 1),
        "wall_s": round(wall, 2)}))


if __name__ == "__main__":
    asyncio.run(main())
