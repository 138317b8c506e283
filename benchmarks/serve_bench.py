"""Online serving benchmark against a running api_server: offered load at
a fixed QPS, reports throughput + TTFT/TPOT percentiles
(reference: benchmarks/ aiohttp clients)."""
import argparse
import asyncio
import json
import random
import time

import aiohttp


async def one_request(session, host, prompt_len, max_tokens, results):
    prompt = " ".join(str(random.randint(0, 9999))
                      for _ in range(prompt_len))
    body = {"messages": [{"role": "user", "content": prompt}],
            "max_tokens": max_tokens, "stream": True,
            "ignore_eos": True}
    t0 = time.time()
    ttft = None
    n_tok = 0
    async with session.post(f"{host}/v1/chat/completions",
                            json=body) as resp:
        async for line in resp.content:
            if not line.startswith(b"data: ") or line[6:].startswith(b"[DONE]"):
                continue
            if ttft is None:
                ttft = time.time() - t0
            n_tok += 1
    results.append({"ttft": ttft, "total": time.time() - t0,
                    "tokens": n_tok})


async def main(args):
    results = []
    async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=3600)) as session:
        tasks = []
        t_start = time.time()
        for i in range(args.num_requests):
            tasks.append(asyncio.create_task(one_request(
                session, args.host, args.prompt_len, args.max_tokens,
                results)))
            if args.qps > 0:
                await asyncio.sleep(1.0 / args.qps)
        await asyncio.gather(*tasks)
    wall = time.time() - t_start
    ttfts = sorted(r["ttft"] for r in results if r["ttft"])
    toks = sum(r["tokens"] for r in results)
    p = lambda v, q: v[int(q * (len(v) - 1))] if v else None
    print(json.dumps({
        "num_requests": len(results),
        "qps_offered": args.qps,
        "wall_s": round(wall, 2),
        "output_tokens_per_s": round(toks / wall, 1),
        "ttft_p50_ms": round(p(ttfts, 0.5) * 1000, 1) if ttfts else None,
        "ttft_p99_ms": round(p(ttfts, 0.99) * 1000, 1) if ttfts else None,
    }))


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="http://127.0.0.1:8000")
    ap.add_argument("--num-requests", type=int, default=64)
    ap.add_argument("--qps", type=float, default=4.0)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--max-tokens", type=int, default=128)
    asyncio.run(main(ap.parse_args()))
