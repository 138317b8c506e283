"""Tool-calling accuracy eval against a running api_server
(reference: benchmarks/evaluate_bfcl.py pattern).

Input JSONL rows:
  {"question": str, "tools": [openai tool defs...],
   "expected": {"name": str, "arguments": {...}}}

Sends each question with the tool declarations, reads the parsed
``tool_calls`` from the response (the server runs the model-family
tool parser, tokenizers/tool_parsers.py), and scores name match and
argument match (JSON-equality after coercion).

    python benchmarks/evaluate_toolcalls.py --data bfcl.jsonl \
        --host http://127.0.0.1:8000 --concurrency 16
"""
import argparse
import asyncio
import json

import aiohttp


def _args_equal(got, want):
    try:
        if isinstance(got, str):
            got = json.loads(got)
        return got == want
    except Exception:
        return False


async def ask(session, host, row, sem, max_tokens):
    async with sem:
        async with session.post(
                f"{host}/v1/chat/completions",
                json={"messages": [{"role": "user",
                                    "content": row["question"]}],
                      "tools": row["tools"],
                      "temperature": 0.0,
                      "max_tokens": max_tokens}) as r:
            body = await r.json()
    msg = body["choices"][0]["message"]
    calls = msg.get("tool_calls") or []
    exp = row["expected"]
    name_ok = bool(calls) and \
        calls[0]["function"]["name"] == exp["name"]
    args_ok = name_ok and _args_equal(
        calls[0]["function"].get("arguments"), exp.get("arguments", {}))
    return name_ok, args_ok


async def evaluate(args):
    rows = [json.loads(ln) for ln in open(args.data) if ln.strip()]
    sem = asyncio.Semaphore(args.concurrency)
    async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=3600)) as s:
        res = await asyncio.gather(*[
            ask(s, args.host, r, sem, args.max_tokens) for r in rows])
    n = len(res)
    print(json.dumps({
        "benchmark": "toolcalls",
        "n": n,
        "name_accuracy": round(100 * sum(a for a, _ in res) / n, 2),
        "full_accuracy": round(100 * sum(b for _, b in res) / n, 2),
    }))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--data", required=True)
    p.add_argument("--host", default="http://127.0.0.1:8000")
    p.add_argument("--concurrency", type=int, default=16)
    p.add_argument("--max-tokens", type=int, default=256)
    args = p.parse_args()
    asyncio.run(evaluate(args))


if __name__ == "__main__":
    main()
