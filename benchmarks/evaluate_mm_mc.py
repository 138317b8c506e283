"""Multimodal multiple-choice eval against a running api_server
(reference: benchmarks/evaluate_mmmu.py pattern).

Input JSONL rows:
  {"image": "/path/to/img.png", "question": str,
   "choices": [str, ...], "answer": int}

Each image ships as a base64 data: URL (no egress in this deployment);
the reply's first A-J letter is scored.

    python benchmarks/evaluate_mm_mc.py --data mmmu.jsonl \
        --host http://127.0.0.1:8000 --concurrency 8
"""
import argparse
import asyncio
import base64
import json
import re
import string

import aiohttp

LETTERS = string.ascii_uppercase


def build_content(row):
    with open(row["image"], "rb") as f:
        b64 = base64.b64encode(f.read()).decode()
    lines = [row["question"], ""]
    for i, c in enumerate(row["choices"]):
        lines.append(f"{LETTERS[i]}. {c}")
    lines.append("")
    lines.append("Answer with the letter of the correct choice only.")
    return [
        {"type": "image_url",
         "image_url": {"url": f"data:image/png;base64,{b64}"}},
        {"type": "text", "text": "\n".join(lines)},
    ]


async def ask(session, host, row, sem, max_tokens):
    async with sem:
        async with session.post(
                f"{host}/v1/chat/completions",
                json={"messages": [{"role": "user",
                                    "content": build_content(row)}],
                      "temperature": 0.0,
                      "max_tokens": max_tokens}) as r:
            body = await r.json()
    text = body["choices"][0]["message"]["content"] or ""
    m = re.search(r"\b([A-J])\b", text)
    got = LETTERS.index(m.group(1)) if m else -1
    return got == row["answer"]


async def evaluate(args):
    rows = [json.loads(ln) for ln in open(args.data) if ln.strip()]
    sem = asyncio.Semaphore(args.concurrency)
    async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=3600)) as s:
        oks = await asyncio.gather(*[
            ask(s, args.host, r, sem, args.max_tokens) for r in rows])
    print(json.dumps({
        "benchmark": "mm-multiple-choice",
        "n": len(rows),
        "accuracy": round(100 * sum(oks) / max(1, len(rows)), 2),
    }))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--data", required=True)
    p.add_argument("--host", default="http://127.0.0.1:8000")
    p.add_argument("--concurrency", type=int, default=8)
    p.add_argument("--max-tokens", type=int, default=16)
    args = p.parse_args()
    asyncio.run(evaluate(args))


if __name__ == "__main__":
    main()
