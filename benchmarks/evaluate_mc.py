"""Multiple-choice accuracy eval against a running api_server
(reference: benchmarks/evaluate_mmlu_pro.py pattern, offline-friendly).

Input: a JSONL of {"question": str, "choices": [str...], "answer": int}
(e.g. an MMLU/MMLU-Pro export). Sends few-shot-free prompts, parses the
first A-J letter of the reply, reports accuracy.

    python benchmarks/evaluate_mc.py --data mmlu.jsonl --host http://...:8000
"""
import argparse
import asyncio
import json
import re
import string

import aiohttp

LETTERS = string.ascii_uppercase


def build_prompt(q):
    lines = [q["question"], ""]
    for i, c in enumerate(q["choices"]):
        lines.append(f"{LETTERS[i]}. {c}")
    lines.append("")
    lines.append("Answer with the letter of the correct choice only.")
    return "\n".join(lines)


async def ask(session, host, q, sem):
    async with sem:
        body = {"messages": [{"role": "user", "content": build_prompt(q)}],
                "max_tokens": 8, "temperature": 0.0}
        async with session.post(f"{host}/v1/chat/completions",
                                json=body) as r:
            data = await r.json()
    text = data["choices"][0]["message"]["content"] or ""
    m = re.search(r"[A-J]", text.upper())
    pred = LETTERS.index(m.group(0)) if m else -1
    return pred == q["answer"]


async def main(args):
    with open(args.data) as f:
        qs = [json.loads(l) for l in f if l.strip()]
    if args.limit:
        qs = qs[:args.limit]
    sem = asyncio.Semaphore(args.concurrency)
    async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=3600)) as session:
        results = await asyncio.gather(
            *[ask(session, args.host, q, sem) for q in qs])
    acc = sum(results) / max(1, len(results))
    print(json.dumps({"n": len(results), "accuracy": round(acc, 4)}))


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--data", required=True)
    ap.add_argument("--host", default="http://127.0.0.1:8000")
    ap.add_argument("--concurrency", type=int, default=16)
    ap.add_argument("--limit", type=int, default=0)
    asyncio.run(main(ap.parse_args()))
