"""RULER-style long-context eval against a running api_server.

The reference benchmark (benchmarks/evaluate_ruler.py) reads the
pre-generated RULER parquet; this deployment has no dataset egress, so
the tasks are GENERATED synthetically with the same shapes — the scores
are comparable across runs of this script (fixed seed), which is what a
serving-engine eval needs (retrieval fidelity at depth, not absolute
leaderboard numbers):

  niah_single   one needle ("the magic number for X is N") in filler
  niah_multikey K needles, query one
  vt            variable tracking: chained assignments X2 = X1, query
                the set of variables equal to a value

    python benchmarks/evaluate_ruler.py --host http://127.0.0.1:8000 \
        --length 4096 --num-per-task 20 --concurrency 16
"""
import argparse
import asyncio
import json
import random
import string

import aiohttp

FILLER = ("The grass is green. The sky is blue. The sun is yellow. "
          "Here we go. There and back again. ")


def _fill_to(words_target, rng, chunks):
    out = []
    n = 0
    while n < words_target:
        c = rng.choice(chunks)
        out.append(c)
        n += len(c.split())
    return out


def gen_niah(rng, length_words, num_keys=1):
    keys = ["".join(rng.choices(string.ascii_lowercase, k=8))
            for _ in range(num_keys)]
    vals = [str(rng.randint(10**6, 10**7 - 1)) for _ in range(num_keys)]
    filler = _fill_to(length_words, rng, [FILLER])
    for k, v in zip(keys, vals):
        pos = rng.randint(0, len(filler))
        filler.insert(pos,
                      f"The special magic number for {k} is: {v}. ")
    qi = rng.randrange(num_keys)
    prompt = ("".join(filler)
              + f"\nWhat is the special magic number for {keys[qi]}? "
              "Answer with the number only.")
    return prompt, [vals[qi]]


def gen_vt(rng, length_words, hops=4):
    val = str(rng.randint(10**4, 10**5 - 1))
    names = ["VAR" + "".join(rng.choices(string.ascii_uppercase, k=5))
             for _ in range(hops)]
    stmts = [f"{names[0]} = {val}. "]
    for i in range(1, hops):
        stmts.append(f"{names[i]} = {names[i - 1]}. ")
    filler = _fill_to(length_words, rng, [FILLER])
    for s in stmts:
        filler.insert(rng.randint(0, len(filler)), s)
    prompt = ("Memorize the variable assignments.\n" + "".join(filler)
              + f"\nWhich variables are equal to {val}? "
              "List the variable names.")
    return prompt, names


async def ask(session, host, prompt, max_tokens, sem):
    async with sem:
        async with session.post(
                f"{host}/v1/completions",
                json={"prompt": prompt, "max_tokens": max_tokens,
                      "temperature": 0.0}) as r:
            body = await r.json()
            return body["choices"][0]["text"]


async def evaluate(args):
    rng = random.Random(args.seed)
    words = args.length * 3 // 4  # ~0.75 words per token
    tasks = []
    for _ in range(args.num_per_task):
        tasks.append(("niah_single", *gen_niah(rng, words, 1)))
        tasks.append(("niah_multikey", *gen_niah(rng, words, 4)))
        tasks.append(("vt", *gen_vt(rng, words)))
    sem = asyncio.Semaphore(args.concurrency)
    async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=3600)) as s:
        outs = await asyncio.gather(*[
            ask(s, args.host, p, args.max_tokens, sem)
            for _t, p, _a in tasks])
    per_task = {}
    for (task, _p, answers), gen in zip(tasks, outs):
        c, n = per_task.setdefault(task, [0, 0])
        hit = all(a.lower() in gen.lower() for a in answers) \
            if task == "vt" else any(a in gen for a in answers)
        per_task[task] = [c + (1 if hit else 0), n + 1]
    total_c = sum(c for c, _ in per_task.values())
    total_n = sum(n for _, n in per_task.values())
    print(json.dumps({
        "benchmark": "ruler-synthetic",
        "length": args.length,
        "accuracy": round(100 * total_c / max(1, total_n), 2),
        "per_task": {t: round(100 * c / n, 2)
                     for t, (c, n) in sorted(per_task.items())},
    }))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--host", default="http://127.0.0.1:8000")
    p.add_argument("--length", type=int, default=4096,
                   help="approx context tokens")
    p.add_argument("--num-per-task", type=int, default=20)
    p.add_argument("--concurrency", type=int, default=16)
    p.add_argument("--max-tokens", type=int, default=48)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args()
    asyncio.run(evaluate(args))


if __name__ == "__main__":
    main()
