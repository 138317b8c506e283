"""Offline batch generation (reference: examples/batch_inference.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

from gllm_amd.engine.llm import LLM
from gllm_amd.sequence import SamplingParams

if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True)
    ap.add_argument("--max-tokens", type=int, default=128)
    ap.add_argument("--temperature", type=float, default=0.7)
    args = ap.parse_args()
    llm = LLM(model=args.model)
    prompts = [
        "Explain what a paged KV cache is in two sentences.",
        "Write a haiku about matrix cores.",
    ]
    sp = SamplingParams(temperature=args.temperature,
                       max_tokens=args.max_tokens)
    for out in llm.generate(prompts, sp):
        print(f"--- seq {out.seq_id} ({out.finish_reason}) ---")
        print(out.text)
