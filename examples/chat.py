"""Interactive offline chat (reference: examples/chat.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

from gllm_amd.engine.llm import LLM
from gllm_amd.sequence import SamplingParams

if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True)
    args = ap.parse_args()
    llm = LLM(model=args.model)
    history = []
    while True:
        try:
            user = input("user> ")
        except EOFError:
            break
        history.append({"role": "user", "content": user})
        out = llm.chat(history, SamplingParams(temperature=0.7,
                                               max_tokens=512))
        print("assistant>", out.text)
        history.append({"role": "assistant", "content": out.text})
