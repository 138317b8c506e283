"""OpenAI-API client against a running api_server
(reference: examples/client.py / chat_client.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse
import json

import requests

if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="http://127.0.0.1:8000")
    ap.add_argument("--stream", action="store_true")
    ap.add_argument("prompt", nargs="?",
                    default="Tell me about the AMD MI355X.")
    args = ap.parse_args()
    body = {"messages": [{"role": "user", "content": args.prompt}],
            "max_tokens": 256, "stream": args.stream}
    url = f"{args.host}/v1/chat/completions"
    if not args.stream:
        r = requests.post(url, json=body, timeout=600)
        print(r.json()["choices"][0]["message"]["content"])
    else:
        with requests.post(url, json=body, stream=True, timeout=600) as r:
            for line in r.iter_lines():
                if not line or not line.startswith(b"data: "):
                    continue
                data = line[6:]
                if data == b"[DONE]":
                    break
                delta = json.loads(data)["choices"][0]["delta"]
                print(delta.get("content") or "", end="", flush=True)
        print()
