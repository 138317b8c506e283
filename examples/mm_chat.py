"""Multimodal chat against a running api_server (Qwen-VL family).

    python -m gllm_amd.entrypoints.api_server --model Qwen2-VL-7B ...
    python examples/mm_chat.py --image photo.png --prompt "Describe it"

Images are sent as base64 data: URLs (this deployment has no egress,
so http image URLs are rejected server-side)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import base64
import json
import urllib.request


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--host", default="http://127.0.0.1:8000")
    p.add_argument("--image", required=True)
    p.add_argument("--prompt", default="Describe this image.")
    p.add_argument("--max-tokens", type=int, default=128)
    args = p.parse_args()

    with open(args.image, "rb") as f:
        b64 = base64.b64encode(f.read()).decode()
    body = {
        "messages": [{"role": "user", "content": [
            {"type": "image_url",
             "image_url": {"url": f"data:image/png;base64,{b64}"}},
            {"type": "text", "text": args.prompt},
        ]}],
        "max_tokens": args.max_tokens,
        "temperature": 0.0,
    }
    req = urllib.request.Request(
        f"{args.host}/v1/chat/completions",
        data=json.dumps(body).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req) as r:
        out = json.load(r)
    print(out["choices"][0]["message"]["content"])


if __name__ == "__main__":
    main()
