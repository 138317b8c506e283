"""Standalone vision-encoder server CLI (encoder disaggregation).

Parity: reference entrypoints/encoder_server.py. Run next to one or
more LM api_server instances started with --mm-encoder-addr (or a
shared --discovery-addr):

    python -m gllm_amd.entrypoints.encoder_server \
        --model /path/to/Qwen2-VL --port 29820 \
        [--discovery-addr host:29800]
"""

import argparse


def main():
    p = argparse.ArgumentParser(description="gllm_amd encoder server")
    p.add_argument("--model", type=str, required=True)
    p.add_argument("--host", type=str, default="0.0.0.0")
    p.add_argument("--port", type=int, default=29820)
    p.add_argument("--load-format", choices=["auto", "dummy"],
                   default="auto")
    p.add_argument("--dtype", type=str, default="bfloat16")
    p.add_argument("--discovery-addr", type=str, default=None)
    args = p.parse_args()

    import torch

    from gllm_amd.config import EngineConfig
    from gllm_amd.disagg.encoder_server import run_encoder_server
    cfg = EngineConfig(model=args.model, load_format=args.load_format,
                       dtype=args.dtype,
                       device="cuda" if torch.cuda.is_available()
                       else "cpu")
    run_encoder_server(cfg, args.host, args.port, args.discovery_addr)


if __name__ == "__main__":
    main()
