#!/bin/bash
# Encoder-disaggregated multimodal serving (docs/multimodal.md):
# the vision tower runs in its own process; LM workers receive ready
# embeddings and never load the tower's weights into their hot path.
set -e
MODEL=${MODEL:-/path/to/Qwen2-VL-7B-Instruct}

# 1. discovery registry
python -m gllm_amd.entrypoints.discovery_server --port 29800 &

# 2. vision encoder (own process / node; registers itself)
python -m gllm_amd.entrypoints.encoder_server \
    --model "$MODEL" --port 29820 --discovery-addr 127.0.0.1:29800 &

# 3. LM api server (resolves the encoder through discovery)
python -m gllm_amd.entrypoints.api_server \
    --model "$MODEL" --port 8000 --discovery-addr 127.0.0.1:29800
