"""GPU serving-stack roundtrip: overlap worker + frontend queues."""

import asyncio
import json
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(600)
def test_gpu_overlap_serving_roundtrip(tmp_path):
    d = tmp_path / "tiny"
    d.mkdir()
    cfg_json = {
        "architectures": ["Qwen2ForCausalLM"], "model_type": "qwen2",
        "hidden_size": 1024, "intermediate_size": 2816,
        "num_hidden_layers": 4, "num_attention_heads": 8,
        "num_key_value_heads": 2, "vocab_size": 32000,
        "max_position_embeddings": 8192, "rms_norm_eps": 1e-6,
        "rope_theta": 1000000.0, "eos_token_id": 0,
    }
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cuda",
                       dtype="bfloat16", page_size=16, max_graph_bs=32,
                       gpu_memory_util=0.3,
                       schedule_method="token_throttling")
    eng = AsyncLLMEngine(cfg)
    eng.start()
    try:
        async def run(n):
            chunks = []
            async for c in eng.generate_stream(
                    list(range(1, 40)),
                    SamplingParams(temperature=0.0, max_tokens=n,
                                   ignore_eos=True)):
                chunks.append(c)
            return [c.token_id for c in chunks]
        loop = asyncio.new_event_loop()
        t1 = loop.run_until_complete(run(8))
        t2 = loop.run_until_complete(run(8))
        assert len(t1) == 8 and t1 == t2
    finally:
        eng.stop()
