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


@pytest.mark.timeout(600)
def test_gpu_vl_mrope_graph_decode(tmp_path):
    """MRoPE (VL) model on GPU with hipGraph decode: the [3, B] position
    buffers must reproduce the eager path exactly (text-only prompts;
    decode positions collapse to a per-seq scalar)."""
    d = tmp_path / "vl"
    d.mkdir()
    cfg_json = {
        "architectures": ["Qwen2VLForConditionalGeneration"],
        "model_type": "qwen2_vl",
        "hidden_size": 512, "intermediate_size": 1024,
        "num_hidden_layers": 2, "num_attention_heads": 8,
        "num_key_value_heads": 2, "vocab_size": 32000,
        "image_token_id": 31999,
        "max_position_embeddings": 4096, "rms_norm_eps": 1e-6,
        "rope_theta": 1000000.0,
        "rope_scaling": {"type": "mrope", "mrope_section": [8, 12, 12]},
        "eos_token_id": 0,
        "vision_config": {
            "depth": 2, "embed_dim": 64, "hidden_size": 512,
            "num_heads": 4, "mlp_ratio": 2.0, "patch_size": 14,
            "temporal_patch_size": 2, "in_channels": 3,
            "spatial_merge_size": 2,
        },
    }
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams

    def run(use_graph):
        cfg = EngineConfig(model=str(d), load_format="dummy",
                           device="cuda", dtype="bfloat16", page_size=16,
                           max_graph_bs=32, gpu_memory_util=0.2,
                           use_graph=use_graph,
                           enable_prefix_caching=False)
        llm = LLM(config=cfg, num_pages_override=512)
        outs = llm.generate(
            [list(range(1, 30)), list(range(5, 45))],
            SamplingParams(temperature=0.0, max_tokens=8,
                           ignore_eos=True))
        del llm
        torch.cuda.empty_cache()
        return [list(o.token_ids) for o in outs]

    got_graph = run(True)
    got_eager = run(False)
    assert got_graph == got_eager


@pytest.mark.timeout(600)
def test_gpu_hybrid_chunked_prefill_matches_full(tmp_path):
    """Hybrid GDN model on GPU: chunked prefill (small maxp -> the WY
    chunk kernel runs across SEVERAL engine iterations with state
    continuation) must match one-shot prefill."""
    d = tmp_path / "hy"
    d.mkdir()
    cfg_json = {
        "architectures": ["Qwen3_5ForCausalLM"], "model_type": "qwen3_5",
        "hidden_size": 512, "intermediate_size": 1024,
        "num_hidden_layers": 4, "full_attention_interval": 2,
        "num_attention_heads": 8, "num_key_value_heads": 2,
        "head_dim": 128, "attn_output_gate": True,
        "partial_rotary_factor": 0.25,
        "linear_num_value_heads": 4, "linear_num_key_heads": 2,
        "linear_key_head_dim": 128, "linear_value_head_dim": 128,
        "linear_conv_kernel_dim": 4,
        "vocab_size": 32000, "max_position_embeddings": 4096,
        "rms_norm_eps": 1e-6, "rope_theta": 10000.0, "eos_token_id": 0,
    }
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams

    def run(maxp):
        cfg = EngineConfig(model=str(d), load_format="dummy",
                           device="cuda", dtype="bfloat16", page_size=16,
                           max_graph_bs=32, gpu_memory_util=0.2,
                           maxp=maxp, enable_prefix_caching=False)
        llm = LLM(config=cfg, num_pages_override=512)
        outs = llm.generate(
            [list(range(1, 150)), list(range(5, 80))],
            SamplingParams(temperature=0.0, max_tokens=8,
                           ignore_eos=True))
        del llm
        torch.cuda.empty_cache()
        return [list(o.token_ids) for o in outs]

    full = run(4096)
    chunked = run(64)
    assert full == chunked
