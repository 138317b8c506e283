"""Qwen2-VL end-to-end multimodal tests (offline path, CPU)."""

import json

import pytest
import torch

VL_TINY = {
    "architectures": ["Qwen2VLForConditionalGeneration"],
    "model_type": "qwen2_vl",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "vocab_size": 160,
    "image_token_id": 150,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "rope_scaling": {"type": "mrope", "mrope_section": [2, 3, 3]},
    "eos_token_id": 0,
    "vision_config": {
        "depth": 2, "embed_dim": 32, "hidden_size": 64, "num_heads": 4,
        "mlp_ratio": 2.0, "patch_size": 14, "temporal_patch_size": 2,
        "in_channels": 3, "spatial_merge_size": 2,
    },
}


def _mk_llm(tmp_path, maxp=64, name="vl"):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    # vision_config must survive the AutoConfig fallback: write plain json
    with open(d / "config.json", "w") as f:
        json.dump(VL_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    import types

    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp)
    llm = LLM(config=cfg, num_pages_override=128)
    # AutoConfig may deliver vision_config as a dict via SimpleNamespace
    return llm


def _mm_prompt():
    # 2 text + 6 image pads (grid 1x4x6 -> 24 patches -> 6 merged) + text
    toks = [1, 2] + [150] * 6 + [3, 4]
    grids = [(1, 4, 6)]
    px = torch.randn(24, 3 * 2 * 14 * 14)
    return toks, {"pixel_values": px, "grids": grids}


def test_vl_generate_deterministic(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path)
    assert llm.runner.uses_mrope
    toks, mm = _mm_prompt()
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    o2 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert len(o1) == 6 and o1 == o2


def test_vl_image_content_changes_output(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path, name="vl2")
    toks, mm = _mm_prompt()
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    torch.manual_seed(123)
    mm2 = {"pixel_values": torch.randn(24, 3 * 2 * 14 * 14) * 3,
           "grids": [(1, 4, 6)]}
    o2 = llm.generate([toks], sp, mm_inputs=[mm2])[0].token_ids
    assert o1 != o2, "vision embeddings must influence generation"


def test_vl_chunked_prefill_matches_full(tmp_path):
    """Image spans straddling chunk boundaries must merge correctly."""
    from gllm_amd.sequence import SamplingParams
    toks, mm = _mm_prompt()
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    llm_full = _mk_llm(tmp_path, maxp=64, name="vlf")
    ref = llm_full.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    llm_chunk = _mk_llm(tmp_path, maxp=4, name="vlc")  # span crosses chunks
    out = llm_chunk.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert out == ref


def test_vl_text_only_still_works(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path, name="vlt")
    sp = [SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True)]
    out = llm.generate([[1, 2, 3, 4, 5]], sp)[0].token_ids
    assert len(out) == 4


VL25_TINY = {
    **VL_TINY,
    "architectures": ["Qwen2_5_VLForConditionalGeneration"],
    "model_type": "qwen2_5_vl",
    "vision_config": {
        "depth": 2, "hidden_size": 32, "out_hidden_size": 64,
        "num_heads": 4, "intermediate_size": 48, "patch_size": 14,
        "temporal_patch_size": 2, "in_channels": 3,
        "spatial_merge_size": 2, "window_size": 56,
        "fullatt_block_indexes": [1],
    },
}


def test_vl25_generate_with_windowed_tower(tmp_path):
    from gllm_amd.sequence import SamplingParams
    d = tmp_path / "vl25"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(VL25_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64)
    llm = LLM(config=cfg, num_pages_override=128)
    from gllm_amd.models.qwen2_vl_vision import Qwen25VisionTransformer
    assert isinstance(llm.runner.model.visual, Qwen25VisionTransformer)
    toks = [1, 2] + [150] * 16 + [3]
    mm = {"pixel_values": torch.randn(64, 3 * 2 * 14 * 14),
          "grids": [(1, 8, 8)]}
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    o2 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert len(o1) == 5 and o1 == o2


def test_mm_prefix_cache_content_keys(tmp_path):
    """Prefix caching with multimodal prompts: identical image reuses
    pages (hit rate > 0, same output); a different image with the SAME
    token ids must NOT alias the cached pages."""
    from gllm_amd.sequence import SamplingParams
    from gllm_amd.core.kv_cache import PrefixMemoryManager
    llm = _mk_llm(tmp_path, name="vlpfx")
    assert isinstance(llm.runner.memory_manager, PrefixMemoryManager)
    # long text prefix so image pads land beyond the first pages too
    toks = list(range(1, 10)) + [150] * 6 + [3, 4]
    torch.manual_seed(41)
    mm1 = {"pixel_values": torch.randn(24, 3 * 2 * 14 * 14),
           "grids": [(1, 4, 6)]}
    mm2 = {"pixel_values": torch.randn(24, 3 * 2 * 14 * 14) * 2,
           "grids": [(1, 4, 6)]}
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm1])[0].token_ids
    o1b = llm.generate([toks], sp, mm_inputs=[mm1])[0].token_ids
    assert o1b == o1
    assert llm.runner.memory_manager.hit_tokens > 0, \
        "identical mm request must hit the prefix cache"
    o2 = llm.generate([toks], sp, mm_inputs=[mm2])[0].token_ids
    assert o2 != o1, \
        "different pixels with identical token ids must not alias pages"
