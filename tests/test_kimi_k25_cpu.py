"""Kimi-K2.5 tests: MoonViT tower + DeepSeek-V3 backbone, 1-D-position
multimodal merge (CPU)."""

import json

import pytest
import torch

from tests.test_deepseek_cpu import DSV2_TINY

KIMI_TINY = {
    **{k: v for k, v in DSV2_TINY.items()},
    "architectures": ["KimiK25ForConditionalGeneration"],
    "model_type": "kimi_k25",
    "media_placeholder_token_id": 120,
    "vision_config": {
        "vt_hidden_size": 32, "vt_num_attention_heads": 4,
        "vt_num_hidden_layers": 2, "vt_intermediate_size": 48,
        "patch_size": 14, "merge_kernel_size": [2, 2],
        "init_pos_emb_height": 8, "init_pos_emb_width": 8,
        "init_pos_emb_time": 4, "in_channels": 3,
        "mm_hidden_size": 32, "text_hidden_size": 64,
    },
}


def _mk_llm(tmp_path, name="k", maxp=64):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(KIMI_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp,
                       enable_prefix_caching=True)  # auto-disabled
    return LLM(config=cfg, num_pages_override=128)


def _mm(seed=3):
    torch.manual_seed(seed)
    # grid 1x4x4 patches -> (4/2)*(4/2) = 4 merged tokens
    return ([1, 2] + [120] * 4 + [3],
            {"pixel_values": torch.randn(16, 3, 14, 14),
             "grids": [(1, 4, 4)]})


def test_kimi_tower_shapes():
    from gllm_amd.models.kimi_k25_vision import (KimiPatchMerger,
                                                 KimiVisionTower)
    torch.manual_seed(0)
    vc = dict(KIMI_TINY["vision_config"])
    tower = KimiVisionTower(vc)
    for p in tower.parameters():
        if p.dim() > 1:
            torch.nn.init.normal_(p, 0, 0.05)
    merger = KimiPatchMerger(vc)
    items = tower(torch.randn(16 + 32, 3, 14, 14), [(1, 4, 4), (2, 4, 4)])
    assert [it.shape for it in items] == [(4, 4, 32), (4, 4, 32)]
    out = torch.cat(merger(items), dim=0)
    assert out.shape == (8, 64)  # temporal pool: t collapses


def test_kimi_generate_and_image_sensitivity(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path)
    from gllm_amd.core.kv_cache import PrefixMemoryManager
    # prefix caching stays on: image runs get content-hash cache keys
    assert isinstance(llm.runner.memory_manager, PrefixMemoryManager)
    assert not llm.runner.uses_mrope
    toks, mm = _mm()
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    o2 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert len(o1) == 6 and o1 == o2
    _, mm2 = _mm(seed=99)
    o3 = llm.generate([toks], sp, mm_inputs=[mm2])[0].token_ids
    assert o3 != o1, "image pixels must influence generation"


def test_kimi_chunked_prefill_matches_full(tmp_path):
    from gllm_amd.sequence import SamplingParams
    toks, mm = _mm()
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    ref = _mk_llm(tmp_path, name="f").generate(
        [toks], sp, mm_inputs=[mm])[0].token_ids
    out = _mk_llm(tmp_path, name="c", maxp=3).generate(
        [toks], sp, mm_inputs=[mm])[0].token_ids
    assert out == ref
