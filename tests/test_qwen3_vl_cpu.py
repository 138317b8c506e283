"""Qwen3-VL deepstack tests (CPU)."""

import json

import pytest
import torch

VL3_TINY = {
    "architectures": ["Qwen3VLForConditionalGeneration"],
    "model_type": "qwen3_vl",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 3,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "head_dim": 16,
    "vocab_size": 160,
    "image_token_id": 150,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "rope_scaling": {"type": "mrope", "mrope_section": [2, 3, 3],
                     "mrope_interleaved": True},
    "eos_token_id": 0,
    "vision_config": {
        "depth": 3, "hidden_size": 32, "out_hidden_size": 64,
        "num_heads": 4, "intermediate_size": 48, "patch_size": 14,
        "temporal_patch_size": 2, "in_channels": 3,
        "spatial_merge_size": 2, "num_position_embeddings": 16,
        "deepstack_visual_indexes": [0, 1], "hidden_act": "silu",
    },
}


def _mk_llm(tmp_path, name="v3", maxp=64):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(VL3_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp)
    return LLM(config=cfg, num_pages_override=128)


def _mm():
    # grid 1x4x4 -> 16 patches -> 4 merged tokens
    torch.manual_seed(5)
    return ([1, 2] + [150] * 4 + [3],
            {"pixel_values": torch.randn(16, 3 * 2 * 14 * 14),
             "grids": [(1, 4, 4)]})


def test_tower_multiscale_output(tmp_path):
    import types
    from gllm_amd.models.qwen3_vl_vision import Qwen3VisionTransformer
    torch.manual_seed(0)
    tower = Qwen3VisionTransformer(
        types.SimpleNamespace(**VL3_TINY["vision_config"]))
    px = torch.randn(16, 3 * 2 * 14 * 14)
    out = tower(px, [(1, 4, 4)])
    # [4 merged tokens, out_hidden * (1 + 2 deepstack levels)]
    assert out.shape == (4, 64 * 3)


def test_qwen3_vl_generate_and_deepstack_matters(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path)
    toks, mm = _mm()
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    o2 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert len(o1) == 6 and o1 == o2

    # zero out the deepstack part of the embeddings -> output changes
    emb = llm.runner.model.encode_images(mm["pixel_values"], mm["grids"])
    emb2 = emb.clone()
    emb2[:, 64:] = 0
    o3 = llm.generate([toks], sp,
                      mm_inputs=[{"embeds": emb2.detach(),
                                  "grids": mm["grids"]}])[0].token_ids
    assert o3 != o1, "deepstack levels must influence generation"


def test_qwen3_vl_chunked_prefill_matches_full(tmp_path):
    from gllm_amd.sequence import SamplingParams
    toks, mm = _mm()
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    ref = _mk_llm(tmp_path, name="f").generate(
        [toks], sp, mm_inputs=[mm])[0].token_ids
    out = _mk_llm(tmp_path, name="c", maxp=3).generate(
        [toks], sp, mm_inputs=[mm])[0].token_ids
    assert out == ref


def test_interleaved_mrope_differs_from_sectioned():
    from gllm_amd.layers.mrope import MRotaryEmbedding
    torch.manual_seed(0)
    q = torch.randn(4, 2 * 16)
    k = torch.randn(4, 2 * 16)
    pos = torch.tensor([[0, 1, 2, 3], [0, 1, 1, 2], [0, 2, 1, 3]])
    a = MRotaryEmbedding(16, 16, 128, 10000.0, [2, 3, 3])
    b = MRotaryEmbedding(16, 16, 128, 10000.0, [2, 3, 3],
                         mrope_interleaved=True)
    qa, _ = a(pos, q.clone(), k.clone())
    qb, _ = b(pos, q.clone(), k.clone())
    assert not torch.allclose(qa, qb, atol=1e-6)
    # identical t/h/w streams => both layouts equal plain rope
    pos_eq = torch.stack([torch.arange(4)] * 3)
    qa2, _ = a(pos_eq, q.clone(), k.clone())
    qb2, _ = b(pos_eq, q.clone(), k.clone())
    assert torch.allclose(qa2, qb2, atol=1e-6)


VL35_TINY = {
    "architectures": ["Qwen3_5ForConditionalGeneration"],
    "model_type": "qwen3_5",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 4,
    "full_attention_interval": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "head_dim": 16,
    "attn_output_gate": True,
    "partial_rotary_factor": 0.5,
    "linear_num_value_heads": 4,
    "linear_num_key_heads": 2,
    "linear_key_head_dim": 8,
    "linear_value_head_dim": 8,
    "linear_conv_kernel_dim": 4,
    "vocab_size": 160,
    "image_token_id": 150,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "rope_scaling": {"type": "mrope", "mrope_section": [1, 1, 2],
                     "mrope_interleaved": True},
    "eos_token_id": 0,
    "vision_config": VL3_TINY["vision_config"],
}


def _mk_llm35(tmp_path, name="v35", maxp=64):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(VL35_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp)
    return LLM(config=cfg, num_pages_override=128)


def test_qwen3_5_vl_generate_and_deepstack_matters(tmp_path):
    """Qwen3.5-VL = Qwen3-VL tower + hybrid-GDN text LM (reference
    qwen3_5.py Qwen3_5ForConditionalGeneration): deterministic
    generation with image inputs, deepstack levels influence output."""
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm35(tmp_path)
    toks, mm = _mm()
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    o2 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert len(o1) == 6 and o1 == o2
    emb = llm.runner.model.encode_images(mm["pixel_values"], mm["grids"])
    emb2 = emb.clone()
    emb2[:, 64:] = 0
    o3 = llm.generate([toks], sp,
                      mm_inputs=[{"embeds": emb2.detach(),
                                  "grids": mm["grids"]}])[0].token_ids
    assert o3 != o1, "deepstack levels must influence generation"


def test_qwen3_5_vl_chunked_prefill_matches_full(tmp_path):
    from gllm_amd.sequence import SamplingParams
    toks, mm = _mm()
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    ref = _mk_llm35(tmp_path, name="f35").generate(
        [toks], sp, mm_inputs=[mm])[0].token_ids
    out = _mk_llm35(tmp_path, name="c35", maxp=3).generate(
        [toks], sp, mm_inputs=[mm])[0].token_ids
    assert out == ref


def test_skip_visual_lm_node_accepts_embeds(tmp_path):
    """Encoder-disagg LM node: with skip_visual the tower is not built,
    and requests carrying ready embeddings still generate (reference
    lm_server.py --skip-visual)."""
    d = tmp_path / "sv"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(VL3_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       skip_visual=True)
    llm = LLM(config=cfg, num_pages_override=128)
    assert llm.runner.model.visual is None
    toks, _ = _mm()
    emb = torch.randn(4, 64 * 3)  # 4 merged tokens, 1 + 2 deepstack
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    out = llm.generate([toks], sp,
                       mm_inputs=[{"embeds": emb, "grids": [(1, 4, 4)]}])
    assert len(out[0].token_ids) == 5
