"""DeepSeek family on GPU: absorbed-latent MLA + DSA run the torch
path on device (the gfx950 MLA kernel is round 2 — ops/__init__.py
logs that explicitly; this is NOT a silent kernel fallback)."""

import json

import pytest
import torch

from tests.test_deepseek_cpu import DSV2_TINY
from tests.test_deepseek_v32_cpu import DSV32_TINY

pytestmark = pytest.mark.gpu


def _mk_llm(tmp_path, cfg_json, name, mode="absorbed"):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cuda",
                       dtype="bfloat16", page_size=16, maxp=256,
                       mla_mode=mode, enable_prefix_caching=False,
                       enforce_eager=True)
    return LLM(config=cfg, num_pages_override=256)


def _gen(llm, prompt, n=6):
    from gllm_amd.sequence import SamplingParams
    sp = [SamplingParams(temperature=0.0, max_tokens=n, ignore_eos=True)]
    return llm.generate([prompt], sp)[0].token_ids


@pytest.mark.timeout(300)
def test_deepseek_v2_gpu_both_mla_modes_run(tmp_path):
    """Both MLA modes must run deterministically on GPU. (Exact
    cross-mode equality is an fp32 property — bf16 rounds q·(W c) and
    (Wᵀq)·c at different points — and is asserted by the fp32 CPU test
    test_mla_absorbed_equals_decompressed.)"""
    prompt = list(range(1, 30))
    for mode in ("absorbed", "decompressed"):
        llm = _mk_llm(tmp_path, DSV2_TINY, mode, mode)
        o1 = _gen(llm, prompt)
        o2 = _gen(llm, prompt)
        assert len(o1) == 6 and o1 == o2, mode


@pytest.mark.timeout(300)
def test_deepseek_v32_dsa_gpu(tmp_path):
    cfg = {**DSV32_TINY, "index_topk": 8}
    llm = _mk_llm(tmp_path, cfg, "v32")
    prompt = list(range(1, 40))
    o1 = _gen(llm, prompt)
    o2 = _gen(llm, prompt)
    assert len(o1) == 6 and o1 == o2
    assert llm.runner.idx_caches is not None
    assert llm.runner.idx_caches[0].is_cuda


@pytest.mark.timeout(600)
def test_deepseek_real_latent_dims_hip_mla(tmp_path):
    """Real latent geometry (kv_lora 512 + rope 64 = the 576-dim MQA
    cache) routes through the gfx950 MLA kernel in-engine; decode must
    be deterministic and graphs must match eager."""
    import json as _json
    cfg = {
        "architectures": ["DeepseekV2ForCausalLM"],
        "model_type": "deepseek_v2",
        "hidden_size": 512, "intermediate_size": 1024,
        "num_hidden_layers": 2, "first_k_dense_replace": 2,
        "num_attention_heads": 16, "num_key_value_heads": 16,
        "q_lora_rank": None, "kv_lora_rank": 512,
        "qk_nope_head_dim": 128, "qk_rope_head_dim": 64,
        "v_head_dim": 128, "vocab_size": 4000,
        "max_position_embeddings": 4096, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "eos_token_id": 1,
    }
    d = tmp_path / "dslite"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        _json.dump(cfg, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams

    def run(use_graph):
        c = EngineConfig(model=str(d), load_format="dummy",
                         device="cuda", dtype="bfloat16", page_size=16,
                         use_graph=use_graph, max_graph_bs=32,
                         gpu_memory_util=0.2,
                         enable_prefix_caching=False)
        llm = LLM(config=c, num_pages_override=512)
        outs = llm.generate(
            [list(range(1, 50)), list(range(7, 20))],
            SamplingParams(temperature=0.0, max_tokens=8,
                           ignore_eos=True))
        import torch as _t
        del llm
        _t.cuda.empty_cache()
        return [list(o.token_ids) for o in outs]

    g = run(True)
    e = run(False)
    assert g == e and all(len(t) == 8 for t in g)
