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
