"""DeepSeek-V2/V3 MLA + grouped-topk tests (CPU reference path)."""

import json

import pytest
import torch

from gllm_amd.ops import torch_ref as R

DSV2_TINY = {
    "architectures": ["DeepseekV2ForCausalLM"],
    "model_type": "deepseek_v2",
    "hidden_size": 64,
    "intermediate_size": 128,
    "moe_intermediate_size": 48,
    "num_hidden_layers": 3,
    "num_attention_heads": 4,
    "n_routed_experts": 8,
    "n_shared_experts": 1,
    "num_experts_per_tok": 2,
    "first_k_dense_replace": 1,
    "moe_layer_freq": 1,
    "routed_scaling_factor": 1.0,
    "scoring_func": "softmax",
    "topk_method": "group_limited_greedy",
    "n_group": 2,
    "topk_group": 1,
    "norm_topk_prob": True,
    "q_lora_rank": 32,
    "kv_lora_rank": 48,
    "qk_nope_head_dim": 16,
    "qk_rope_head_dim": 8,
    "v_head_dim": 16,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "eos_token_id": 0,
}

DSV3_TINY = {
    **DSV2_TINY,
    "architectures": ["DeepseekV3ForCausalLM"],
    "model_type": "deepseek_v3",
    "topk_method": "noaux_tc",
    "scoring_func": "sigmoid",
    "q_lora_rank": None,
}


def test_grouped_topk_masks_groups():
    torch.manual_seed(0)
    scores = torch.randn(5, 8)
    w, ids = R.grouped_topk(scores, topk=2, n_group=2, topk_group=1,
                            renormalize=True)
    # all selected experts must come from ONE group (groups of 4)
    for t in range(5):
        groups = set(int(i) // 4 for i in ids[t])
        assert len(groups) == 1
    assert torch.allclose(w.sum(-1), torch.ones(5))


def test_grouped_topk_noaux_bias_shifts_selection():
    torch.manual_seed(1)
    scores = torch.zeros(1, 8)
    bias = torch.zeros(8)
    bias[5] = 10.0  # force expert 5's group + expert 5
    w, ids = R.grouped_topk(scores, topk=2, n_group=2, topk_group=1,
                            renormalize=False, scoring="sigmoid",
                            e_bias=bias)
    assert 5 in ids[0].tolist()
    # weights come from the UNBIASED sigmoid scores
    assert torch.allclose(w, torch.sigmoid(torch.zeros(1, 2)), atol=1e-5)


def test_mla_paged_attention_matches_dense():
    torch.manual_seed(2)
    H, Dk, Dv, ps = 4, 24, 16, 4
    S = 13
    k_cache = torch.zeros(8, ps, H, Dk)
    v_cache = torch.zeros(8, ps, H, Dv)
    bt = torch.tensor([[1, 2, 3, 4]], dtype=torch.int32)
    k = torch.randn(S, H, Dk)
    v = torch.randn(S, H, Dv)
    for pos in range(S):
        k_cache[bt[0, pos // ps], pos % ps] = k[pos]
        v_cache[bt[0, pos // ps], pos % ps] = v[pos]
    q = torch.randn(S, H, Dk)
    out = R.mla_paged_attention(q, k_cache, v_cache, bt,
                                torch.tensor([S], dtype=torch.int32),
                                torch.tensor([0, S], dtype=torch.int32),
                                0.2)
    # dense reference
    scores = torch.einsum("lhd,shd->hls", q.float(), k.float()) * 0.2
    mask = torch.ones(S, S, dtype=torch.bool).tril()
    scores.masked_fill_(~mask.unsqueeze(0), float("-inf"))
    ref = torch.einsum("hls,shd->lhd", torch.softmax(scores, -1), v.float())
    assert torch.allclose(out.float(), ref, atol=1e-4)


@pytest.mark.parametrize("cfg_json", [DSV2_TINY, DSV3_TINY],
                         ids=["dsv2", "dsv3"])
def test_deepseek_generates(tmp_path, cfg_json):
    d = tmp_path / cfg_json["model_type"]
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4,
                       enable_prefix_caching=True)
    llm = LLM(config=cfg, num_pages_override=128)
    sp = SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)
    out = llm.generate([list(range(1, 25))], [sp])
    assert len(out[0].token_ids) == 5
    # chunked prefill equivalence
    cfg2 = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                        dtype="float32", page_size=4, maxp=8,
                        enable_prefix_caching=True)
    llm2 = LLM(config=cfg2, num_pages_override=128)
    out2 = llm2.generate([list(range(1, 25))], [sp])
    assert out2[0].token_ids == out[0].token_ids


def test_mla_absorbed_equals_decompressed(tmp_path):
    """The latent-MQA absorbed path (default) must emit exactly the
    decompressed per-head path's tokens: q.(W_UK c) == (W_UK^T q).c."""
    import json as _json
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams

    outs = {}
    for mode in ("absorbed", "decompressed"):
        d = tmp_path / mode
        d.mkdir()
        with open(d / "config.json", "w") as f:
            _json.dump(DSV2_TINY, f)
        cfg = EngineConfig(model=str(d), load_format="dummy",
                           device="cpu", dtype="float32", page_size=4,
                           maxp=64, mla_mode=mode,
                           enable_prefix_caching=False)
        llm = LLM(config=cfg, num_pages_override=128)
        if mode == "absorbed":
            # latent cache: 1 head, lora+rope dims; v is a view
            k = llm.runner.k_caches[0]
            v = llm.runner.v_caches[0]
            assert k.shape[2] == 1 and k.shape[3] == 48 + 8
            assert v.data_ptr() == k.data_ptr()  # zero-copy
        sp = [SamplingParams(temperature=0.0, max_tokens=6,
                             ignore_eos=True)] * 2
        res = llm.generate([list(range(1, 20)), [5, 6, 7]], sp)
        outs[mode] = [o.token_ids for o in res]
    assert outs["absorbed"] == outs["decompressed"]
