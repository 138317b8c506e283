"""DeepSeek-V3.2 DSA (lightning indexer + sparse MLA) tests, CPU path.

The correctness oracle (reference deepseek_v32.py:455-470): for any
causal horizon <= index_topk the top-k selects every key, so DSA must
be EXACTLY dense MLA. A tiny index_topk then must actually change the
output (keys really get dropped) while staying deterministic and
chunk-invariant (the paged index-K cache carries across prefill
chunks)."""

import json

import pytest
import torch

from tests.test_deepseek_cpu import DSV2_TINY

DSV32_TINY = {
    **DSV2_TINY,
    "architectures": ["DeepseekV32ForCausalLM"],
    "model_type": "deepseek_v32",
    "index_n_heads": 4,
    "index_head_dim": 16,
    "index_topk": 64,
}


def _mk_llm(tmp_path, cfg_json, name, maxp=64):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp,
                       enable_prefix_caching=False)
    return LLM(config=cfg, num_pages_override=128)


def _gen(llm, prompt, n=5):
    from gllm_amd.sequence import SamplingParams
    sp = [SamplingParams(temperature=0.0, max_tokens=n, ignore_eos=True)]
    return llm.generate([prompt], sp)[0].token_ids


def test_dsa_topk_covering_equals_dense(tmp_path):
    """index_topk >= seq_len: sparse selection covers all keys, so the
    output must EQUAL the same model with the indexer disabled."""
    llm = _mk_llm(tmp_path, DSV32_TINY, "v32")
    model = llm.runner.model
    from gllm_amd.models.deepseek_v32 import DSAMLAAttention
    assert isinstance(model.layers[0].self_attn, DSAMLAAttention)
    assert llm.runner.idx_caches is not None
    prompt = list(range(1, 20))
    sparse = _gen(llm, prompt)

    # disable the indexer -> plain dense MLA
    originals = []
    for layer in model.layers:
        originals.append(layer.self_attn._dsa_select)
        layer.self_attn._dsa_select = \
            (lambda *a, **k: None)
    dense = _gen(llm, prompt)
    for layer, fn in zip(model.layers, originals):
        layer.self_attn._dsa_select = fn
    assert sparse == dense


def test_dsa_small_topk_changes_output_and_is_deterministic(tmp_path):
    cfg = {**DSV32_TINY, "index_topk": 4}
    llm = _mk_llm(tmp_path, cfg, "v32small")
    prompt = list(range(1, 24))
    o1 = _gen(llm, prompt, n=6)
    o2 = _gen(llm, prompt, n=6)
    assert o1 == o2 and len(o1) == 6

    model = llm.runner.model
    originals = [ly.self_attn._dsa_select for ly in model.layers]
    for ly in model.layers:
        ly.self_attn._dsa_select = (lambda *a, **k: None)
    dense = _gen(llm, prompt, n=6)
    for ly, fn in zip(model.layers, originals):
        ly.self_attn._dsa_select = fn
    assert o1 != dense, "topk=4 over a 24-token prompt must drop keys"


def test_dsa_chunked_prefill_matches_full(tmp_path):
    """Index keys written chunk by chunk must give the same selection as
    a single-shot prefill (paged index cache state carry)."""
    cfg = {**DSV32_TINY, "index_topk": 6}
    prompt = list(range(1, 26))
    full = _gen(_mk_llm(tmp_path, cfg, "v32full", maxp=64), prompt)
    chunked = _gen(_mk_llm(tmp_path, cfg, "v32chunk", maxp=5), prompt)
    assert chunked == full


def test_sparse_oracle_mask_semantics():
    """ops.mla_paged_attention with a covering topk_positions equals the
    dense call; with a restricted set it equals manual masked attention."""
    from gllm_amd.ops import torch_ref as R
    torch.manual_seed(3)
    H, Dk, Dv, page = 2, 8, 6, 4
    s_len = 10
    k_cache = torch.randn(8, page, H, Dk)
    v_cache = torch.randn(8, page, H, Dv)
    block_table = torch.tensor([[2, 5, 7]], dtype=torch.int32)
    seq_lens = torch.tensor([s_len], dtype=torch.int32)
    qsl = torch.tensor([0, 1], dtype=torch.int32)
    q = torch.randn(1, H, Dk)

    dense = R.mla_paged_attention(q, k_cache, v_cache, block_table,
                                  seq_lens, qsl, 0.5)
    cover = torch.arange(s_len, dtype=torch.int32).unsqueeze(0)
    sparse_all = R.mla_paged_attention(q, k_cache, v_cache, block_table,
                                       seq_lens, qsl, 0.5,
                                       topk_positions=cover)
    assert torch.allclose(dense, sparse_all, atol=1e-6)

    sel = torch.tensor([[1, 4, 7, -1]], dtype=torch.int32)
    sparse = R.mla_paged_attention(q, k_cache, v_cache, block_table,
                                   seq_lens, qsl, 0.5, topk_positions=sel)
    # manual reference over the 3 selected positions
    pages = block_table[0, :3].long()
    k = k_cache[pages].reshape(-1, H, Dk)[:s_len]
    v = v_cache[pages].reshape(-1, H, Dv)[:s_len]
    idx = torch.tensor([1, 4, 7])
    sc = torch.einsum("hd,shd->hs", q[0].float(), k[idx].float()) * 0.5
    p = torch.softmax(sc, dim=-1)
    ref = torch.einsum("hs,shd->hd", p, v[idx].float())
    assert torch.allclose(sparse[0], ref.to(sparse.dtype), atol=1e-5)
