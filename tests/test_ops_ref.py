"""Torch reference-op sanity tests (these refs are the HIP oracle)."""

import math

import pytest
import torch

from gllm_amd.ops import torch_ref as R


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(5, 32)
    w = torch.randn(32)
    out = R.rmsnorm(x, w, 1e-6)
    ref = x / torch.sqrt((x * x).mean(-1, keepdim=True) + 1e-6) * w
    assert torch.allclose(out, ref, atol=1e-5)


def test_fused_add_rmsnorm_inplace():
    torch.manual_seed(0)
    x = torch.randn(5, 32)
    r = torch.randn(5, 32)
    x0, r0 = x.clone(), r.clone()
    R.fused_add_rmsnorm(x, r, torch.ones(32), 1e-6)
    assert torch.allclose(r, x0 + r0, atol=1e-6)
    assert torch.allclose(x, R.rmsnorm(x0 + r0, torch.ones(32), 1e-6),
                          atol=1e-6)


def test_silu_and_mul():
    x = torch.randn(4, 16)
    out = R.silu_and_mul(x)
    ref = torch.nn.functional.silu(x[:, :8]) * x[:, 8:]
    assert torch.allclose(out, ref, atol=1e-5)


def test_rope_neox_rotates_positions():
    D = 8
    cache_len = 32
    inv = 1.0 / (10000 ** (torch.arange(0, D, 2).float() / D))
    t = torch.arange(cache_len).float()
    freqs = torch.outer(t, inv)
    cache = torch.cat([freqs.cos(), freqs.sin()], -1)
    q = torch.randn(3, 2 * D)
    k = torch.randn(3, 1 * D)
    pos = torch.tensor([0, 1, 5])
    q1, k1 = R.rotary_embedding(pos, q.clone(), k.clone(), D, cache, True)
    # position 0 is identity
    # rebuild manually for position 5, head 0 of q
    x = q[2, :D]
    c, s = freqs[5].cos(), freqs[5].sin()
    o1 = x[:D // 2] * c - x[D // 2:] * s
    o2 = x[D // 2:] * c + x[:D // 2] * s
    assert torch.allclose(q1[0, :D], q[0, :D], atol=1e-6)
    assert torch.allclose(q1[2, :D], torch.cat([o1, o2]), atol=1e-5)


def test_paged_attention_vs_sdpa_prefill_and_decode():
    torch.manual_seed(0)
    Hq, Hkv, D, ps = 4, 2, 16, 4
    num_pages = 32
    k_cache = torch.zeros(num_pages, ps, Hkv, D)
    v_cache = torch.zeros(num_pages, ps, Hkv, D)
    # two seqs: seq0 len 7 all new (prefill), seq1 ctx 6 + 1 new (decode)
    lens_ctx = [0, 6]
    lens_new = [7, 1]
    total_ctx = [c + n for c, n in zip(lens_ctx, lens_new)]
    block_table = torch.tensor([[0, 1, 0], [2, 3, 0]], dtype=torch.int32)
    full_k = [torch.randn(t, Hkv, D) for t in total_ctx]
    full_v = [torch.randn(t, Hkv, D) for t in total_ctx]
    # place ALL context kv into the cache
    for b in range(2):
        for pos in range(total_ctx[b]):
            page = block_table[b, pos // ps]
            k_cache[page, pos % ps] = full_k[b][pos]
            v_cache[page, pos % ps] = full_v[b][pos]
    q = torch.randn(sum(lens_new), Hq, D)
    qsl = torch.tensor([0, 7, 8], dtype=torch.int32)
    out = R.paged_attention(q, k_cache, v_cache, block_table,
                            torch.tensor(total_ctx, dtype=torch.int32),
                            qsl, 1.0 / math.sqrt(D))
    # reference with SDPA per seq
    for b in range(2):
        qs, qe = int(qsl[b]), int(qsl[b + 1])
        qq = q[qs:qe].transpose(0, 1)            # [Hq, L, D]
        kk = full_k[b].repeat_interleave(2, 1).transpose(0, 1)
        vv = full_v[b].repeat_interleave(2, 1).transpose(0, 1)
        L, S = qe - qs, total_ctx[b]
        mask = torch.ones(L, S, dtype=torch.bool).tril(S - L)
        ref = torch.nn.functional.scaled_dot_product_attention(
            qq, kk, vv, attn_mask=mask.unsqueeze(0),
            scale=1.0 / math.sqrt(D))
        assert torch.allclose(out[qs:qe].transpose(0, 1), ref, atol=1e-4), \
            f"seq {b} mismatch"


def test_reshape_and_cache_scatter():
    ps = 4
    k_cache = torch.zeros(8, ps, 2, 8)
    v_cache = torch.zeros(8, ps, 2, 8)
    k = torch.randn(3, 2, 8)
    v = torch.randn(3, 2, 8)
    slots = torch.tensor([5, 6, 17])
    R.reshape_and_cache(k, v, k_cache, v_cache, slots)
    assert torch.equal(k_cache[1, 1], k[0])
    assert torch.equal(k_cache[1, 2], k[1])
    assert torch.equal(v_cache[4, 1], v[2])


def test_topk_softmax_renorm():
    g = torch.tensor([[1.0, 2.0, 3.0, 0.0]])
    w, ids = R.topk_softmax(g, 2)
    assert ids[0].tolist() == [2, 1]
    assert abs(float(w.sum()) - 1.0) < 1e-6


def test_min_p_filter():
    """vLLM-style min_p: tokens below min_p * max_prob are dropped."""
    import torch
    from gllm_amd.layers.sampler import Sampler
    probs = torch.tensor([[0.5, 0.3, 0.15, 0.05],
                          [0.25, 0.25, 0.25, 0.25]])
    out = Sampler._apply_min_p(probs.clone(),
                               torch.tensor([0.4, 0.0]))
    # row 0: threshold 0.2 -> keep 0.5, 0.3; renormalized
    assert out[0, 2] == 0 and out[0, 3] == 0
    assert abs(out[0, 0] - 0.5 / 0.8) < 1e-6
    # row 1: disabled -> unchanged
    assert torch.allclose(out[1], probs[1])


def test_min_p_sampling_end_to_end():
    import torch
    from gllm_amd.layers.sampler import Sampler, SamplingMetadata
    B, V = 4, 50
    torch.manual_seed(0)
    logits = torch.randn(B, V) * 4
    meta = SamplingMetadata(
        temperatures=torch.full((B,), 1.0),
        top_ps=torch.ones(B), top_ks=torch.full((B,), -1,
                                                dtype=torch.int32),
        penalties=torch.ones(B), all_greedy=False, any_penalty=False,
        min_ps=torch.full((B,), 0.9),
        generators=[torch.Generator().manual_seed(i) for i in range(B)])
    out = Sampler()(logits, meta)
    # min_p=0.9 keeps essentially only the argmax at these scales
    assert torch.equal(out.next_tokens, logits.argmax(-1))


def test_presence_frequency_penalties():
    """OpenAI additive penalties over OUTPUT tokens only."""
    import torch
    from gllm_amd.layers.sampler import Sampler, SamplingMetadata
    B, V = 2, 10
    logits = torch.zeros(B, V)
    out_ids0 = torch.tensor([3, 3, 5])   # token 3 twice, 5 once
    meta = SamplingMetadata(
        temperatures=torch.zeros(B),
        top_ps=torch.ones(B),
        top_ks=torch.full((B,), -1, dtype=torch.int32),
        penalties=torch.ones(B), all_greedy=True, any_penalty=True,
        pres_freq_rows=[(0, out_ids0, 0.5, 0.25)])
    out = Sampler._apply_penalties(logits.clone(), meta)
    assert abs(out[0, 3] - (-0.5 - 0.25 * 2)) < 1e-6
    assert abs(out[0, 5] - (-0.5 - 0.25)) < 1e-6
    assert out[0, 0] == 0.0 and torch.all(out[1] == 0)


def test_frequency_penalty_end_to_end(tmp_path):
    """A strong frequency penalty must change greedy output vs none
    (and prompt tokens must NOT be penalized: outputs only)."""
    import json as _json
    import torch
    d = tmp_path / "fp"
    d.mkdir()
    cfg_json = {
        "architectures": ["Qwen2ForCausalLM"], "model_type": "qwen2",
        "hidden_size": 64, "intermediate_size": 128,
        "num_hidden_layers": 2, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 128,
        "max_position_embeddings": 2048, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "eos_token_id": 0,
    }
    with open(d / "config.json", "w") as f:
        _json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       enable_prefix_caching=False)
    llm = LLM(config=cfg, num_pages_override=128)
    prompt = [1, 2, 3, 4, 5]
    base = llm.generate([prompt], [SamplingParams(
        temperature=0.0, max_tokens=10, ignore_eos=True)])[0].token_ids
    pen = llm.generate([prompt], [SamplingParams(
        temperature=0.0, max_tokens=10, ignore_eos=True,
        frequency_penalty=100.0)])[0].token_ids
    assert len(set(pen)) == len(pen), \
        f"freq penalty must forbid repeats, got {pen}"
    assert base != pen or len(set(base)) == len(base)
