"""GDN (Gated DeltaNet) reference ops + hybrid model tests."""

import json

import pytest
import torch

from gllm_amd.ops import gdn_ref as G


def test_gating_formula():
    A_log = torch.tensor([0.0, 1.0])
    a = torch.zeros(3, 2)
    b = torch.zeros(3, 2)
    dt = torch.zeros(2)
    g, beta = G.gdn_gating(A_log, a, b, dt)
    sp0 = torch.nn.functional.softplus(torch.tensor(0.0))
    assert torch.allclose(g[0], torch.tensor([-sp0, -torch.e * sp0]),
                          atol=1e-5)
    assert torch.allclose(beta, torch.full((3, 2), 0.5))


def test_conv_chunked_equals_full():
    torch.manual_seed(0)
    C, K, T = 6, 4, 13
    w = torch.randn(C, K)
    x = torch.randn(T, C)
    st_full = torch.zeros(C, K - 1)
    full = G.causal_conv1d_prefill(x, w, st_full, False)
    # two chunks with state carry
    st = torch.zeros(C, K - 1)
    o1 = G.causal_conv1d_prefill(x[:5], w, st, False)
    o2 = G.causal_conv1d_prefill(x[5:], w, st, True)
    assert torch.allclose(torch.cat([o1, o2]), full, atol=1e-5)
    # decode updates equal the tail
    st2 = torch.zeros(C, K - 1)
    o3 = G.causal_conv1d_prefill(x[:12], w, st2, False)
    o4 = G.causal_conv1d_update(x[12], w, st2)
    assert torch.allclose(o4, full[12], atol=1e-5)
    assert torch.allclose(st2, st, atol=1e-5)


def test_delta_rule_chunked_equals_full():
    torch.manual_seed(1)
    T, Hk, Hv, Dk, Dv = 11, 2, 4, 8, 6
    q = torch.randn(T, Hk, Dk)
    k = torch.randn(T, Hk, Dk)
    v = torch.randn(T, Hv, Dv)
    g = -torch.rand(T, Hv) * 0.3
    beta = torch.sigmoid(torch.randn(T, Hv))
    S_full = torch.zeros(Hv, Dv, Dk)
    full = G.gated_delta_rule(q, k, v, g, beta, 0.5, S_full)
    S = torch.zeros(Hv, Dv, Dk)
    o1 = G.gated_delta_rule(q[:4], k[:4], v[:4], g[:4], beta[:4], 0.5, S)
    o2 = G.gated_delta_rule(q[4:], k[4:], v[4:], g[4:], beta[4:], 0.5, S)
    assert torch.allclose(torch.cat([o1, o2]), full, atol=1e-4)
    assert torch.allclose(S, S_full, atol=1e-4)


HYBRID_TINY = {
    "architectures": ["Qwen3_5ForCausalLM"],
    "model_type": "qwen3_5",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 4,
    "full_attention_interval": 2,     # 2 GDN + 2 full-attn layers
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
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "eos_token_id": 0,
}


def _mk_llm(tmp_path, maxp=64, name="h"):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(HYBRID_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp,
                       enable_prefix_caching=True)  # auto-disabled
    return LLM(config=cfg, num_pages_override=128)


def test_hybrid_generates_and_chunked_prefill_state_carry(tmp_path):
    from gllm_amd.sequence import SamplingParams
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    prompt = [list(range(1, 30))]
    llm = _mk_llm(tmp_path, maxp=64, name="full")
    ref = llm.generate(prompt, sp)[0].token_ids
    assert len(ref) == 6
    # prefix caching stays ON for hybrid models (snapshot restore)
    from gllm_amd.core.kv_cache import PrefixMemoryManager
    assert isinstance(llm.runner.memory_manager, PrefixMemoryManager)
    # KV allocated only for the full-attention layers
    assert len(llm.runner.k_caches) == 2

    # chunked prefill must carry conv + recurrent state across chunks
    llm2 = _mk_llm(tmp_path, maxp=8, name="chunked")
    out = llm2.generate(prompt, sp)[0].token_ids
    assert out == ref

    # slots released
    assert llm.runner.ssm_pool.alloc.num_used == 0


def test_hybrid_batched_equals_single(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path, name="b")
    sp = SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)
    p1, p2 = [1, 2, 3, 4, 5, 6, 7], [9, 10, 11]
    batched = llm.generate([p1, p2], [sp, sp])
    s1 = llm.generate([p1], [sp])[0].token_ids
    s2 = llm.generate([p2], [sp])[0].token_ids
    assert batched[0].token_ids == s1
    assert batched[1].token_ids == s2


HYBRID_MOE_TINY = {
    **HYBRID_TINY,
    "architectures": ["Qwen3NextForCausalLM"],
    "model_type": "qwen3_next",
    "num_experts": 4,
    "num_experts_per_tok": 2,
    "moe_intermediate_size": 48,
    "shared_expert_intermediate_size": 64,
    "norm_topk_prob": True,
    "decoder_sparse_step": 1,
    "mlp_only_layers": [0],           # layer 0 keeps a dense MLP
}


def test_hybrid_moe_generates(tmp_path):
    """Qwen3-Next-style hybrid: GDN + full attention + routed MoE MLP
    (reference qwen3_5_moe.py)."""
    import json as _json
    d = tmp_path / "hmoe"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        _json.dump(HYBRID_MOE_TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       enable_prefix_caching=False)
    llm = LLM(config=cfg, num_pages_override=128)
    from gllm_amd.models.moe_family import MoEBlock
    from gllm_amd.models.llama_family import DenseMLP
    layers = llm.runner.model.layers
    assert isinstance(layers[0].mlp, DenseMLP)      # mlp_only_layers
    assert isinstance(layers[1].mlp, MoEBlock)
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    o1 = llm.generate([[1, 2, 3, 4, 5, 6]], sp)[0].token_ids
    o2 = llm.generate([[1, 2, 3, 4, 5, 6]], sp)[0].token_ids
    assert len(o1) == 5 and o1 == o2


def test_chunked_delta_rule_equals_sequential():
    """The chunk-parallel WY form must match the sequential recurrence
    (state carry included) across chunk sizes and state carry-in."""
    import torch
    from gllm_amd.ops.gdn_ref import (gated_delta_rule,
                                      gated_delta_rule_chunked)
    torch.manual_seed(1)
    T, Hk, Dk, Hv, Dv = 53, 2, 16, 4, 8
    q = torch.randn(T, Hk, Dk)
    k = torch.randn(T, Hk, Dk)
    v = torch.randn(T, Hv, Dv)
    g = -torch.rand(T, Hv)
    beta = torch.sigmoid(torch.randn(T, Hv))
    for chunk in (1, 7, 16, 64, 128):
        s_ref = torch.randn(Hv, Dv, Dk)
        s_chk = s_ref.clone()
        o_ref = gated_delta_rule(q, k, v, g, beta, 0.25, s_ref)
        o_chk = gated_delta_rule_chunked(q, k, v, g, beta, 0.25, s_chk,
                                         chunk=chunk)
        assert torch.allclose(o_ref, o_chk, atol=1e-4), chunk
        assert torch.allclose(s_ref, s_chk, atol=1e-4), chunk


def test_hybrid_checkpoint_loading(tmp_path):
    """Real-checkpoint mapping: fused in_proj_qkvz/ba split, conv1d /
    A_log / dt_bias, gated q_proj into the fused qkv, dense MLP +
    norms. Loads a synthetic safetensors checkpoint and checks the
    parameters landed where the forward reads them."""
    import os
    import torch
    d = tmp_path / "ckpt"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(HYBRID_TINY, f)
    g = torch.Generator().manual_seed(11)
    H = HYBRID_TINY["hidden_size"]
    I = HYBRID_TINY["intermediate_size"]
    V = HYBRID_TINY["vocab_size"]
    hd = HYBRID_TINY["head_dim"]
    nh = HYBRID_TINY["num_attention_heads"]
    nkv = HYBRID_TINY["num_key_value_heads"]
    nvh = HYBRID_TINY["linear_num_value_heads"]
    nkh = HYBRID_TINY["linear_num_key_heads"]
    dk = HYBRID_TINY["linear_key_head_dim"]
    dv = HYBRID_TINY["linear_value_head_dim"]
    K = HYBRID_TINY["linear_conv_kernel_dim"]
    key_dim, value_dim = nkh * dk, nvh * dv
    conv_dim = 2 * key_dim + value_dim

    def rnd(*s):
        return torch.randn(*s, generator=g) * 0.05

    sd = {"model.embed_tokens.weight": rnd(V, H),
          "model.norm.weight": torch.ones(H) + rnd(H) * 0.01,
          "lm_head.weight": rnd(V, H)}
    for L in range(HYBRID_TINY["num_hidden_layers"]):
        p = f"model.layers.{L}."
        if (L + 1) % HYBRID_TINY["full_attention_interval"]:
            la = p + "linear_attn."
            sd[la + "in_proj_qkvz.weight"] = rnd(
                2 * key_dim + 2 * value_dim, H)
            sd[la + "in_proj_ba.weight"] = rnd(2 * nvh, H)
            sd[la + "conv1d.weight"] = rnd(conv_dim, 1, K)
            sd[la + "A_log"] = rnd(nvh).abs()
            sd[la + "dt_bias"] = rnd(nvh)
            sd[la + "norm.weight"] = torch.ones(dv) + rnd(dv) * 0.01
            sd[la + "out_proj.weight"] = rnd(H, value_dim)
        else:
            sa = p + "self_attn."
            sd[sa + "q_proj.weight"] = rnd(nh * 2 * hd, H)  # q|gate
            sd[sa + "k_proj.weight"] = rnd(nkv * hd, H)
            sd[sa + "v_proj.weight"] = rnd(nkv * hd, H)
            sd[sa + "o_proj.weight"] = rnd(H, nh * hd)
            sd[sa + "q_norm.weight"] = torch.ones(hd)
            sd[sa + "k_norm.weight"] = torch.ones(hd)
        sd[p + "mlp.gate_proj.weight"] = rnd(I, H)
        sd[p + "mlp.up_proj.weight"] = rnd(I, H)
        sd[p + "mlp.down_proj.weight"] = rnd(H, I)
        sd[p + "input_layernorm.weight"] = torch.ones(H)
        sd[p + "post_attention_layernorm.weight"] = torch.ones(H)
    from safetensors.torch import save_file
    save_file(sd, os.path.join(str(d), "model.safetensors"))

    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="auto", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       enable_prefix_caching=False)
    llm = LLM(config=cfg, num_pages_override=128)
    la0 = llm.runner.model.layers[0].linear_attn
    assert torch.allclose(la0.in_proj_qkvz.weight,
                          sd["model.layers.0.linear_attn"
                             ".in_proj_qkvz.weight"])
    assert torch.allclose(la0.A_log,
                          sd["model.layers.0.linear_attn.A_log"])
    assert torch.allclose(
        la0.conv1d_weight,
        sd["model.layers.0.linear_attn.conv1d.weight"].reshape(
            conv_dim, K))
    at1 = llm.runner.model.layers[1].self_attn
    assert torch.allclose(
        at1.qkv_proj.weight[:nh * 2 * hd],
        sd["model.layers.1.self_attn.q_proj.weight"])
    sp = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    o1 = llm.generate([[1, 2, 3, 4, 5, 6, 7]], sp)[0].token_ids
    o2 = llm.generate([[1, 2, 3, 4, 5, 6, 7]], sp)[0].token_ids
    assert len(o1) == 5 and o1 == o2


def test_hybrid_prefix_cache_state_restore(tmp_path):
    """Prefix caching for hybrid models: a repeated prompt must hit the
    cache (pages reused AND recurrent state restored from the boundary
    snapshot) and emit exactly the cold-run tokens; a different prompt
    with the same length must not be affected."""
    from gllm_amd.sequence import SamplingParams
    from gllm_amd.core.kv_cache import PrefixMemoryManager
    llm = _mk_llm(tmp_path, maxp=8, name="pfx")  # page 4, chunks of 8
    mgr = llm.runner.memory_manager
    assert isinstance(mgr, PrefixMemoryManager)
    prompt = list(range(1, 30))
    sp = [SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)]
    cold = llm.generate([prompt], sp)[0].token_ids
    assert llm.runner.ssm_pool.snapshots, "no snapshots taken"
    warm = llm.generate([prompt], sp)[0].token_ids
    assert warm == cold
    assert mgr.hit_tokens > 0, "repeated hybrid prompt must hit"
    # unrelated prompt unaffected
    other = [5] * 29
    o1 = llm.generate([other], sp)[0].token_ids
    o2 = llm.generate([other], sp)[0].token_ids
    assert o1 == o2


def test_hybrid_prefix_multiturn_reuse(tmp_path):
    """Multi-turn pattern: prompt2 = prompt1 + generated + more text
    hits the decode-extended pages with state restore."""
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path, maxp=64, name="mt")
    sp = [SamplingParams(temperature=0.0, max_tokens=7, ignore_eos=True)]
    p1 = list(range(1, 26))
    out1 = llm.generate([p1], sp)[0].token_ids
    p2 = p1 + out1 + [9, 8, 7]
    hits_before = llm.runner.memory_manager.hit_tokens
    out2 = llm.generate([p2], sp)[0].token_ids
    # cold reference from a fresh engine
    llm2 = _mk_llm(tmp_path, maxp=64, name="mt2")
    llm2.generate([p1], sp)  # warm nothing relevant; fresh check below
    cold = _mk_llm(tmp_path, maxp=64, name="mt3").generate(
        [p2], sp)[0].token_ids
    assert out2 == cold
    assert llm.runner.memory_manager.hit_tokens > hits_before


def test_batched_chunked_matches_ragged():
    """Padded batched WY (beta=0 padding) == per-seq chunked == the
    sequential recurrence, including state carry."""
    import torch
    from gllm_amd.ops import gdn_ref
    torch.manual_seed(3)
    Hk, Hv, Dk, Dv = 2, 4, 128, 64
    lens = [70, 1, 130]
    B, Tmax = len(lens), max(lens)
    scale = Dk ** -0.5
    qb = torch.zeros(B, Tmax, Hk, Dk)
    kb = torch.zeros(B, Tmax, Hk, Dk)
    vb = torch.zeros(B, Tmax, Hv, Dv)
    gb = torch.zeros(B, Tmax, Hv)
    bb = torch.zeros(B, Tmax, Hv)
    per = []
    for i, n in enumerate(lens):
        q = torch.randn(n, Hk, Dk)
        k = torch.randn(n, Hk, Dk)
        v = torch.randn(n, Hv, Dv) / 4
        g = -torch.rand(n, Hv) * 0.1
        b = torch.rand(n, Hv)
        qb[i, :n], kb[i, :n], vb[i, :n] = q, k, v
        gb[i, :n], bb[i, :n] = g, b
        per.append((q, k, v, g, b))
    states_b = torch.randn(B, Hv, Dv, Dk) / 8
    states_ref = states_b.clone()
    ob = gdn_ref.gated_delta_rule_chunked_batched(
        qb, kb, vb, gb, bb, scale, states_b, chunk=64)
    for i, n in enumerate(lens):
        q, k, v, g, b = per[i]
        st = states_ref[i].clone()
        o_ref = gdn_ref.gated_delta_rule(q, k, v, g, b, scale, st)
        assert torch.allclose(ob[i, :n].float(), o_ref.float(),
                              atol=1e-3, rtol=1e-3), i
        assert torch.allclose(states_b[i], st, atol=1e-3, rtol=1e-3), i
