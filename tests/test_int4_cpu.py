"""int4 AWQ/GPTQ weight-only quantization tests.

Same contract as fp8: loading a packed int4 checkpoint must produce
EXACTLY the outputs of a plain checkpoint holding the manually
dequantized weights (pack/unpack conventions in
layers/quantization/int4.py)."""

import json
import os

import pytest
import torch

from gllm_amd.layers.quantization.int4 import (dequant_awq, dequant_gptq,
                                               pack_awq, pack_gptq)

CFG = {
    "architectures": ["Qwen2ForCausalLM"],
    "model_type": "qwen2",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 0,
}
GROUP = 16
QUANT_KEYS = ("q_proj", "k_proj", "v_proj", "o_proj",
              "gate_proj", "up_proj", "down_proj")
PROMPTS = [list(range(1, 18)), [9, 8, 7]]
MAX_TOKENS = 6


@pytest.mark.parametrize("method", ["gptq", "awq"])
def test_pack_dequant_roundtrip(method):
    torch.manual_seed(0)
    w = torch.randn(48, 64)  # [N, K]
    pack = pack_gptq if method == "gptq" else pack_awq
    deq = dequant_gptq if method == "gptq" else dequant_awq
    qw, qz, s = pack(w, GROUP)
    back = deq(qw, qz, s, GROUP, torch.float32)
    assert back.shape == w.shape
    # 4-bit asymmetric per-group quant: max error ~ scale/2
    err = (back - w).abs().max()
    assert err < 0.2, err


def _base_state_dict():
    g = torch.Generator().manual_seed(77)
    H, I, V = CFG["hidden_size"], CFG["intermediate_size"], CFG["vocab_size"]
    hd = H // CFG["num_attention_heads"]
    kv = CFG["num_key_value_heads"] * hd
    sd = {}

    def rnd(*shape):
        return torch.randn(*shape, generator=g) * 0.08

    sd["model.embed_tokens.weight"] = rnd(V, H)
    for L in range(CFG["num_hidden_layers"]):
        p = f"model.layers.{L}."
        sd[p + "self_attn.q_proj.weight"] = rnd(H, H)
        sd[p + "self_attn.q_proj.bias"] = rnd(H)
        sd[p + "self_attn.k_proj.weight"] = rnd(kv, H)
        sd[p + "self_attn.k_proj.bias"] = rnd(kv)
        sd[p + "self_attn.v_proj.weight"] = rnd(kv, H)
        sd[p + "self_attn.v_proj.bias"] = rnd(kv)
        sd[p + "self_attn.o_proj.weight"] = rnd(H, H)
        sd[p + "mlp.gate_proj.weight"] = rnd(I, H)
        sd[p + "mlp.up_proj.weight"] = rnd(I, H)
        sd[p + "mlp.down_proj.weight"] = rnd(H, I)
        sd[p + "input_layernorm.weight"] = torch.ones(H) + rnd(H) * 0.05
        sd[p + "post_attention_layernorm.weight"] = \
            torch.ones(H) + rnd(H) * 0.05
    sd["model.norm.weight"] = torch.ones(H) + rnd(H) * 0.05
    sd["lm_head.weight"] = rnd(V, H)
    return sd


def _is_quantized(name):
    return name.endswith(".weight") and \
        any(f".{k}." in name for k in QUANT_KEYS)


def _write(d, sd, method):
    os.makedirs(d, exist_ok=True)
    cfg = dict(CFG)
    if method:
        cfg["quantization_config"] = {"quant_method": method, "bits": 4,
                                      "group_size": GROUP}
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    from safetensors.torch import save_file
    save_file(sd, os.path.join(d, "model.safetensors"))


def _gen(model_dir):
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=model_dir, load_format="auto", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       enable_prefix_caching=False)
    llm = LLM(config=cfg, num_pages_override=128)
    sp = [SamplingParams(temperature=0.0, max_tokens=MAX_TOKENS,
                         ignore_eos=True)] * len(PROMPTS)
    return [o.token_ids for o in llm.generate(PROMPTS, sp)]


def _run_int4_tp_rank(rank, model_dir, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    cfg = EngineConfig(model=model_dir, load_format="auto", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       tp_size=2, master_port=port,
                       enable_prefix_caching=False)
    eng = PPEngine(cfg, num_pages_override=128)
    seqs = [Sequence(i, p, SamplingParams(temperature=0.0,
                                          max_tokens=MAX_TOKENS,
                                          ignore_eos=True))
            for i, p in enumerate(PROMPTS)]
    eng.add_requests(seqs)
    eng.run_until_done()
    if rank == 0:
        q.put([s.output_token_ids for s in seqs])
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("method,port", [("gptq", 29771), ("awq", 29781)])
def test_int4_tp2_equals_single(tmp_path, method, port):
    """TP=2 sharding of packed int4 tensors (qweight/qzeros/scales
    sliced along stored N for column layers, along packed K rows and
    whole quant groups for row-parallel) must reproduce the
    single-process int4 outputs exactly."""
    import multiprocessing as mp
    pack = pack_gptq if method == "gptq" else pack_awq
    base = _base_state_dict()
    q_sd = {}
    for name, w in base.items():
        if _is_quantized(name):
            qw, qz, s = pack(w, GROUP)
            stem = name[:-len(".weight")]
            q_sd[stem + ".qweight"] = qw
            q_sd[stem + ".qzeros"] = qz
            q_sd[stem + ".scales"] = s
        else:
            q_sd[name] = w
    d = str(tmp_path / f"{method}_tp")
    _write(d, q_sd, method)
    ref = _gen(d)
    ctx = mp.get_context("spawn")
    rq = ctx.Queue()
    procs = [ctx.Process(target=_run_int4_tp_rank,
                         args=(r, d, port, rq)) for r in range(2)]
    for p in procs:
        p.start()
    got = rq.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got == ref


@pytest.mark.parametrize("method", ["gptq", "awq"])
def test_int4_checkpoint_equals_dequantized_twin(tmp_path, method):
    pack = pack_gptq if method == "gptq" else pack_awq
    deq = dequant_gptq if method == "gptq" else dequant_awq
    base = _base_state_dict()
    q_sd, twin_sd = {}, {}
    for name, w in base.items():
        if _is_quantized(name):
            qw, qz, s = pack(w, GROUP)
            stem = name[:-len(".weight")]
            q_sd[stem + ".qweight"] = qw
            q_sd[stem + ".qzeros"] = qz
            q_sd[stem + ".scales"] = s
            twin_sd[name] = deq(qw, qz, s, GROUP, torch.float32)
        else:
            q_sd[name] = w
            twin_sd[name] = w
    dq = str(tmp_path / f"{method}_q")
    dt = str(tmp_path / f"{method}_t")
    _write(dq, q_sd, method)
    _write(dt, twin_sd, None)
    assert _gen(dq) == _gen(dt)


# --------------------------------------------------------- int4 MoE experts
MOE_CFG = {
    "architectures": ["Qwen2MoeForCausalLM"],
    "model_type": "qwen2_moe",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "num_experts": 4,
    "num_experts_per_tok": 2,
    "moe_intermediate_size": 48,
    "shared_expert_intermediate_size": 96,
    "norm_topk_prob": False,
    "decoder_sparse_step": 1,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 0,
}
MOE_GROUP = 8  # moe_intermediate/tp = 24 stays group- and pack-aligned


def _moe_state_dict():
    g = torch.Generator().manual_seed(432)
    c = MOE_CFG
    H, V = c["hidden_size"], c["vocab_size"]
    hd = H // c["num_attention_heads"]
    kv = c["num_key_value_heads"] * hd
    Im, Is = c["moe_intermediate_size"], c["shared_expert_intermediate_size"]
    sd = {}

    def rnd(*shape):
        return torch.randn(*shape, generator=g) * 0.08

    sd["model.embed_tokens.weight"] = rnd(V, H)
    for L in range(c["num_hidden_layers"]):
        p = f"model.layers.{L}."
        sd[p + "self_attn.q_proj.weight"] = rnd(H, H)
        sd[p + "self_attn.q_proj.bias"] = rnd(H)
        sd[p + "self_attn.k_proj.weight"] = rnd(kv, H)
        sd[p + "self_attn.k_proj.bias"] = rnd(kv)
        sd[p + "self_attn.v_proj.weight"] = rnd(kv, H)
        sd[p + "self_attn.v_proj.bias"] = rnd(kv)
        sd[p + "self_attn.o_proj.weight"] = rnd(H, H)
        sd[p + "mlp.gate.weight"] = rnd(c["num_experts"], H)
        for e in range(c["num_experts"]):
            ep = p + f"mlp.experts.{e}."
            sd[ep + "gate_proj.weight"] = rnd(Im, H)
            sd[ep + "up_proj.weight"] = rnd(Im, H)
            sd[ep + "down_proj.weight"] = rnd(H, Im)
        sd[p + "mlp.shared_expert.gate_proj.weight"] = rnd(Is, H)
        sd[p + "mlp.shared_expert.up_proj.weight"] = rnd(Is, H)
        sd[p + "mlp.shared_expert.down_proj.weight"] = rnd(H, Is)
        sd[p + "mlp.shared_expert_gate.weight"] = rnd(1, H)
        sd[p + "input_layernorm.weight"] = torch.ones(H) + rnd(H) * 0.05
        sd[p + "post_attention_layernorm.weight"] = \
            torch.ones(H) + rnd(H) * 0.05
    sd["model.norm.weight"] = torch.ones(H) + rnd(H) * 0.05
    sd["lm_head.weight"] = rnd(V, H)
    return sd


def _moe_is_quantized(name):
    if not name.endswith(".weight"):
        return False
    if name.endswith("shared_expert_gate.weight") or \
            name.endswith("mlp.gate.weight"):
        return False  # routers stay dense
    return any(f".{k}." in name for k in QUANT_KEYS)


def _write_moe(d, sd, method):
    os.makedirs(d, exist_ok=True)
    cfg = dict(MOE_CFG)
    if method:
        cfg["quantization_config"] = {"quant_method": method, "bits": 4,
                                      "group_size": MOE_GROUP}
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    from safetensors.torch import save_file
    save_file(sd, os.path.join(d, "model.safetensors"))


def _make_moe_checkpoints(tmp_path, method):
    pack = pack_gptq if method == "gptq" else pack_awq
    deq = dequant_gptq if method == "gptq" else dequant_awq
    base = _moe_state_dict()
    q_sd, twin_sd = {}, {}
    for name, w in base.items():
        if _moe_is_quantized(name):
            qw, qz, s = pack(w, MOE_GROUP)
            stem = name[:-len(".weight")]
            q_sd[stem + ".qweight"] = qw
            q_sd[stem + ".qzeros"] = qz
            q_sd[stem + ".scales"] = s
            twin_sd[name] = deq(qw, qz, s, MOE_GROUP, torch.float32)
        else:
            q_sd[name] = w
            twin_sd[name] = w
    dq = str(tmp_path / f"{method}_moe_q")
    dt = str(tmp_path / f"{method}_moe_t")
    _write_moe(dq, q_sd, method)
    _write_moe(dt, twin_sd, None)
    return dq, dt


@pytest.mark.parametrize("method", ["gptq", "awq"])
def test_int4_moe_checkpoint_equals_dequantized_twin(tmp_path, method):
    """Packed int4 expert banks (w13/w2 qweight/qzeros/scales routed
    through the experts.<e>.<proj>.<qtensor> names) must load and
    execute exactly like the dequantized twin checkpoint."""
    dq, dt = _make_moe_checkpoints(tmp_path, method)
    assert _gen(dq) == _gen(dt)


@pytest.mark.timeout(300)
@pytest.mark.parametrize("method,port", [("gptq", 29772), ("awq", 29782)])
def test_int4_moe_tp2_equals_single(tmp_path, method, port):
    """TP=2 over the packed int4 MoE checkpoint — covers stored-N
    slicing of the w13 banks and packed-K / whole-group slicing of the
    w2 banks."""
    import multiprocessing as mp
    dq, _ = _make_moe_checkpoints(tmp_path, method)
    ref = _gen(dq)
    ctx = mp.get_context("spawn")
    rq = ctx.Queue()
    procs = [ctx.Process(target=_run_int4_tp_rank,
                         args=(r, dq, port, rq)) for r in range(2)]
    for p in procs:
        p.start()
    got = rq.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got == ref


@pytest.mark.timeout(300)
def test_int4_tp2_kv_replication_equals_single(tmp_path):
    """tp=2 over num_key_value_heads=1: the packed qkv loaders must
    replicate the shared K/V shard (src_rank = tp_rank //
    kv_replication) instead of slicing it."""
    import multiprocessing as mp
    cfg = dict(CFG)
    cfg["num_key_value_heads"] = 1
    base = _base_state_dict()
    hd = CFG["hidden_size"] // CFG["num_attention_heads"]
    g = torch.Generator().manual_seed(99)
    for L in range(CFG["num_hidden_layers"]):
        p = f"model.layers.{L}.self_attn."
        base[p + "k_proj.weight"] = torch.randn(
            hd, CFG["hidden_size"], generator=g) * 0.08
        base[p + "k_proj.bias"] = torch.randn(hd, generator=g) * 0.08
        base[p + "v_proj.weight"] = torch.randn(
            hd, CFG["hidden_size"], generator=g) * 0.08
        base[p + "v_proj.bias"] = torch.randn(hd, generator=g) * 0.08
    q_sd = {}
    for name, w in base.items():
        if _is_quantized(name):
            qw, qz, s = pack_gptq(w, GROUP)
            stem = name[:-len(".weight")]
            q_sd[stem + ".qweight"] = qw
            q_sd[stem + ".qzeros"] = qz
            q_sd[stem + ".scales"] = s
        else:
            q_sd[name] = w
    d = str(tmp_path / "gptq_kvrep")
    os.makedirs(d, exist_ok=True)
    cfg["quantization_config"] = {"quant_method": "gptq", "bits": 4,
                                  "group_size": GROUP}
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    from safetensors.torch import save_file
    save_file(q_sd, os.path.join(d, "model.safetensors"))
    ref = _gen(d)
    ctx = mp.get_context("spawn")
    rq = ctx.Queue()
    procs = [ctx.Process(target=_run_int4_tp_rank,
                         args=(r, d, 29791, rq)) for r in range(2)]
    for p in procs:
        p.start()
    got = rq.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got == ref


def test_pack_dequant_property_sweep():
    """Property sweep over shapes/groups: dequant(pack(w)) stays within
    half a quant step of w for both formats (hypothesis-style grid,
    deterministic seeds)."""
    for seed, (N, K, g) in enumerate([(8, 16, 8), (24, 32, 16),
                                      (16, 64, 32), (40, 48, 8),
                                      (8, 128, 128)]):
        torch.manual_seed(seed)
        w = torch.randn(N, K) * (0.01 + seed)
        for pack, deq in ((pack_gptq, dequant_gptq),
                          (pack_awq, dequant_awq)):
            qw, qz, s = pack(w, g)
            back = deq(qw, qz, s, g, torch.float32)
            # bound: half a step from rounding + up to half a step
            # from the rounded zero-point (plus the GPTQ stored-z-1
            # clamp at z=0) => one full quant step per group
            step = s.float().repeat_interleave(g, dim=0).t()  # [N, K]
            assert ((back - w).abs() <= step + 1e-5).all(), \
                (seed, N, K, g)
