"""compressed-tensors pack-quantized int4 MoE (Kimi-K2.5 expert format).

Reference: model_loader.py:538-591 normalizes the compressed-tensors
config to an int4-MoE hint (routed experts int4 group-32 symmetric,
dense layers bf16); fused_moe_triton/layer.py:229 consumes
weight_packed/weight_scale. Loading such a checkpoint must reproduce
the outputs of a twin checkpoint holding the dequantized experts."""

import json
import os

import torch

from gllm_amd.layers.quantization.int4 import (dequant_ct_int4,
                                               pack_ct_int4)

CFG = {
    "architectures": ["Qwen2MoeForCausalLM"],
    "model_type": "qwen2_moe",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "num_experts": 4,
    "num_experts_per_tok": 2,
    "moe_intermediate_size": 32,
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
GROUP = 32
PROMPTS = [list(range(1, 18)), [9, 8, 7]]
MAX_TOKENS = 6
CT_QCFG = {
    "quant_method": "compressed-tensors",
    "format": "pack-quantized",
    "config_groups": {
        "group_0": {"weights": {"num_bits": 4, "group_size": GROUP,
                                "symmetric": True, "type": "int"}}},
}


def test_ct_pack_dequant_roundtrip():
    torch.manual_seed(3)
    w = torch.randn(48, 64) * 0.2
    packed, scale = pack_ct_int4(w, GROUP)
    assert packed.dtype == torch.int32 and packed.shape == (48, 8)
    back = dequant_ct_int4(packed, scale, GROUP, torch.float32)
    # symmetric 4-bit: per-group max error is scale/2
    bound = scale.float().repeat_interleave(GROUP, dim=1) * 0.5 + 1e-6
    assert ((back - w).abs() <= bound).all()


def _moe_state_dict():
    g = torch.Generator().manual_seed(321)
    c = CFG
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


def _is_expert_weight(name):
    return ".mlp.experts." in name and name.endswith(".weight")


def _write(d, sd, quantized):
    from safetensors.torch import save_file
    os.makedirs(d, exist_ok=True)
    cfg = dict(CFG)
    if quantized:
        cfg["quantization_config"] = CT_QCFG
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    out = {}
    for name, w in sd.items():
        if quantized and _is_expert_weight(name):
            packed, scale = pack_ct_int4(w, GROUP)
            base = name[:-len(".weight")]
            out[base + ".weight_packed"] = packed
            out[base + ".weight_scale"] = scale
        else:
            out[name] = w.clone()
    save_file(out, os.path.join(d, "model.safetensors"))


def _make_checkpoints(tmp_path):
    sd = _moe_state_dict()
    dq = str(tmp_path / "moe_ct_dq")
    d4 = str(tmp_path / "moe_ct_i4")
    deq_sd = dict(sd)
    for name, w in sd.items():
        if _is_expert_weight(name):
            packed, scale = pack_ct_int4(w, GROUP)
            deq_sd[name] = dequant_ct_int4(packed, scale, GROUP,
                                           torch.float32)
    _write(d4, sd, quantized=True)
    _write(dq, deq_sd, quantized=False)
    return d4, dq


def _gen_tokens(model_dir):
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


def test_ct_int4_moe_checkpoint_equals_dequantized_twin(tmp_path):
    d4, dq = _make_checkpoints(tmp_path)
    assert _gen_tokens(d4) == _gen_tokens(dq)
