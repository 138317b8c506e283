"""FusedMoE layer tests: routing math, TP-sharding equivalence, EP over
gloo (world 2), and the Mixtral/Qwen2-MoE model forward on CPU."""

import json
import multiprocessing as mp
import os

import pytest
import torch

from gllm_amd.layers.moe.layer import FusedMoE


def _ref_moe(x, w13, w2, gate_logits, topk, renorm=True):
    """Dense reference: full softmax-topk, per-token expert sum."""
    probs = torch.softmax(gate_logits.float(), -1)
    weights, ids = torch.topk(probs, topk, -1)
    if renorm:
        weights = weights / weights.sum(-1, keepdim=True)
    out = torch.zeros_like(x)
    T = x.shape[0]
    for t in range(T):
        for k in range(topk):
            e = int(ids[t, k])
            h = torch.nn.functional.linear(x[t:t + 1], w13[e])
            d = h.shape[-1] // 2
            act = torch.nn.functional.silu(h[:, :d]) * h[:, d:]
            y = torch.nn.functional.linear(act, w2[e])
            out[t] += weights[t, k].to(x.dtype) * y[0]
    return out


def test_fused_moe_matches_dense_reference():
    torch.manual_seed(0)
    E, K, H, I, T = 8, 2, 32, 64, 10
    layer = FusedMoE(E, K, H, I, renormalize=True,
                     params_dtype=torch.float32)
    with torch.no_grad():
        layer.w13_weight.normal_(0, 0.1)
        layer.w2_weight.normal_(0, 0.1)
    x = torch.randn(T, H)
    logits = torch.randn(T, E)
    out = layer(x, logits)
    ref = _ref_moe(x, layer.w13_weight, layer.w2_weight, logits, K)
    assert torch.allclose(out, ref, atol=1e-4), \
        (out - ref).abs().max()


def _run_ep_rank(rank, tmp, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.parallel import init_distributed
    # EP spans dp*tp; tp=2 gives 2 EP ranks with REPLICATED activations
    # (dp=2 would mean DP attention: per-replica batches + MoE gather,
    # covered by tests/test_dp_cpu.py)
    init_distributed(rank=rank, pp_size=1, dp_size=1, tp_size=2,
                     master_port=port)
    torch.manual_seed(0)
    E, K, H, I, T = 8, 2, 32, 64, 6
    layer = FusedMoE(E, K, H, I, renormalize=True, use_ep=True,
                     params_dtype=torch.float32)
    # load full expert weights via the weight_loader path
    g = torch.Generator().manual_seed(7)
    full_w1 = torch.randn(E, I, H, generator=g) * 0.1
    full_w3 = torch.randn(E, I, H, generator=g) * 0.1
    full_w2 = torch.randn(E, H, I, generator=g) * 0.1
    for e in range(E):
        layer._load_w13(layer.w13_weight, full_w1[e], e, 0)
        layer._load_w13(layer.w13_weight, full_w3[e], e, 1)
        layer._load_w2(layer.w2_weight, full_w2[e], e)
    x = torch.randn(T, H, generator=g)
    logits = torch.randn(T, E, generator=g)
    out = layer(x, logits)
    if rank == 0:
        w13 = torch.cat([full_w1, full_w3], dim=1)
        ref = _ref_moe(x, w13, full_w2, logits, K)
        # plain lists: tensors over mp.Queue ride /dev/shm files that
        # vanish when the child exits before the parent reads them
        q.put((out.tolist(), ref.tolist()))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_ep_two_ranks_matches_dense(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_ep_rank,
                         args=(r, str(tmp_path), 29671, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out, ref = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    out, ref = torch.tensor(out), torch.tensor(ref)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


MIXTRAL_TINY = {
    "architectures": ["MixtralForCausalLM"],
    "model_type": "mixtral",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "num_local_experts": 4,
    "num_experts_per_tok": 2,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "eos_token_id": 0,
}

QWEN2MOE_TINY = {
    **MIXTRAL_TINY,
    "architectures": ["Qwen2MoeForCausalLM"],
    "model_type": "qwen2_moe",
    "num_experts": 4,
    "moe_intermediate_size": 48,
    "shared_expert_intermediate_size": 96,
    "norm_topk_prob": False,
    "decoder_sparse_step": 1,
}


@pytest.mark.parametrize("cfg_json", [MIXTRAL_TINY, QWEN2MOE_TINY],
                         ids=["mixtral", "qwen2moe"])
def test_moe_model_generates(tmp_path, cfg_json):
    d = tmp_path / cfg_json["model_type"]
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4,
                       enable_prefix_caching=False)
    llm = LLM(config=cfg, num_pages_override=128)
    outs = llm.generate([[1, 2, 3, 4, 5, 6]],
                        [SamplingParams(temperature=0.0, max_tokens=5,
                                        ignore_eos=True)])
    assert len(outs[0].token_ids) == 5
    outs2 = llm.generate([[1, 2, 3, 4, 5, 6]],
                         [SamplingParams(temperature=0.0, max_tokens=5,
                                         ignore_eos=True)])
    assert outs2[0].token_ids == outs[0].token_ids
