"""Real-checkpoint loading + TP/PP sharding equality.

Builds a tiny Qwen2 safetensors checkpoint on disk, then checks that
TP=2 and PP=2 runs (gloo, spawned) produce exactly the single-process
outputs — covering the safetensors loader, stacked-weight mapping, TP
shard copies, vocab-parallel embedding/LM-head and the logits
all-gather.
"""

import json
import multiprocessing as mp
import os

import pytest
import torch

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

PROMPTS = [list(range(1, 22)), [7, 8, 9, 10]]
MAX_TOKENS = 6


def _write_checkpoint(d):
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(CFG, f)
    g = torch.Generator().manual_seed(99)
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
    from safetensors.torch import save_file
    save_file(sd, os.path.join(d, "model.safetensors"))


def _mk_cfg(model_dir, pp, tp, port):
    from gllm_amd.config import EngineConfig
    return EngineConfig(model=model_dir, load_format="auto", device="cpu",
                        dtype="float32", page_size=4, maxp=64,
                        pp_size=pp, tp_size=tp, master_port=port,
                        enable_prefix_caching=False)


def _single_reference(model_dir):
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    llm = LLM(config=_mk_cfg(model_dir, 1, 1, 0), num_pages_override=128)
    sp = [SamplingParams(temperature=0.0, max_tokens=MAX_TOKENS,
                         ignore_eos=True)] * len(PROMPTS)
    return [o.token_ids for o in llm.generate(PROMPTS, sp)]


def _run_rank(rank, model_dir, pp, tp, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    eng = PPEngine(_mk_cfg(model_dir, pp, tp, port),
                   num_pages_override=128)
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
@pytest.mark.parametrize("pp,tp,port", [(2, 1, 29721), (1, 2, 29741)],
                         ids=["pp2", "tp2"])
def test_checkpoint_parallel_equals_single(tmp_path, pp, tp, port):
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)
    ref = _single_reference(d)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_rank,
                         args=(r, d, pp, tp, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got == ref


def _run_overlap_tp_rank(rank, model_dir, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.engine.overlap_engine import OverlapEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    eng = OverlapEngine(_mk_cfg(model_dir, 1, 2, port),
                        num_pages_override=128)
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
def test_overlap_tp2_equals_single(tmp_path):
    """TP=2 under the overlap engine (launch-first/collect-later with
    replicated samplers) must equal the single-process sync run."""
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)
    ref = _single_reference(d)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_overlap_tp_rank,
                         args=(r, d, 29743, q)) for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got == ref
