"""DP attention (replica parallelism) tests on CPU/gloo.

Reference parity: worker.py:750-889 (_schedule_forward_dp lockstep with
dummy batches) + models/utils.py:39-96 (dp_ep_moe_routed gather). Here
each DP replica owns its own scheduler+KV; MoE experts span replicas
(EP = DP x TP), so every round runs a dp_meta_barrier and idle replicas
enter 1-token dummy forwards. The test loads a real safetensors MoE
checkpoint and checks a dp=2 run (skewed per-replica workloads, so the
dummy path is exercised) reproduces the single-process outputs exactly.
"""

import json
import multiprocessing as mp
import os

import pytest
import torch

CFG = {
    "architectures": ["MixtralForCausalLM"],
    "model_type": "mixtral",
    "hidden_size": 64,
    "intermediate_size": 64,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "num_local_experts": 4,
    "num_experts_per_tok": 2,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 0,
}

PROMPTS = [list(range(1, 18)), [7, 8, 9, 10], [5, 6, 7, 8, 9, 10, 11]]
# different lengths per replica => one replica runs dummy rounds while
# the other still decodes
MAX_TOKENS = [8, 3, 5]


def _write_checkpoint(d):
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(CFG, f)
    g = torch.Generator().manual_seed(31)
    H, I, V = CFG["hidden_size"], CFG["intermediate_size"], CFG["vocab_size"]
    E = CFG["num_local_experts"]
    hd = H // CFG["num_attention_heads"]
    kv = CFG["num_key_value_heads"] * hd
    sd = {}

    def rnd(*shape):
        return torch.randn(*shape, generator=g) * 0.08

    sd["model.embed_tokens.weight"] = rnd(V, H)
    for L in range(CFG["num_hidden_layers"]):
        p = f"model.layers.{L}."
        sd[p + "self_attn.q_proj.weight"] = rnd(H, H)
        sd[p + "self_attn.k_proj.weight"] = rnd(kv, H)
        sd[p + "self_attn.v_proj.weight"] = rnd(kv, H)
        sd[p + "self_attn.o_proj.weight"] = rnd(H, H)
        m = p + "block_sparse_moe."
        sd[m + "gate.weight"] = rnd(E, H)
        for e in range(E):
            sd[m + f"experts.{e}.w1.weight"] = rnd(I, H)
            sd[m + f"experts.{e}.w3.weight"] = rnd(I, H)
            sd[m + f"experts.{e}.w2.weight"] = rnd(H, I)
        sd[p + "input_layernorm.weight"] = torch.ones(H) + rnd(H) * 0.05
        sd[p + "post_attention_layernorm.weight"] = \
            torch.ones(H) + rnd(H) * 0.05
    sd["model.norm.weight"] = torch.ones(H) + rnd(H) * 0.05
    sd["lm_head.weight"] = rnd(V, H)
    from safetensors.torch import save_file
    save_file(sd, os.path.join(d, "model.safetensors"))


def _mk_cfg(model_dir, dp, use_ep, port):
    from gllm_amd.config import EngineConfig
    return EngineConfig(model=model_dir, load_format="auto", device="cpu",
                        dtype="float32", page_size=4, maxp=64,
                        dp_size=dp, use_ep=use_ep, master_port=port,
                        enable_prefix_caching=False)


def _single_reference(model_dir):
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    llm = LLM(config=_mk_cfg(model_dir, 1, False, 0),
              num_pages_override=128)
    sp = [SamplingParams(temperature=0.0, max_tokens=mt, ignore_eos=True)
          for mt in MAX_TOKENS]
    return [o.token_ids for o in llm.generate(PROMPTS, sp)]


def _run_dp_rank(rank, model_dir, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    eng = PPEngine(_mk_cfg(model_dir, 2, True, port),
                   num_pages_override=128)
    assert eng.dp_size == 2
    # replica takes its round-robin share of the request stream
    mine = [(i, p) for i, p in enumerate(PROMPTS) if i % 2 == eng.dp_rank]
    seqs = [Sequence(i, p, SamplingParams(temperature=0.0,
                                          max_tokens=MAX_TOKENS[i],
                                          ignore_eos=True))
            for i, p in mine]
    eng.add_requests(seqs)
    eng.run_until_done()
    q.put((rank, [(s.seq_id, s.output_token_ids) for s in seqs]))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_ep_moe_equals_single(tmp_path):
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)
    ref = _single_reference(d)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_dp_rank, args=(r, d, 29761, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(2):
        rank, outs = q.get(timeout=240)
        for seq_id, toks in outs:
            got[seq_id] = toks
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert [got[i] for i in range(len(PROMPTS))] == ref


def _run_dp_overlap_rank(rank, model_dir, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.engine.overlap_engine import OverlapEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    eng = OverlapEngine(_mk_cfg(model_dir, 2, True, port),
                        num_pages_override=128)
    mine = [(i, p) for i, p in enumerate(PROMPTS) if i % 2 == eng.dp_rank]
    seqs = [Sequence(i, p, SamplingParams(temperature=0.0,
                                          max_tokens=MAX_TOKENS[i],
                                          ignore_eos=True))
            for i, p in mine]
    eng.add_requests(seqs)
    eng.run_until_done()
    q.put((rank, [(s.seq_id, s.output_token_ids) for s in seqs]))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_overlap_equals_single(tmp_path):
    """DP + launch-first/collect-later overlap (reference
    overlap_worker.py:258-309): lockstep dummy launches keep the MoE
    collectives matched while collection stays local."""
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)
    ref = _single_reference(d)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_dp_overlap_rank,
                         args=(r, d, 29767, q)) for r in range(2)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(2):
        _rank, outs = q.get(timeout=240)
        for sid, toks in outs:
            got[sid] = toks
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert [got[i] for i in range(len(PROMPTS))] == ref


def _run_dp_fuzz_rank(rank, model_dir, port, prompts, max_tokens, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    eng = PPEngine(_mk_cfg(model_dir, 2, True, port),
                   num_pages_override=128)
    mine = [(i, p) for i, p in enumerate(prompts) if i % 2 == eng.dp_rank]
    seqs = [Sequence(i, p, SamplingParams(temperature=0.0,
                                          max_tokens=max_tokens[i],
                                          ignore_eos=True))
            for i, p in mine]
    # staggered intake: half the requests arrive after the first rounds
    eng.add_requests(seqs[:len(seqs) // 2 + 1])
    eng.run_until_done(max_steps=3)
    eng.add_requests(seqs[len(seqs) // 2 + 1:])
    eng.run_until_done()
    q.put((rank, [(s.seq_id, s.output_token_ids) for s in seqs]))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("seed,port", [(0, 29763), (1, 29765)])
def test_dp2_fuzz_staggered_load_equals_single(tmp_path, seed, port):
    """Randomized skewed workloads + staggered arrival across replicas:
    dummy-round interleavings must never change results."""
    rng = torch.Generator().manual_seed(seed)

    def ri(lo, hi):
        return int(torch.randint(lo, hi, (1,), generator=rng))

    prompts = [[ri(1, 120) for _ in range(ri(3, 24))] for _ in range(7)]
    max_tokens = [ri(1, 9) for _ in range(7)]
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)

    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    llm = LLM(config=_mk_cfg(d, 1, False, 0), num_pages_override=128)
    sp = [SamplingParams(temperature=0.0, max_tokens=mt, ignore_eos=True)
          for mt in max_tokens]
    ref = [o.token_ids for o in llm.generate(prompts, sp)]

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_dp_fuzz_rank,
                         args=(r, d, port, prompts, max_tokens, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(2):
        _rank, outs = q.get(timeout=240)
        for sid, toks in outs:
            got[sid] = toks
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert [got[i] for i in range(len(prompts))] == ref


def test_abort_unknown_seq_id_is_ignored():
    """An abort for a seq this scheduler never saw (another DP replica's,
    or already finished) must not poison a future recycled seq id."""
    from gllm_amd.config import EngineConfig
    from gllm_amd.core.kv_cache import MemoryManager
    from gllm_amd.core.scheduler import Scheduler
    from gllm_amd.sequence import SamplingParams, Sequence
    cfg = EngineConfig(model="x", device="cpu", page_size=4,
                       enable_prefix_caching=False)
    mm = MemoryManager(num_pages=64, page_size=4)
    sched = Scheduler(cfg, mm)
    sched.abort_seqs([17])          # never seen
    assert 17 not in sched.abort_ids
    seq = Sequence(17, [1, 2, 3], SamplingParams(temperature=0.0,
                                                 max_tokens=4,
                                                 ignore_eos=True))
    sched.add_seqs([seq])
    b = sched.schedule_once()
    assert b is not None and not seq.finish_reason


def _run_dp_pp_rank(rank, model_dir, port, q):
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    cfg = EngineConfig(model=model_dir, load_format="auto", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       pp_size=2, dp_size=2, use_ep=True, master_port=port,
                       enable_prefix_caching=False)
    eng = PPEngine(cfg, num_pages_override=128)
    assert eng.dp_size == 2 and eng.pp_size == 2
    # every stage of a replica adds the replica's round-robin share
    # (replicated schedulers)
    mine = [(i, p) for i, p in enumerate(PROMPTS) if i % 2 == eng.dp_rank]
    seqs = [Sequence(i, p, SamplingParams(temperature=0.0,
                                          max_tokens=MAX_TOKENS[i],
                                          ignore_eos=True))
            for i, p in mine]
    eng.add_requests(seqs)
    eng.run_until_done()
    q.put((rank, [(s.seq_id, s.output_token_ids) for s in seqs]))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_pp2_ep_moe_equals_single(tmp_path):
    """DP across a 2-stage pipeline (reference worker.py
    _schedule_forward_dp_pp): per-stage DP meta barriers, stage-local
    dummy forwards on idle replicas, per-replica token broadcast
    groups. Skewed workloads exercise the dummy path on both stages."""
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)
    ref = _single_reference(d)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_dp_pp_rank, args=(r, d, 29763, q))
             for r in range(4)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(4):
        rank, outs = q.get(timeout=240)
        for seq_id, toks in outs:
            got.setdefault(seq_id, toks)
            assert got[seq_id] == toks, f"rank disagreement on {seq_id}"
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert [got[i] for i in range(len(PROMPTS))] == ref
