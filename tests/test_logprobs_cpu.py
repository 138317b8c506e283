"""Prompt-logprob correctness (offline LLM, CPU).

Cross-validation: the prompt logprob at position k (model predicting
token P[k] from prefix P[:k]) must agree with the SAMPLED-token top-k
logprobs of a separate run whose whole prompt is P[:k] — both are the
log-softmax of the same logits row, computed by independent code paths
(runtime/model_runner.py _prompt_logprobs vs layers/sampler.py
_with_logprobs)."""

import json

import pytest

TINY = {
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
    "eos_token_id": 0,
}


def _mk_llm(tmp_path, maxp=64, name="lp"):
    d = tmp_path / name
    d.mkdir(exist_ok=True)
    with open(d / "config.json", "w") as f:
        json.dump(TINY, f)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=maxp,
                       enable_prefix_caching=False)
    return LLM(config=cfg, num_pages_override=128)


def test_prompt_logprobs_structure_and_cross_check(tmp_path):
    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path)
    prompt = [3, 17, 42, 99, 5, 66, 12]
    sp = [SamplingParams(temperature=0.0, max_tokens=1, ignore_eos=True,
                         prompt_logprobs=4)]
    out = llm.generate([prompt], sp)[0]
    plp = out.prompt_logprobs
    assert plp is not None and len(plp) == len(prompt) - 1
    for chosen, topk in plp:
        assert chosen <= 0.0 and len(topk) == 4

    # cross-check position k=4 against a run whose prompt is prompt[:4]
    k = 4
    sp2 = [SamplingParams(temperature=0.0, max_tokens=1, ignore_eos=True,
                          logprobs=4)]
    short = llm.generate([prompt[:k]], sp2)[0]
    _, sampled_topk = short.logprobs[0]
    _, prompt_topk = plp[k - 1]
    assert set(sampled_topk) == set(prompt_topk)
    for tok in sampled_topk:
        assert abs(sampled_topk[tok] - prompt_topk[tok]) < 1e-4


def test_prompt_logprobs_chunked_matches_full(tmp_path):
    from gllm_amd.sequence import SamplingParams
    prompt = list(range(1, 20))
    sp = [SamplingParams(temperature=0.0, max_tokens=1, ignore_eos=True,
                         prompt_logprobs=3)]
    full = _mk_llm(tmp_path, maxp=64, name="full").generate(
        [prompt], sp)[0].prompt_logprobs
    chunked = _mk_llm(tmp_path, maxp=4, name="chunk").generate(
        [prompt], sp)[0].prompt_logprobs
    assert len(full) == len(chunked) == len(prompt) - 1
    for (a, ta), (b, tb) in zip(full, chunked):
        assert abs(a - b) < 1e-4
        assert set(ta) == set(tb)


def _run_pp2_lp_rank(rank, model_dir, port, q):
    import os

    import torch
    os.environ.update(RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    torch.set_num_threads(1)
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    cfg = EngineConfig(model=model_dir, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       pp_size=2, master_port=port,
                       enable_prefix_caching=False)
    eng = PPEngine(cfg, num_pages_override=128)
    seqs = [Sequence(0, [3, 17, 42, 99, 5, 66, 12],
                     SamplingParams(temperature=0.0, max_tokens=3,
                                    ignore_eos=True, logprobs=2,
                                    prompt_logprobs=3))]
    eng.add_requests(seqs)
    eng.run_until_done()
    if rank == 0:
        s = seqs[0]
        q.put((s.output_token_ids,
               [(lp, sorted(top.items())) for lp, top in s.out_logprobs],
               s.prompt_logprobs_out))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_pp2_ships_logprobs_to_rank0(tmp_path):
    """Under PP>1 sampling runs on the last stage; per-token logprobs
    ride the token broadcast and prompt logprobs ship at finish, so
    rank 0 serves the same values a single-process run computes."""
    import multiprocessing as mp

    from gllm_amd.sequence import SamplingParams
    llm = _mk_llm(tmp_path)
    sp = [SamplingParams(temperature=0.0, max_tokens=3, ignore_eos=True,
                         logprobs=2, prompt_logprobs=3)]
    ref = llm.generate([[3, 17, 42, 99, 5, 66, 12]], sp)[0]
    ref_seq = ref

    d = tmp_path / "lp"  # same dir _mk_llm used (same dummy seed)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_pp2_lp_rank,
                         args=(r, str(d), 29796, q)) for r in range(2)]
    for p in procs:
        p.start()
    toks, out_lps, plp = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert toks == ref.token_ids
    ref_lps = [(lp, sorted(top.items())) for lp, top in ref.logprobs]
    assert len(out_lps) == len(ref_lps) == 3
    for (a, at), (b, bt) in zip(out_lps, ref_lps):
        assert abs(a - b) < 1e-4
        assert [t for t, _ in at] == [t for t, _ in bt]
    assert plp is not None and len(plp) == len(ref.prompt_logprobs)
    for (ca, ta), (cb, tb) in zip(plp, ref.prompt_logprobs):
        assert abs(ca - cb) < 1e-4 and sorted(ta) == sorted(tb)
