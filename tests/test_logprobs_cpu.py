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
