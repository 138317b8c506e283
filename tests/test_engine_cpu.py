"""End-to-end CPU engine tests: tiny Qwen2, dummy weights, greedy decode
(config 1 of BASELINE.json: plumbing without a GPU)."""

import pytest
import torch

from gllm_amd.engine.llm import LLM
from gllm_amd.sequence import SamplingParams


@pytest.fixture()
def llm(tiny_config):
    return LLM(config=tiny_config, num_pages_override=128)


def greedy(n):
    return SamplingParams(temperature=0.0, max_tokens=n, ignore_eos=True)


def test_generate_greedy_deterministic(llm):
    prompts = [[1, 2, 3, 4, 5], [7, 8, 9]]
    out1 = llm.generate(prompts, [greedy(8), greedy(8)])
    out2 = llm.generate(prompts, [greedy(8), greedy(8)])
    assert [o.token_ids for o in out1] == [o.token_ids for o in out2]
    assert all(len(o.token_ids) == 8 for o in out1)


def test_chunked_prefill_matches_full_prefill(tiny_config):
    prompt = list(range(1, 40))
    llm_big = LLM(config=tiny_config, num_pages_override=128)
    ref = llm_big.generate([prompt], [greedy(6)])[0].token_ids

    tiny_config.maxp = 8  # force 5 chunks
    llm_small = LLM(config=tiny_config, num_pages_override=128)
    out = llm_small.generate([prompt], [greedy(6)])[0].token_ids
    assert out == ref


def test_prefix_cache_reuse_matches_cold(llm):
    prompt = list(range(1, 30))
    cold = llm.generate([prompt], [greedy(6)])[0].token_ids
    warm = llm.generate([prompt], [greedy(6)])[0].token_ids
    assert warm == cold
    assert llm.runner.memory_manager.get_cache_hit_rate() > 0


def test_batched_equals_single(llm):
    p1, p2 = [1, 2, 3, 4, 5, 6, 7], [9, 10, 11]
    batched = llm.generate([p1, p2], [greedy(5), greedy(5)])
    solo1 = llm.generate([p1], [greedy(5)])[0].token_ids
    solo2 = llm.generate([p2], [greedy(5)])[0].token_ids
    assert batched[0].token_ids == solo1
    assert batched[1].token_ids == solo2


def test_sampling_with_seed_reproducible(llm):
    sp = SamplingParams(temperature=0.8, top_p=0.9, top_k=20, max_tokens=6,
                        seed=1234, ignore_eos=True)
    o1 = llm.generate([[1, 2, 3]], [sp])[0].token_ids
    o2 = llm.generate([[1, 2, 3]], [sp])[0].token_ids
    assert o1 == o2


def test_repetition_penalty_changes_output(llm):
    base = SamplingParams(temperature=0.0, max_tokens=10, ignore_eos=True)
    pen = SamplingParams(temperature=0.0, max_tokens=10, ignore_eos=True,
                         repetition_penalty=5.0)
    o1 = llm.generate([[3, 3, 3, 3]], [base])[0].token_ids
    o2 = llm.generate([[3, 3, 3, 3]], [pen])[0].token_ids
    # with an enormous penalty the argmax token set must differ somewhere
    assert o1 != o2


def test_token_throttling_engine(tiny_config):
    tiny_config.schedule_method = "token_throttling"
    llm = LLM(config=tiny_config, num_pages_override=128)
    outs = llm.generate([list(range(1, 20)), list(range(5, 30))],
                        [greedy(5), greedy(5)])
    assert all(len(o.token_ids) == 5 for o in outs)


def test_logprobs_returned(llm):
    sp = SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True,
                        logprobs=3)
    out = llm.generate([[1, 2, 3, 4]], [sp])[0]
    assert out.logprobs is not None and len(out.logprobs) == 4
    chosen, top = out.logprobs[0]
    assert chosen <= 0.0 and len(top) == 3
    # the chosen (greedy) token must be the top-1 entry
    assert out.token_ids[0] in top


CHATGLM_TINY = {
    "architectures": ["ChatGLMModel"],
    "model_type": "chatglm",
    "hidden_size": 64,
    "ffn_hidden_size": 128,
    "num_layers": 2,
    "num_attention_heads": 4,
    "multi_query_attention": True,
    "multi_query_group_num": 2,
    "kv_channels": 16,
    "padded_vocab_size": 128,
    "seq_length": 2048,
    "layernorm_epsilon": 1e-5,
    "add_qkv_bias": True,
    "rope_ratio": 1.0,
    "eos_token_id": 0,
    "vocab_size": 128,
}

MISTRAL_TINY = {
    "architectures": ["MistralForCausalLM"],
    "model_type": "mistral",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "sliding_window": 8,
    "eos_token_id": 0,
}


@pytest.mark.parametrize("cfg_json", [CHATGLM_TINY, MISTRAL_TINY],
                         ids=["chatglm", "mistral_swa"])
def test_more_arches_generate(tmp_path, cfg_json):
    import json
    d = tmp_path / cfg_json["model_type"]
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(cfg_json, f)
    from gllm_amd.config import EngineConfig
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4,
                       enable_prefix_caching=False)
    llm2 = LLM(config=cfg, num_pages_override=128)
    out = llm2.generate([list(range(1, 30))], [greedy(6)])
    assert len(out[0].token_ids) == 6
    out2 = llm2.generate([list(range(1, 30))], [greedy(6)])
    assert out2[0].token_ids == out[0].token_ids


def test_penalty_pool_used_and_freed(llm):
    pen = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True,
                         repetition_penalty=2.0)
    pool = llm.runner.penalty_pool
    free0 = pool.alloc.num_free
    out = llm.generate([[3, 4, 5]], [pen])[0]
    assert len(out.token_ids) == 6
    # slot released with the seq
    assert pool.alloc.num_free == free0
    # mask rows contain prompt + generated tokens for the next alloc
    from gllm_amd.sequence import Sequence
    s = Sequence(123, [3, 4, 5], pen)
    slot = pool.ensure(s)
    assert slot >= 0
    assert pool.mask[slot, 3] == 1 and pool.mask[slot, 4] == 1
    pool.free(s)
