"""Overlap engine must produce exactly the synchronous engine's tokens."""

import json

import pytest

from gllm_amd.config import EngineConfig
from gllm_amd.engine.llm import LLM
from gllm_amd.engine.overlap_engine import OverlapEngine
from gllm_amd.sequence import SamplingParams, Sequence


def _cfg(model_dir, **kw):
    base = dict(model=model_dir, load_format="dummy", device="cpu",
                dtype="float32", page_size=4, maxp=32, maxd=32,
                schedule_method="chunked_prefill",
                enable_prefix_caching=True)
    base.update(kw)
    return EngineConfig(**base)


@pytest.fixture()
def model_dir(tiny_model_dir):
    return tiny_model_dir


def run_overlap(model_dir, prompts, sps, **kw):
    eng = OverlapEngine(_cfg(model_dir, **kw), num_pages_override=128)
    seqs = [Sequence(i, p, sp, eos_token_id=0)
            for i, (p, sp) in enumerate(zip(prompts, sps))]
    eng.add_requests(seqs)
    eng.run_until_done()
    return [s.output_token_ids for s in seqs], eng


def run_sync(model_dir, prompts, sps, **kw):
    llm = LLM(config=_cfg(model_dir, **kw), num_pages_override=128)
    outs = llm.generate(prompts, sps)
    return [o.token_ids for o in outs]


def test_overlap_matches_sync_greedy(model_dir):
    prompts = [list(range(1, 20)), list(range(30, 41)), [5, 6]]
    sps = [SamplingParams(temperature=0.0, max_tokens=7, ignore_eos=True)
           for _ in prompts]
    ref = run_sync(model_dir, prompts, sps)
    got, eng = run_overlap(model_dir, prompts, sps)
    assert got == ref
    # all KV pages returned
    assert eng.runner.memory_manager.allocator.num_used == 0 or \
        eng.runner.memory_manager.get_num_free_pages() > 0


def test_overlap_eos_stops(model_dir):
    # eos_token_id=0 and dummy logits will hit various tokens; use
    # stop_token_ids to force an early stop on whatever token comes first
    prompts = [[1, 2, 3, 4, 5]]
    sps = [SamplingParams(temperature=0.0, max_tokens=50, ignore_eos=True)]
    ref = run_sync(model_dir, prompts, sps)
    # pick the 3rd generated token as a stop token
    stop_tok = ref[0][2]
    sps2 = [SamplingParams(temperature=0.0, max_tokens=50,
                           stop_token_ids=[stop_tok], ignore_eos=True)]
    ref2 = run_sync(model_dir, prompts, sps2)
    got, eng = run_overlap(model_dir, prompts, sps2)
    assert got == ref2
    assert got[0][-1] == stop_tok


def test_overlap_prefix_cache_consistent(model_dir):
    prompts = [list(range(1, 25))]
    sps = [SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)]
    got1, eng = run_overlap(model_dir, prompts, sps)
    # same engine, same prompt again: prefix cache must not be poisoned
    seqs = [Sequence(99, prompts[0],
                     SamplingParams(temperature=0.0, max_tokens=5,
                                    ignore_eos=True), eos_token_id=0)]
    eng.add_requests(seqs)
    eng.run_until_done()
    assert seqs[0].output_token_ids == got1[0]
