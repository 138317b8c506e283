"""Multi-process PP engine test on CPU (gloo, world_size 2).

Verifies the replicated-scheduler PP design end to end: 2 pipeline
stages produce exactly the same greedy tokens as the single-process
engine on the same dummy-weight model.
"""

import json
import multiprocessing as mp
import os

import pytest
import torch

TINY = {
    "architectures": ["Qwen2ForCausalLM"],
    "model_type": "qwen2",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 4,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 0,
}

PROMPTS = [list(range(1, 30)), list(range(40, 40 + 17)), [5, 6, 7]]
MAX_TOKENS = 6


def _mk_model_dir(tmp):
    d = os.path.join(tmp, "tiny4")
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(TINY, f)
    return d


def _mk_config(model_dir, pp_size, port):
    from gllm_amd.config import EngineConfig
    return EngineConfig(model=model_dir, load_format="dummy", device="cpu",
                        dtype="float32", page_size=4, maxp=16, maxd=32,
                        pp_size=pp_size, master_port=port,
                        schedule_method="token_throttling",
                        enable_prefix_caching=False)


def _run_rank(rank, model_dir, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.set_num_threads(1)
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence
    cfg = _mk_config(model_dir, 2, port)
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
def test_pp2_matches_single_process(tmp_path):
    model_dir = _mk_model_dir(str(tmp_path))

    # single-process reference
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams
    cfg1 = _mk_config(model_dir, 1, 0)
    llm = LLM(config=cfg1, num_pages_override=128)
    ref = [o.token_ids for o in llm.generate(
        PROMPTS, [SamplingParams(temperature=0.0, max_tokens=MAX_TOKENS,
                                 ignore_eos=True)] * len(PROMPTS))]

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29611
    procs = [ctx.Process(target=_run_rank,
                         args=(r, model_dir, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got == ref
