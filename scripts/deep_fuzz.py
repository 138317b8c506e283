"""Deep offline fuzz: repeated-prompt consistency under prefix caching,
hybrid SSM snapshot restore, chunk-budget variation and sampling-param
mixes. One-off QA driver (heavier than the suite's test_fuzz_cpu):

    python scripts/deep_fuzz.py --rounds 30 --seed 0
"""
import argparse
import json
import os
import random
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

QWEN_TINY = {
    "architectures": ["Qwen2ForCausalLM"], "model_type": "qwen2",
    "hidden_size": 64, "intermediate_size": 128, "num_hidden_layers": 2,
    "num_attention_heads": 4, "num_key_value_heads": 2, "vocab_size": 128,
    "max_position_embeddings": 2048, "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0, "eos_token_id": 0,
}
HYBRID_TINY = {
    **QWEN_TINY, "architectures": ["Qwen3_5ForCausalLM"],
    "model_type": "qwen3_5", "full_attention_interval": 2,
    "num_hidden_layers": 4, "head_dim": 16, "attn_output_gate": True,
    "partial_rotary_factor": 0.5, "linear_num_value_heads": 4,
    "linear_num_key_heads": 2, "linear_key_head_dim": 8,
    "linear_value_head_dim": 8, "linear_conv_kernel_dim": 4,
}


def mk_llm(cfg_json, maxp, page):
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    d = tempfile.mkdtemp(prefix="fuzz_")
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg_json, f)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=page, maxp=maxp,
                       enable_prefix_caching=True)
    return LLM(config=cfg, num_pages_override=192)


def run(rounds, seed):
    from gllm_amd.sequence import SamplingParams
    rng = random.Random(seed)
    failures = 0
    for cfg_json, name in ((QWEN_TINY, "dense"), (HYBRID_TINY, "hybrid")):
        for r in range(rounds):
            maxp = rng.choice([4, 8, 16, 64])
            page = rng.choice([4, 8])
            llm = mk_llm(cfg_json, maxp, page)
            # a pool of prompts, some shared prefixes
            base = [rng.randint(1, 120) for _ in range(rng.randint(8, 40))]
            prompts = []
            for _ in range(6):
                if rng.random() < 0.5:
                    p = base[:rng.randint(4, len(base))] + \
                        [rng.randint(1, 120)
                         for _ in range(rng.randint(0, 10))]
                else:
                    p = [rng.randint(1, 120)
                         for _ in range(rng.randint(3, 40))]
                prompts.append(p)
            sps = [SamplingParams(
                temperature=0.0,
                max_tokens=rng.randint(1, 8),
                repetition_penalty=rng.choice([1.0, 1.0, 1.3]),
                frequency_penalty=rng.choice([0.0, 0.0, 0.5]),
                ignore_eos=True) for _ in prompts]
            first = [o.token_ids for o in llm.generate(prompts, sps)]
            # repeats (prefix hits + snapshot restore) must reproduce
            again = [o.token_ids for o in llm.generate(prompts, sps)]
            if first != again:
                failures += 1
                print(f"FAIL {name} round {r}: maxp={maxp} page={page}")
                for i, (a, b) in enumerate(zip(first, again)):
                    if a != b:
                        print("  prompt", prompts[i], "->", a, "vs", b)
            # shuffled subset as a third pass
            idx = list(range(len(prompts)))
            rng.shuffle(idx)
            sub = idx[:4]
            third = [o.token_ids for o in llm.generate(
                [prompts[i] for i in sub], [sps[i] for i in sub])]
            if third != [first[i] for i in sub]:
                failures += 1
                print(f"FAIL {name} round {r} (shuffled subset)")
            # page-leak invariant
            mm = llm.runner.memory_manager
            in_use = sum(1 for rc in getattr(mm, "page_ref", [])
                         if rc > 0)
            if not hasattr(mm, "page_ref"):
                in_use = 0
            if in_use != 0 and mm.get_num_free_pages() + in_use \
                    < mm.num_pages // 2:
                print(f"WARN {name} round {r}: pages in_use={in_use}")
    print("deep fuzz done:", "OK" if failures == 0 else
          f"{failures} FAILURES")
    return failures


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=20)
    ap.add_argument("--seed", type=int, default=0)
    a = ap.parse_args()
    sys.exit(1 if run(a.rounds, a.seed) else 0)
