"""Flagship serving benchmark (driver contract).

Measures the BASELINE.json metric: output tokens/sec (whole node) for
Qwen2.5-32B at PP = N on MI355X, synthetic data, random-init (dummy)
weights, bf16. A "step" is one pipeline tick of the serving engine
(schedule -> forward -> sample -> commit) with a saturated continuous
decode batch; `value` is sampled output tokens/sec for the WHOLE model
(one model spanning all N GPUs via PP), so scaling is "strong".

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
N>1 is launched by the driver via torch.distributed.run (one rank per
GPU over RCCL); ranks read RANK/WORLD_SIZE from the env.
"""

import argparse
import json
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

QWEN25_32B = {
    "architectures": ["Qwen2ForCausalLM"],
    "model_type": "qwen2",
    "hidden_size": 5120,
    "intermediate_size": 27648,
    "num_hidden_layers": 64,
    "num_attention_heads": 40,
    "num_key_value_heads": 8,
    "vocab_size": 152064,
    "max_position_embeddings": 32768,
    "rms_norm_eps": 1e-6,
    "rope_theta": 1000000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 151643,
}

SMALL_DEBUG = {  # --model debug: quick bring-up config
    **QWEN25_32B,
    "hidden_size": 1024, "intermediate_size": 2816,
    "num_hidden_layers": 8, "num_attention_heads": 8,
    "num_key_value_heads": 2, "vocab_size": 32000,
}

LLAMA3_8B = {  # BASELINE config: Llama-3 8B PP=1 bf16
    "architectures": ["LlamaForCausalLM"],
    "model_type": "llama",
    "hidden_size": 4096,
    "intermediate_size": 14336,
    "num_hidden_layers": 32,
    "num_attention_heads": 32,
    "num_key_value_heads": 8,
    "vocab_size": 128256,
    "max_position_embeddings": 8192,
    "rms_norm_eps": 1e-5,
    "rope_theta": 500000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 128001,
}

MIXTRAL_8X7B = {  # BASELINE config: Mixtral 8x7B PP + EP over xGMI
    "architectures": ["MixtralForCausalLM"],
    "model_type": "mixtral",
    "hidden_size": 4096,
    "intermediate_size": 14336,
    "num_hidden_layers": 32,
    "num_attention_heads": 32,
    "num_key_value_heads": 8,
    "num_local_experts": 8,
    "num_experts_per_tok": 2,
    "vocab_size": 32000,
    "max_position_embeddings": 32768,
    "rms_norm_eps": 1e-5,
    "rope_theta": 1000000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 2,
}

MIXTRAL_DEBUG = {  # CPU-testable MoE path
    **MIXTRAL_8X7B,
    "hidden_size": 512, "intermediate_size": 1024,
    "num_hidden_layers": 4, "num_attention_heads": 8,
    "num_key_value_heads": 2, "vocab_size": 32000,
}

DEEPSEEK_V3 = {  # BASELINE config: DeepSeek-V3 PP=8 + EP, absorbed MLA
    "architectures": ["DeepseekV3ForCausalLM"],
    "model_type": "deepseek_v3",
    "hidden_size": 7168,
    "intermediate_size": 18432,
    "moe_intermediate_size": 2048,
    "num_hidden_layers": 61,
    "first_k_dense_replace": 3,
    "num_attention_heads": 128,
    "num_key_value_heads": 128,
    "n_routed_experts": 256,
    "n_shared_experts": 1,
    "num_experts_per_tok": 8,
    "n_group": 8,
    "topk_group": 4,
    "routed_scaling_factor": 2.5,
    "scoring_func": "sigmoid",
    "norm_topk_prob": True,
    "q_lora_rank": 1536,
    "kv_lora_rank": 512,
    "qk_nope_head_dim": 128,
    "qk_rope_head_dim": 64,
    "v_head_dim": 128,
    "vocab_size": 129280,
    "max_position_embeddings": 163840,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 1,
}

DEEPSEEK_V2_LITE = {  # real DeepSeek-V2-Lite dims: 1-GPU MLA+MoE bench
    "architectures": ["DeepseekV2ForCausalLM"],
    "model_type": "deepseek_v2",
    "hidden_size": 2048,
    "intermediate_size": 10944,
    "moe_intermediate_size": 1408,
    "num_hidden_layers": 27,
    "first_k_dense_replace": 1,
    "num_attention_heads": 16,
    "num_key_value_heads": 16,
    "n_routed_experts": 64,
    "n_shared_experts": 2,
    "num_experts_per_tok": 6,
    "n_group": 1,
    "topk_group": 1,
    "routed_scaling_factor": 1.0,
    "scoring_func": "softmax",
    "norm_topk_prob": False,
    "q_lora_rank": None,
    "kv_lora_rank": 512,
    "qk_nope_head_dim": 128,
    "qk_rope_head_dim": 64,
    "v_head_dim": 128,
    "vocab_size": 102400,
    "max_position_embeddings": 163840,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 1,
}

QWEN3_NEXT_9B = {  # hybrid GDN + full attention, 1-GPU scale
    "architectures": ["Qwen3NextForCausalLM"],
    "model_type": "qwen3_next",
    "hidden_size": 2048,
    "intermediate_size": 5504,
    "num_hidden_layers": 24,
    "num_attention_heads": 16,
    "num_key_value_heads": 2,
    "head_dim": 128,
    "linear_num_value_heads": 16,
    "linear_num_key_heads": 8,
    "linear_key_head_dim": 128,
    "linear_value_head_dim": 128,
    "linear_conv_kernel_dim": 4,
    "full_attention_interval": 4,
    "vocab_size": 151936,
    "max_position_embeddings": 32768,
    "rms_norm_eps": 1e-6,
    "rope_theta": 1000000.0,
    "partial_rotary_factor": 0.25,
    "tie_word_embeddings": False,
    "eos_token_id": 2,
}

MODELS = {
    "qwen2.5-32b": ("Qwen2.5-32B", QWEN25_32B),
    "deepseek-v2-lite": ("DeepSeek-V2-Lite", DEEPSEEK_V2_LITE),
    "qwen3-next-9b": ("Qwen3-Next-9B-hybrid", QWEN3_NEXT_9B),
    "debug": ("debug-0.2B", SMALL_DEBUG),
    "llama3-8b": ("Llama-3-8B", LLAMA3_8B),
    "mixtral-8x7b": ("Mixtral-8x7B", MIXTRAL_8X7B),
    "mixtral-debug": ("mixtral-debug", MIXTRAL_DEBUG),
    "deepseek-v3": ("DeepSeek-V3", DEEPSEEK_V3),
}


def write_model_dir(cfg_json):
    d = tempfile.mkdtemp(prefix="bench_model_")
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg_json, f)
    return d


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--prompt-len", type=int, default=1024)
    ap.add_argument("--model", type=str, default="qwen2.5-32b",
                    choices=sorted(MODELS))
    ap.add_argument("--tp", type=int, default=1,
                    help="TP degree; PP = gpus // tp (BASELINE's MoE "
                         "configs pair PP with EP over the TP ranks)")
    ap.add_argument("--use-ep", action="store_true",
                    help="expert parallelism over the dp*tp ranks of "
                         "each stage")
    ap.add_argument("--page-size", type=int, default=16)
    ap.add_argument("--quant", type=str, default=None,
                    choices=["fp8", "int4"],
                    help="block-quantize the dummy weights (fp8 e4m3 "
                         "128x128 blocks) and run the native fp8 path")
    ap.add_argument("--schedule", type=str, default="token_throttling")
    ap.add_argument("--qps", type=float, default=32.0,
                    help=">0: pace request arrivals at this rate for the "
                         "TTFT measurement instead of a burst")
    ap.add_argument("--no-graph", action="store_true")
    ap.add_argument("--no-overlap", action="store_true")
    ap.add_argument("--max-graph-bs", type=int, default=512)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    assert world == args.gpus or world == 1, \
        f"WORLD_SIZE {world} != --gpus {args.gpus}"
    n = max(world, 1)

    use_gpu = torch.cuda.is_available()
    device = f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}" if use_gpu \
        else "cpu"
    model_name, model_json = MODELS[args.model]
    if args.quant == "fp8":
        model_json = dict(model_json)
        model_json["quantization_config"] = {
            "quant_method": "fp8", "weight_block_size": [128, 128]}
        model_name += "-fp8"
    elif args.quant == "int4":
        model_json = dict(model_json)
        model_json["quantization_config"] = {
            "quant_method": "gptq", "bits": 4, "group_size": 128}
        model_name += "-int4"
    model_dir = write_model_dir(model_json)
    assert n % args.tp == 0, (n, args.tp)
    pp = n // args.tp

    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.overlap_engine import OverlapEngine
    from gllm_amd.engine.pp_engine import PPEngine
    from gllm_amd.sequence import SamplingParams, Sequence

    cfg = EngineConfig(
        model=model_dir, load_format="dummy",
        dtype="bfloat16" if use_gpu else "float32",
        device=device, pp_size=pp, tp_size=args.tp, use_ep=args.use_ep,
        page_size=args.page_size,
        schedule_method=args.schedule, maxp=8192, maxd=1024,
        use_graph=not args.no_graph, max_graph_bs=args.max_graph_bs,
        enable_prefix_caching=False,
        master_addr=os.environ.get("MASTER_ADDR", "127.0.0.1"),
        master_port=int(os.environ.get("MASTER_PORT", "29500")),
    )
    if n == 1 and not args.no_overlap:
        eng = OverlapEngine(cfg, num_pages_override=None if use_gpu else 512)
    else:
        eng = PPEngine(cfg, num_pages_override=None if use_gpu else 512)

    # ---- synthetic load: batch of fixed prompts, unbounded decode ----
    g = torch.Generator().manual_seed(1234)
    vocab = model_json["vocab_size"]
    seqs = []
    t_submit = time.time()
    for i in range(args.batch):
        ids = torch.randint(1, vocab - 1, (args.prompt_len,),
                            generator=g).tolist()
        sp = SamplingParams(temperature=0.0, ignore_eos=True,
                            max_tokens=4096)
        s = Sequence(i, ids, sp, eos_token_id=None, arrival_time=t_submit)
        seqs.append(s)
    # ---- ramp: arrivals (burst or paced at --qps) until every seq has
    # produced its first token; TTFT measured from each seq's arrival ----
    # Paced arrivals under multi-GPU: rank 0's wall clock decides how
    # many requests release each tick and BROADCASTS the count, so every
    # replicated scheduler applies the identical release schedule
    # (deterministic-by-construction; wall time never drives a
    # non-rank-0 decision). BASELINE.json's "p50 TTFT at fixed QPS"
    # comes from this paced ramp.
    ttfts = {}
    arrivals = {}
    if args.qps > 0:
        import torch.distributed as dist
        to_release = list(seqs)
        next_t = time.time()
        while True:
            k = 0
            if rank == 0:
                now = time.time()
                while k < len(to_release) and now >= next_t:
                    k += 1
                    next_t += 1.0 / args.qps
            if world > 1:
                obj = [k]
                dist.broadcast_object_list(obj, src=0)
                k = int(obj[0])
            now = time.time()
            for _ in range(k):
                s = to_release.pop(0)
                arrivals[s.seq_id] = now
                eng.add_requests([s])
            eng.step_tick()
            now = time.time()
            for s in seqs:
                if s.seq_id in arrivals and s.seq_id not in ttfts \
                        and s.num_output_tokens > 0:
                    ttfts[s.seq_id] = (now - arrivals[s.seq_id]) * 1000.0
            # deterministic exit: engine state is identical on every rank
            if not to_release and \
                    all(s.num_output_tokens > 0 for s in seqs):
                break
    else:
        eng.add_requests(seqs)
        while len(ttfts) < len(seqs):
            eng.step_tick()
            now = time.time()
            for s in seqs:
                if s.seq_id not in ttfts and s.num_output_tokens > 0:
                    ttfts[s.seq_id] = (now - t_submit) * 1000.0
            if not eng.scheduler.has_work():
                break
    ttft_sorted = sorted(ttfts.values())
    ttft_p50 = ttft_sorted[len(ttft_sorted) // 2] if ttft_sorted else None

    # ---- warmup ----
    for _ in range(args.warmup):
        eng.step_tick()

    # ---- timed region ----
    eng.barrier_sync()
    t0 = time.time()
    sampled = 0
    for _ in range(args.steps):
        sampled += eng.step_tick()
    eng.barrier_sync()
    elapsed = time.time() - t0

    # max elapsed over ranks
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    value = sampled / elapsed
    if rank == 0:
        out = {
            "metric": "output tokens/sec (whole node)",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": ({"fp8": "fp8-w8a8", "int4": "int4-w4a16"}.get(
                args.quant) or "bf16") if use_gpu else "fp32",
            "data": "synthetic",
            "ttft_p50_ms": round(ttft_p50, 1) if ttft_p50 else None,
            "ttft_qps": args.qps if args.qps > 0 else "burst",
            "config": {
                "model": model_name,
                "global_batch": args.batch,
                "seq_len": args.prompt_len,
                "parallelism": f"pp{pp}" +
                (f"tp{args.tp}" if args.tp > 1 else "") +
                ("ep" if args.use_ep else ""),
                "schedule": args.schedule,
            },
        }
        print(json.dumps(out))
    if world > 1:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
