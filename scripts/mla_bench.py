"""MLA kernel microbench on MI355X: decode + prefill regimes.

Run (GPU box):  python scripts/mla_bench.py
Reports ms/call, effective KV bandwidth and TF/s for the absorbed-MLA
attention kernel at DeepSeek-V3 shapes (H=128, latent 576/512).
"""

import math
import time

import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from gllm_amd import ops


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def mk(B, ps, ctx, H, DK=576):
    max_pages = max(-(-c // ps) for c in ctx)
    total = sum(-(-c // ps) for c in ctx) + 1
    kc = torch.randn(total, ps, 1, DK, dtype=torch.bfloat16, device="cuda")
    bt = torch.zeros(B, max_pages, dtype=torch.int32, device="cuda")
    nxt = 1
    for b, c in enumerate(ctx):
        n = -(-c // ps)
        bt[b, :n] = torch.arange(nxt, nxt + n)
        nxt += n
    return kc, bt


def run_case(name, q_lens, ctx, H):
    B, ps, DK, DV = len(q_lens), 16, 576, 512
    kc, bt = mk(B, ps, ctx, H)
    vc = kc[..., :DV]
    T = sum(q_lens)
    q = torch.randn(T, H, DK, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl_l = [0]
    for x in q_lens:
        qsl_l.append(qsl_l[-1] + x)
    qsl = torch.tensor(qsl_l, dtype=torch.int32, device="cuda")
    scale = 1.0 / math.sqrt(DK)
    ms = bench(lambda: ops.mla_paged_attention(
        q, kc, vc, bt, seq_lens, qsl, scale, seq_lens_cpu=ctx,
        query_start_loc_cpu=qsl_l))
    kv_bytes = sum(ctx) * DK * 2
    # causal flops: per seq sum over rows of visible kv
    flops = 0
    for ql, c in zip(q_lens, ctx):
        past = c - ql
        vis = sum(past + i + 1 for i in range(ql))
        flops += vis * H * (DK + DV) * 2
    print(f"{name:34s} {ms:8.3f} ms   {kv_bytes/ms/1e6:8.1f} GB/s KV"
          f"   {flops/ms/1e9:8.1f} TF/s")


if __name__ == "__main__":
    torch.manual_seed(0)
    run_case("decode B=64 ctx=1k H=128", [1] * 64, [1024] * 64, 128)
    run_case("decode B=64 ctx=4k H=128", [1] * 64, [4096] * 64, 128)
    run_case("decode B=256 ctx=1k H=128", [1] * 256, [1024] * 256, 128)
    run_case("decode B=8 ctx=16k H=128", [1] * 8, [16384] * 8, 128)
    run_case("prefill 4x1k H=128", [1024] * 4, [1024] * 4, 128)
    run_case("prefill 1x8k H=128", [8192], [8192], 128)
    run_case("chunk 1k past 15k H=128", [1024], [16384], 128)
