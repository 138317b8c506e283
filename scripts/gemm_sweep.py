"""Decode/prefill GEMM shape sweep: hipBLASLt default vs TunableOp.

Shapes are Qwen2.5-32B projections. Run on an MI355X:
    python scripts/gemm_sweep.py             # default heuristics
    PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
        python scripts/gemm_sweep.py         # online-tuned
Prints effective TB/s of weight streaming per shape (the decode-regime
bound) and TF/s.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

SHAPES = [
    # (name, N out, K in)
    ("qkv", 7168, 5120),
    ("o", 5120, 5120),
    ("gate_up", 55296, 5120),
    ("down", 5120, 27648),
    ("lm_head", 152064, 5120),
]
MS = [32, 64, 128, 256, 512, 8192]


def bench_shape(M, N, K, iters=20):
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    for _ in range(3):
        F.linear(x, w)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        F.linear(x, w)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    bytes_w = 2.0 * N * K
    tbps = bytes_w / dt / 1e12
    tf = 2.0 * M * N * K / dt / 1e12
    return dt * 1e6, tbps, tf


def main():
    print(f"TunableOp={os.environ.get('PYTORCH_TUNABLEOP_ENABLED', '0')}")
    for M in MS:
        for name, N, K in SHAPES:
            if M == 8192 and name == "lm_head":
                continue
            us, tbps, tf = bench_shape(M, N, K)
            print(f"M={M:5d} {name:8s} [{N:6d}x{K:5d}] {us:9.1f} us  "
                  f"W-stream {tbps:5.2f} TB/s  {tf:7.1f} TF/s")


if __name__ == "__main__":
    main()


def bench_shape_nn(M, N, K, iters=20):
    """x [M,K] @ W [K,N] (pre-transposed weight, NN layout)."""
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(K, N, dtype=torch.bfloat16, device="cuda")
    for _ in range(3):
        torch.matmul(x, w)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        torch.matmul(x, w)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    return dt * 1e6, 2.0 * N * K / dt / 1e12, 2.0 * M * N * K / dt / 1e12


if os.environ.get("SWEEP_NN"):
    print("=== NN layout (x @ W[K,N]) ===")
    for M in [32, 64, 128, 256]:
        for name, N, K in SHAPES:
            us, tbps, tf = bench_shape_nn(M, N, K)
            print(f"NN M={M:5d} {name:8s} [{K:5d}x{N:6d}] {us:9.1f} us  "
                  f"W-stream {tbps:5.2f} TB/s  {tf:7.1f} TF/s")


if os.environ.get("SWEEP_SKINNY"):
    from gllm_amd import ops
    print("=== skinny_gemm ===")
    for M in [32, 64, 128, 256]:
        for name, N, K in SHAPES:
            x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
            w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
            for _ in range(3):
                ops.skinny_gemm(x, w)
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(20):
                ops.skinny_gemm(x, w)
            torch.cuda.synchronize()
            dt = (time.time() - t0) / 20
            print(f"SK M={M:5d} {name:8s} {dt*1e6:9.1f} us  "
                  f"W-stream {2.0*N*K/dt/1e12:5.2f} TB/s")

if os.environ.get("SWEEP_FP8"):
    from gllm_amd import ops
    from gllm_amd.layers.quantization import fp8 as qfp8
    print("=== fp8_linear (quant+gemm end to end) ===")
    for M in [32, 64, 128, 256]:
        for name, N, K in SHAPES:
            if K % 128 or N > 60000:
                continue
            x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
            wq = torch.randint(0, 255, (N, K), dtype=torch.uint8,
                               device="cuda").view(torch.float8_e4m3fn)
            ws = torch.rand(-(-N // 128), K // 128, device="cuda") * 0.01
            for _ in range(3):
                ops.fp8_linear(x, wq, ws)
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(20):
                ops.fp8_linear(x, wq, ws)
            torch.cuda.synchronize()
            dt = (time.time() - t0) / 20
            print(f"FP8 M={M:5d} {name:8s} {dt*1e6:9.1f} us  "
                  f"W-stream {1.0*N*K/dt/1e12:5.2f} TB/s "
                  f"(bf16-equiv {2.0*N*K/dt/1e12:5.2f})")
