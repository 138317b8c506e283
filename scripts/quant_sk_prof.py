"""Quantized skinny GEMM microbench (fp8 w8a8 / int4 w4a16).

Times each Qwen2.5-32B decode shape through the public ops wrappers
(quant + GEMM + reduce end to end) and prints actual-weight-byte TB/s.

    python scripts/quant_sk_prof.py fp8|int4 [shape] [iters]

Under rocprofv3 PMC (single shape keeps the counter file small):
    rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
        SQ_ACTIVE_INST_ANY SQ_ACTIVE_INST_VALU SQ_INSTS_MFMA \
        --output-format csv -d <dir> -- python scripts/quant_sk_prof.py int4 down 10
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
from gllm_amd import ops  # noqa: E402

SHAPES = {
    "qkv": (64, 7168, 5120),
    "o": (64, 5120, 5120),
    "gate_up": (64, 55296, 5120),
    "down": (64, 5120, 27648),
}


def run(mode, name, iters):
    M, N, K = SHAPES[name]
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    if mode == "fp8":
        from gllm_amd.layers.quantization import fp8 as qfp8
        w = torch.randn(N, K, dtype=torch.bfloat16) / 32
        wq, ws = qfp8.block_quant_fp8(w)
        wq, ws = wq.cuda(), ws.cuda()
        wbytes = N * K
        fn = lambda: ops.fp8_linear(x, wq, ws)  # noqa: E731
    else:
        from types import SimpleNamespace
        from gllm_amd.layers.quantization import int4 as qi4
        w = torch.randn(N, K) / 32
        qweight, qzeros, scales = qi4.pack_gptq(w, group_size=128)
        layer = SimpleNamespace(int4_cfg=("gptq", 128), qweight=qweight,
                                qzeros=qzeros, scales=scales)
        wq4, sb, grp = qi4.repack_canonical(layer)
        sbt = sb.permute(1, 2, 0).contiguous().cuda()
        wq4 = wq4.cuda()
        wbytes = N * K // 2
        fn = lambda: ops.int4_linear(x, wq4, sbt, grp)  # noqa: E731
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    print(f"{mode} {name:8s} [{N:6d}x{K:5d}] {dt*1e6:8.1f} us  "
          f"W-actual {wbytes/dt/1e12:5.2f} TB/s  "
          f"bf16-equiv {2*N*K/dt/1e12:5.2f} TB/s")


def main():
    mode = sys.argv[1] if len(sys.argv) > 1 else "fp8"
    shape = sys.argv[2] if len(sys.argv) > 2 else None
    iters = int(sys.argv[3]) if len(sys.argv) > 3 else 30
    for name in ([shape] if shape else SHAPES):
        run(mode, name, iters)


def splitk_sweep():
    """Direct fp8_skinny_gemm calls with explicit splitk values."""
    from gllm_amd.layers.quantization import fp8 as qfp8
    from gllm_amd.ops import _gpu_kernels, per_token_group_quant_fp8
    k = _gpu_kernels()
    for name in SHAPES:
        M, N, K = SHAPES[name]
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16) / 32
        wq, ws = qfp8.block_quant_fp8(w)
        wq, ws = wq.cuda().view(torch.uint8), ws.cuda().contiguous()
        aq, _, ast = per_token_group_quant_fp8(x, transposed=True)
        aq = aq.view(torch.uint8)
        out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
        for sk in (1, 2, 4, 8, 16):
            if (K // sk) % 128:
                continue
            wsb = torch.empty(sk * M * N, dtype=torch.float32,
                              device="cuda")
            try:
                k.fp8_skinny_gemm(out, aq, ast, wq, ws, None, wsb, sk)
            except RuntimeError as e:
                print(f"{name} sk={sk}: {e}")
                continue
            torch.cuda.synchronize()
            import time
            t0 = time.time()
            for _ in range(30):
                k.fp8_skinny_gemm(out, aq, ast, wq, ws, None, wsb, sk)
            torch.cuda.synchronize()
            dt = (time.time() - t0) / 30
            print(f"{name:8s} splitk={sk:2d} {dt*1e6:7.1f} us "
                  f"W-actual {N*K/dt/1e12:5.2f} TB/s")


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "sweep":
        splitk_sweep()
    else:
        main()
