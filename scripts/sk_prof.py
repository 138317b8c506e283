"""Run the skinny GEMM in a loop for rocprofv3 PMC collection.

    rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
        SQ_BUSY_CYCLES GRBM_GUI_ACTIVE -d gpurun_out/skpmc -- \
        python scripts/sk_prof.py down
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
from gllm_amd import ops  # noqa: E402

SHAPES = {
    "down": (64, 5120, 27648),
    "qkv": (64, 7168, 5120),
    "gate_up": (64, 55296, 5120),
    "down256": (256, 5120, 27648),
}


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "down"
    M, N, K = SHAPES[name]
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    for _ in range(3):
        ops.skinny_gemm(x, w)
    torch.cuda.synchronize()
    import time
    t0 = time.time()
    iters = int(os.environ.get("SK_ITERS", "30"))
    for _ in range(iters):
        ops.skinny_gemm(x, w)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    print(f"{name}: {dt*1e6:.1f} us, {2.0*N*K/dt/1e12:.2f} TB/s W-stream")


if __name__ == "__main__":
    main()
