"""fp8 skinny at M in {96, 128, 192, 256} per ring (FP8_RING env)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from gllm_amd import ops  # noqa: E402
from gllm_amd.layers.quantization import fp8 as qfp8  # noqa: E402

for M in (96, 128, 192, 256):
    for name, N, K in (("gate_up", 55296, 5120), ("down", 5120, 27648)):
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16) / 32
        wq, ws = qfp8.block_quant_fp8(w)
        wq, ws = wq.cuda(), ws.cuda()
        for _ in range(3):
            ops.fp8_linear(x, wq, ws)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(30):
            ops.fp8_linear(x, wq, ws)
        torch.cuda.synchronize()
        dt = (time.time() - t0) / 30
        print(f"M={M:3d} {name:8s} {dt*1e6:7.1f} us "
              f"W-actual {N*K/dt/1e12:5.2f} TB/s")
