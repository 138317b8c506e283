"""Tiny MLA decode run for rocprofv3 (keep the trace small)."""

import math
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from gllm_amd import ops


def main():
    torch.manual_seed(0)
    B, ps, H, DK, DV = 64, 16, 128, 576, 512
    ctx = [1024] * B
    max_pages = 1024 // ps
    total = B * max_pages + 1
    kc = torch.randn(total, ps, 1, DK, dtype=torch.bfloat16, device="cuda")
    vc = kc[..., :DV]
    bt = torch.zeros(B, max_pages, dtype=torch.int32, device="cuda")
    nxt = 1
    for b in range(B):
        bt[b] = torch.arange(nxt, nxt + max_pages)
        nxt += max_pages
    q = torch.randn(B, H, DK, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl_l = list(range(B + 1))
    qsl = torch.tensor(qsl_l, dtype=torch.int32, device="cuda")
    scale = 1.0 / math.sqrt(DK)
    for _ in range(10):
        ops.mla_paged_attention(q, kc, vc, bt, seq_lens, qsl, scale,
                                seq_lens_cpu=ctx, query_start_loc_cpu=qsl_l)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
