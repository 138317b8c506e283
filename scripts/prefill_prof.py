"""Focused prefill attention run for rocprofv3 PMC (Qwen32B shapes)."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from gllm_amd import ops

B, Hq, Hkv, D, ps, S = 8, 40, 8, 128, 16, 1024
torch.manual_seed(0)
n_pages = B * (S // ps) + 1
kc = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16, device="cuda")
vc = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16, device="cuda")
bt = torch.arange(1, n_pages, dtype=torch.int32, device="cuda").reshape(B, S // ps)
T = B * S
q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
seq_lens = torch.full((B,), S, dtype=torch.int32, device="cuda")
qsl = torch.arange(0, T + 1, S, dtype=torch.int32, device="cuda")
for _ in range(10):
    ops.paged_attention(q, kc, vc, bt, seq_lens, qsl, D ** -0.5,
                        max_query_len=S)
torch.cuda.synchronize()
