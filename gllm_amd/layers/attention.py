"""Paged attention layer (reference: layers/attention.py FlashAttention).

One module instance per decoder layer; forward stores new K/V into the
paged cache (HIP scatter kernel) then runs the varlen paged-attention
kernel family (prefill MFMA kernel / split-KV decode kernel, dispatched
in ops.paged_attention).
"""


import torch
import torch.nn as nn

from gllm_amd import ops
from gllm_amd.runtime.forward_context import ForwardContext


class Attention(nn.Module):
    def __init__(self, layer_idx: int, num_heads: int, num_kv_heads: int,
                 head_dim: int, scale: float, sliding_window: int = 0):
        super().__init__()
        self.layer_idx = layer_idx       # LOCAL layer index on this PP stage
        self.num_heads = num_heads       # per TP rank
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        self.scale = scale
        self.sliding_window = sliding_window

    def forward(self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                fctx: ForwardContext) -> torch.Tensor:
        T = q.shape[0]
        q = q.unflatten(-1, (self.num_heads, self.head_dim))
        k = k.unflatten(-1, (self.num_kv_heads, self.head_dim))
        v = v.unflatten(-1, (self.num_kv_heads, self.head_dim))
        if fctx.is_profile_run:
            # Peak-memory profile pass runs before the KV cache exists
            # (reference layers/attention.py:112-116): skip attention math,
            # return a same-shape tensor.
            return q.reshape(T, -1)
        k_cache = fctx.k_caches[self.layer_idx]
        v_cache = fctx.v_caches[self.layer_idx]
        ops.reshape_and_cache(k, v, k_cache, v_cache, fctx.slot_mapping)
        out = ops.paged_attention(
            q, k_cache, v_cache, fctx.block_table, fctx.seq_lens,
            fctx.query_start_loc, self.scale,
            max_query_len=fctx.max_query_len,
            sliding_window=self.sliding_window)
        return out.reshape(T, -1)
