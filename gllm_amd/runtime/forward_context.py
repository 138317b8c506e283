"""Per-forward batch metadata handed through the model to every layer.

Equivalent in role to the reference's InputData (gllm/input_data.py) GPU
tensors, but passed explicitly down ``model.forward`` instead of being
read from a global.
"""

import dataclasses
from typing import List, Optional

import torch


@dataclasses.dataclass
class ForwardContext:
    # --- ragged batch geometry ---
    num_tokens: int
    positions: torch.Tensor          # [T] int64
    slot_mapping: torch.Tensor       # [T] int64 (flat KV slot per new token)
    block_table: torch.Tensor        # [B, max_pages] int32
    seq_lens: torch.Tensor           # [B] int32 (total context incl. chunk)
    query_start_loc: torch.Tensor    # [B+1] int32
    max_query_len: int               # 1 => pure decode
    max_seq_len: int
    # --- KV cache (this stage's layers) ---
    k_caches: List[torch.Tensor]     # per local layer [pages, page, Hkv, D]
    v_caches: List[torch.Tensor]
    # --- sampling metadata (set on the last PP stage) ---
    # rows of the hidden states whose logits we need: query_start_loc[1:]-1
    logits_indices: Optional[torch.Tensor] = None
    # profile/warmup run: attention may be skipped (no KV yet)
    is_profile_run: bool = False
    # overlap mode: token ids contain negative ring placeholders
    has_placeholders: bool = False
    # hybrid GDN models: per-seq SSM state slots (core/ssm.py)
    ssm_pool: Optional[object] = None
    ssm_slots: Optional[List[int]] = None
    ssm_has_init: Optional[List[bool]] = None
    # graph-replay path: persistent device slot buffer (all rows are
    # initialized decode steps; padding rows point at the dummy slot)
    ssm_slots_dev: Optional[object] = None
    # multimodal: rows of this batch whose embeddings come from the
    # vision tower (replaced after embed_tokens on the first stage)
    mm_rows: Optional[torch.Tensor] = None     # [N] long
    mm_embeds: Optional[torch.Tensor] = None   # [N, hidden]
    # Qwen3-VL deepstack: [N, D*hidden] multiscale features added at the
    # image rows after decoder layers 0..D-1
    mm_deepstack: Optional[torch.Tensor] = None
    # DSA (DeepSeek-V3.2): per-layer paged index-K caches
    # [pages, page_size, index_head_dim] (models/deepseek_v32.py)
    idx_caches: Optional[List[torch.Tensor]] = None
    # host copies of the ragged geometry (filled by the batch builder
    # from its numpy staging for free) so per-seq torch paths — DSA
    # selection, GDN chunking, prompt logprobs — iterate without a
    # device sync per int() read
    seq_lens_cpu: Optional[List[int]] = None
    query_start_loc_cpu: Optional[List[int]] = None

    @property
    def is_pure_decode(self) -> bool:
        return self.max_query_len == 1

    def host_qsl(self) -> List[int]:
        if self.query_start_loc_cpu is None:
            self.query_start_loc_cpu = self.query_start_loc.tolist()
        return self.query_start_loc_cpu

    def host_seq_lens(self) -> List[int]:
        if self.seq_lens_cpu is None:
            self.seq_lens_cpu = self.seq_lens.tolist()
        return self.seq_lens_cpu
