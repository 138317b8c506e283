"""SSM working-state pool for hybrid linear-attention models.

Parity: reference SSMSegment working pool (memory_manager.py:87-256) —
per-layer conv_state + recurrent-state tensors addressed by a per-seq
slot. Snapshot pools (prefix-cache state restore) are a round-2 item;
round 1 disables prefix caching for hybrid models instead.
"""

import dataclasses
from typing import List

import torch

from gllm_amd.sequence import Sequence
from gllm_amd.utils.id_allocator import IDAllocator


@dataclasses.dataclass
class SSMSpec:
    num_ssm_layers: int
    conv_dim: int        # channels of the causal conv (per TP rank)
    conv_kernel: int
    num_v_heads: int     # per TP rank
    head_k_dim: int
    head_v_dim: int


class SSMPool:
    def __init__(self, spec: SSMSpec, num_slots: int, device: str,
                 dtype=torch.float32):
        self.spec = spec
        self.device = device
        L = spec.num_ssm_layers
        self.conv_state = torch.zeros(
            L, num_slots, spec.conv_dim, spec.conv_kernel - 1,
            dtype=dtype, device=device)
        self.ssm_state = torch.zeros(
            L, num_slots, spec.num_v_heads, spec.head_v_dim,
            spec.head_k_dim, dtype=torch.float32, device=device)
        self.alloc = IDAllocator(num_slots)

    def ensure(self, seq: Sequence) -> int:
        if seq.ssm_slot >= 0:
            return seq.ssm_slot
        slot = self.alloc.allocate()
        seq.ssm_slot = slot
        return slot

    def free(self, seq: Sequence) -> None:
        if seq.ssm_slot >= 0:
            self.alloc.free(seq.ssm_slot)
            seq.ssm_slot = -1
