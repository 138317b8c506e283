"""SSM working-state + snapshot pools for hybrid linear-attention
models.

Parity: reference SSMSegment (memory_manager.py:87-256) — per-layer
conv_state + recurrent-state tensors addressed by a per-seq slot, plus
SNAPSHOT storage keyed by prefix-cache chain id: when a sequence's
computed length lands exactly on a page boundary at registration time,
its recurrent state is cloned under that boundary's chain id; a later
prefix hit restores it (the KV pages alone are not enough for hybrid
models — the linear-attention layers carry state). The prefix
manager's hit_filter trims hits to the deepest snapshotted boundary.
"""

import collections
import dataclasses

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
        # the LAST slot is the hipGraph dummy/scratch slot (padding rows
        # of a captured decode bucket write their garbage there); the
        # allocator never hands it out
        self.dummy_slot = num_slots - 1
        self.alloc = IDAllocator(num_slots - 1)

    def ensure(self, seq: Sequence) -> int:
        if seq.ssm_slot >= 0:
            return seq.ssm_slot
        slot = self.alloc.allocate()
        seq.ssm_slot = slot
        seq.ssm_state_ready = False
        return slot

    def free(self, seq: Sequence) -> None:
        if seq.ssm_slot >= 0:
            self.alloc.free(seq.ssm_slot)
            seq.ssm_slot = -1
            seq.ssm_state_ready = False

    # ---- snapshots (prefix-cache state restore) ----
    max_snapshots = 64

    @property
    def snapshots(self):
        if not hasattr(self, "_snapshots"):
            self._snapshots = collections.OrderedDict()
        return self._snapshots

    def has_snapshot(self, chain_id: int) -> bool:
        return chain_id in self.snapshots

    def snapshot(self, chain_id: int, slot: int) -> None:
        snaps = self.snapshots
        snaps[chain_id] = (self.conv_state[:, slot].clone(),
                           self.ssm_state[:, slot].clone())
        snaps.move_to_end(chain_id)
        while len(snaps) > self.max_snapshots:
            snaps.popitem(last=False)

    def restore(self, chain_id: int, slot: int) -> bool:
        snap = self.snapshots.get(chain_id)
        if snap is None:
            return False
        self.snapshots.move_to_end(chain_id)
        self.conv_state[:, slot].copy_(snap[0])
        self.ssm_state[:, slot].copy_(snap[1])
        return True
