"""Persistent repetition-penalty mask pool.

Parity: reference memory_manager.py:723-828 (per-seq token-presence
masks kept resident so the sampler applies penalties without gathering
host token histories). Device-resident uint8 [slots, vocab]; updated by
GPU scatters, so the overlap engine can penalize without waiting for
the sampled token to reach the host.
"""


import torch

from gllm_amd.sequence import Sequence
from gllm_amd.utils.id_allocator import IDAllocator


class PenaltyPool:
    def __init__(self, num_slots: int, vocab_size: int, device: str):
        self.vocab_size = vocab_size
        self.device = device
        self.mask = torch.zeros(num_slots, vocab_size, dtype=torch.uint8,
                                device=device)
        self.alloc = IDAllocator(num_slots)

    def ensure(self, seq: Sequence) -> int:
        """Slot for ``seq``, allocating + seeding with the prompt tokens
        on first use. Returns -1 when the pool is exhausted (caller falls
        back to the host-row path)."""
        if seq.penalty_slot >= 0:
            return seq.penalty_slot
        if self.alloc.num_free == 0:
            return -1
        slot = self.alloc.allocate()
        seq.penalty_slot = slot
        row = self.mask[slot]
        row.zero_()
        toks = torch.tensor(
            [t for t in seq.token_ids if 0 <= t < self.vocab_size],
            dtype=torch.long, device=self.device)
        if toks.numel():
            row[toks] = 1
        return slot

    def append(self, slots: torch.Tensor, tokens: torch.Tensor) -> None:
        """Mark sampled tokens (GPU-side, no host sync)."""
        ok = (tokens >= 0) & (tokens < self.vocab_size)
        self.mask[slots[ok], tokens[ok]] = 1

    def free(self, seq: Sequence) -> None:
        if seq.penalty_slot >= 0:
            self.alloc.free(seq.penalty_slot)
            seq.penalty_slot = -1
