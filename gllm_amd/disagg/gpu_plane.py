"""GPU-direct disagg data plane: hipIpc slot pool over xGMI.

Reference role: transfer/nixl_transfer.py (NIXL/UCX GPU WRITEs into
pre-registered LM-side slot regions, completion notifications as the
data gate). MI355X-native shape: the LM client allocates a pool of
fixed-size bf16 slots with ``_kernels.car_alloc`` (hipMalloc +
hipIpcGetMemHandle), ships the 64-byte handle to the encoder over the
existing TCP control channel ONCE, and the encoder maps it with
``car_open`` — after that every job's embeddings move as one
device-to-device copy over xGMI (or through the copy engine on a single
GPU) and the TCP result carries only (slot, n_tokens).

Intra-node only (hipIpc). Cross-node encoders keep the TCP payload
path; the client downgrades automatically when registration fails.
"""

from typing import Optional

import torch

from gllm_amd.logger import logger


class SlotPool:
    """LM-side owner of the shared embedding slots."""

    def __init__(self, n_slots: int, slot_elems: int):
        from gllm_amd import _kernels as K
        self.K = K
        self.n_slots = n_slots
        self.slot_elems = slot_elems
        data_bytes = n_slots * slot_elems * 2  # bf16
        self.ptr, self.handle = K.car_alloc(data_bytes)
        self._next = 0

    def acquire(self) -> int:
        """Round-robin slot assignment (the client blocks on each job's
        result before reusing a slot, so n_slots bounds in-flight jobs)."""
        s = self._next
        self._next = (self._next + 1) % self.n_slots
        return s

    def view(self, slot: int, n_tokens: int, dim: int) -> torch.Tensor:
        assert n_tokens * dim <= self.slot_elems, "embedding > slot"
        flat = self.K.car_view_tensor(
            self.ptr + 256 + slot * self.slot_elems * 2, n_tokens * dim)
        return flat.view(n_tokens, dim)

    def close(self):
        self.K.car_free(self.ptr)


class RemotePool:
    """Encoder-side mapping of a client's pool."""

    def __init__(self, reg):
        from gllm_amd import _kernels as K
        self.K = K
        self.ptr = K.car_open(bytes(reg.handle))
        self.n_slots = reg.n_slots
        self.slot_elems = reg.slot_elems

    def write(self, slot: int, emb: torch.Tensor) -> int:
        """Copy [T, D] embeddings into the mapped slot; returns T."""
        t, d = emb.shape
        assert t * d <= self.slot_elems, "embedding > slot"
        dst = self.K.car_view_tensor(
            self.ptr + 256 + slot * self.slot_elems * 2, t * d)
        dst.view(t, d).copy_(emb.to(torch.bfloat16))
        torch.cuda.synchronize()  # the TCP notification is the gate
        return t

    def close(self):
        self.K.car_close(self.ptr)


def try_make_pool(n_slots: int = 8,
                  slot_elems: int = 16384 * 1024) -> Optional[SlotPool]:
    """Build the LM-side pool when a GPU + the extension are available;
    None => TCP payload fallback."""
    if not torch.cuda.is_available():
        return None
    try:
        return SlotPool(n_slots, slot_elems)
    except Exception:
        logger.exception("hipIpc slot pool alloc failed; TCP fallback")
        return None
