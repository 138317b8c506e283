"""Custom all-reduce over hipIpc-mapped peer buffers (xGMI).

Reference parity: distributed/custom_all_reduce.py (sgl_kernel NVLink
two-shot AR + cuda_wrapper.py ctypes IPC — here the hipIpc calls live
in the extension, ops/csrc/custom_ar.hip, and the kernel reads peers
straight over point-to-point xGMI links).

Round-2 (v2): one-shot for small payloads + two-shot reduce-scatter/
gather above ~256 KB, with the barrier epoch kept in a DEVICE cell so
the collective is hipGraph-capturable without the reference's
graph-buffer registration handshake (custom_all_reduce.py:266-391) —
every rank issues the same collective sequence, so the cells stay in
lock-step across replays and eager calls alike. ON by default for
TP 2..8 (disable with GLLM_CUSTOM_AR=0); init failure (no IPC) falls
back to RCCL cleanly. Eligibility mirrors should_custom_ar: bf16/fp32,
payload <= max_bytes.
"""

import os
from typing import Optional

import torch

_MAX_BYTES = 8 << 20
_INSTANCE = None
_INIT_TRIED = False


class CustomAllReduce:
    def __init__(self, group, rank: int, world: int,
                 max_bytes: int = _MAX_BYTES):
        from gllm_amd import _kernels as K
        import torch.distributed as dist
        self.K = K
        self.rank = rank
        self.world = world
        self.max_bytes = max_bytes
        self.my_ptr, handle = K.car_alloc_v2(max_bytes)
        # exchange 64-byte hipIpc handles over the group (byte tensors,
        # reference custom_all_reduce.py:57-78)
        h = torch.zeros(len(handle), dtype=torch.uint8)
        h[:] = torch.tensor(list(handle), dtype=torch.uint8)
        gathered = [torch.zeros_like(h) for _ in range(world)]
        dist.all_gather(gathered, h, group=group)
        self.ptrs = []
        self._opened = []
        for r in range(world):
            if r == rank:
                self.ptrs.append(self.my_ptr)
            else:
                p = K.car_open(bytes(gathered[r].tolist()))
                self.ptrs.append(p)
                self._opened.append(p)
        self.epoch = 0
        # all peers mapped before anyone reduces
        dist.barrier(group=group)

    def eligible(self, t: torch.Tensor) -> bool:
        if t.dtype not in (torch.bfloat16, torch.float32):
            return False
        if t.numel() * t.element_size() > self.max_bytes:
            return False
        return True

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        t = t.contiguous()
        self.K.car_all_reduce_v2(t, self.ptrs, self.rank, self.world,
                                 self.max_bytes)
        return t

    def close(self):
        for p in self._opened:
            self.K.car_close(p)
        self.K.car_free(self.my_ptr)


def init_custom_all_reduce() -> Optional[CustomAllReduce]:
    """Build the AR for the current TP group (call after
    init_distributed). No-op unless GLLM_CUSTOM_AR=1."""
    global _INSTANCE, _INIT_TRIED
    _INIT_TRIED = True
    if os.environ.get("GLLM_CUSTOM_AR", "1") == "0":
        return None
    if not torch.cuda.is_available():
        return None
    from gllm_amd.parallel import state as S
    world = S.get_tp_size()
    if not (2 <= world <= 8):
        return None
    try:
        _INSTANCE = CustomAllReduce(S.get_tp_group(), S.get_tp_rank(),
                                    world)
    except Exception:  # pragma: no cover - driver without IPC support
        from gllm_amd.logger import logger
        logger.exception("custom AR init failed; falling back to RCCL")
        _INSTANCE = None
    return _INSTANCE


def try_custom_all_reduce(t: torch.Tensor) -> Optional[torch.Tensor]:
    if _INSTANCE is None:
        return None
    if not _INSTANCE.eligible(t):
        return None
    return _INSTANCE.all_reduce(t)
