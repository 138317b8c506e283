"""Distributed process-group state: PP x DP x TP grid over RCCL/xGMI.

MI355X-native stance: one process per GPU;
``torch.distributed`` backend "nccl" IS RCCL on ROCm; xGMI is
all-to-all point-to-point (7 links x ~153 GB/s per GPU), so PP
send/recv rides exactly one link per stage pair and TP all-reduce for
small decode messages is later served by a direct-write hipIpc two-shot
AR (parallel/custom_all_reduce.py) instead of a per-link-bound ring.

Rank layout (matches the reference grid, dist_utils.py:162-196):
  global_rank = pp * (dp_size*tp_size) + dp * tp_size + tp

Group families built at init:
  * one TP group per (pp, dp)      — forward all-reduce / all-gather
  * one DP group per (pp,)         — DP-attention meta barrier + gather
  * one EP group per (pp,)         — routed-expert partial all-reduce
    (EP spans dp*tp ranks of a stage)
Reference builds TP groups twice to get a dedicated control-plane
communicator (dist_utils.py:69-76); we instead keep control traffic on
zmq/CPU entirely, so a single TP communicator suffices.
"""

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist

# module-global state (mirrors the reference's dist_utils globals)
_RANK = 0
_WORLD = 1
_PP_RANK = 0
_PP_SIZE = 1
_DP_RANK = 0
_DP_SIZE = 1
_TP_RANK = 0
_TP_SIZE = 1
_TP_GROUP: Optional[dist.ProcessGroup] = None
_DP_GROUP: Optional[dist.ProcessGroup] = None
_EP_GROUP: Optional[dist.ProcessGroup] = None
_DEVICE = "cpu"
_INITIALIZED = False


def init_distributed(config=None, rank: Optional[int] = None,
                     world_size: Optional[int] = None,
                     backend: Optional[str] = None,
                     pp_size: int = 1, dp_size: int = 1, tp_size: int = 1,
                     master_addr: str = "127.0.0.1",
                     master_port: int = 29500) -> None:
    """Initialize the grid. With config given, sizes come from it.
    Safe to call with world_size == 1 and no env: stays single-process."""
    global _RANK, _WORLD, _PP_RANK, _PP_SIZE, _DP_RANK, _DP_SIZE
    global _TP_RANK, _TP_SIZE, _TP_GROUP, _DP_GROUP, _EP_GROUP
    global _DEVICE, _INITIALIZED

    if config is not None:
        pp_size, dp_size, tp_size = (config.pp_size, config.dp_size,
                                     config.tp_size)
        master_addr = config.master_addr
        master_port = config.master_port

    world = pp_size * dp_size * tp_size
    if world_size is not None:
        assert world_size == world
    if rank is None:
        rank = int(os.environ.get("RANK", "0"))
    _RANK, _WORLD = rank, world
    stage = dp_size * tp_size
    _PP_RANK = rank // stage
    _DP_RANK = (rank % stage) // tp_size
    _TP_RANK = rank % tp_size
    _PP_SIZE, _DP_SIZE, _TP_SIZE = pp_size, dp_size, tp_size

    use_gpu = torch.cuda.is_available()
    _DEVICE = f"cuda:{os.environ.get('LOCAL_RANK', rank % max(1, torch.cuda.device_count()) if use_gpu else 0)}" if use_gpu else "cpu"

    if world == 1:
        _INITIALIZED = True
        return

    if backend is None:
        backend = "nccl" if use_gpu else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", master_addr)
        os.environ.setdefault("MASTER_PORT", str(master_port))
        if use_gpu:
            torch.cuda.set_device(int(str(_DEVICE).split(":")[1]))
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(minutes=30))

    # ---- TP groups: one per (pp, dp) ----
    for pp in range(pp_size):
        for dp in range(dp_size):
            ranks = [pp * stage + dp * tp_size + t for t in range(tp_size)]
            g = dist.new_group(ranks) if tp_size > 1 else None
            if rank in ranks:
                _TP_GROUP = g
    # ---- DP groups: one per pp stage, fixed tp rank ----
    for pp in range(pp_size):
        for t in range(tp_size):
            ranks = [pp * stage + dp * tp_size + t for dp in range(dp_size)]
            g = dist.new_group(ranks) if dp_size > 1 else None
            if rank in ranks:
                _DP_GROUP = g
    # ---- EP groups: all ranks of a stage ----
    for pp in range(pp_size):
        ranks = [pp * stage + i for i in range(stage)]
        g = dist.new_group(ranks) if stage > 1 else None
        if rank in ranks:
            _EP_GROUP = g
    _INITIALIZED = True
    if use_gpu and tp_size > 1:
        # opt-in hipIpc/xGMI custom AR (GLLM_CUSTOM_AR=1)
        from gllm_amd.parallel.custom_all_reduce import \
            init_custom_all_reduce
        init_custom_all_reduce()


def destroy_distributed() -> None:
    global _INITIALIZED, _TP_GROUP, _DP_GROUP, _EP_GROUP
    if dist.is_initialized():
        dist.destroy_process_group()
    _TP_GROUP = _DP_GROUP = _EP_GROUP = None
    _INITIALIZED = False


# ---------------------------------------------------------------- accessors
def get_rank(): return _RANK
def get_world_size(): return _WORLD
def get_pp_rank(): return _PP_RANK
def get_pp_size(): return _PP_SIZE
def get_dp_rank(): return _DP_RANK
def get_dp_size(): return _DP_SIZE
def get_tp_rank(): return _TP_RANK
def get_tp_size(): return _TP_SIZE
def get_ep_rank(): return _DP_RANK * _TP_SIZE + _TP_RANK
def get_ep_size(): return _DP_SIZE * _TP_SIZE
def get_tp_group(): return _TP_GROUP
def get_dp_group(): return _DP_GROUP
def get_ep_group(): return _EP_GROUP
def is_first_pp_rank(): return _PP_RANK == 0
def is_last_pp_rank(): return _PP_RANK == _PP_SIZE - 1


def get_next_pp_rank() -> int:
    return (_RANK + _DP_SIZE * _TP_SIZE) % _WORLD


def get_prev_pp_rank() -> int:
    return (_RANK - _DP_SIZE * _TP_SIZE) % _WORLD


# ---------------------------------------------------------------- collectives
def tensor_parallel_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """Sum-all-reduce across the TP group. Hook point for the hipIpc
    two-shot xGMI AR (decode-size messages) once registered."""
    if _TP_SIZE == 1:
        return t
    from gllm_amd.parallel import custom_all_reduce
    out = custom_all_reduce.try_custom_all_reduce(t)
    if out is not None:
        return out
    dist.all_reduce(t, group=_TP_GROUP)
    return t


def tensor_parallel_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    """All-gather across TP, concatenating along ``dim`` (0 or -1)."""
    if _TP_SIZE == 1:
        return t
    t = t.contiguous()
    # all_gather_into_tensor concatenates along dim 0 of a flat output
    out = torch.empty((_TP_SIZE * t.shape[0],) + tuple(t.shape[1:]),
                      dtype=t.dtype, device=t.device)
    dist.all_gather_into_tensor(out, t, group=_TP_GROUP)
    if dim == 0:
        return out
    assert dim in (-1, t.dim() - 1), "only dim 0 / -1 supported"
    chunks = out.reshape((_TP_SIZE,) + tuple(t.shape))
    return torch.cat(list(chunks.unbind(0)), dim=-1)


def ep_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if _EP_GROUP is None:
        return t
    dist.all_reduce(t, group=_EP_GROUP)
    return t


# ------------------------------------------------------- DP attention
# Replica parallelism (reference dist_utils.py:266-352 + worker.py
# _schedule_forward_dp): each DP replica owns its own scheduler + KV;
# MoE layers span replicas via EP = DP x TP, so every forward round is
# entered in lockstep. The per-round metadata barrier gathers
# (num_tokens_this_round, has_pending_work) from each replica; idle
# replicas run a 1-token dummy forward so the MoE collectives match.
# The per-forward token counts are published module-globally for the
# MoE layer's padded gather (set around each forward by the engine).
_DP_FWD_COUNTS: Optional[List[int]] = None


def dp_meta_barrier(num_tokens: int, has_work: bool):
    """All-gather (num_tokens, has_work) across the DP group. Returns
    (counts, flags) lists indexed by dp rank."""
    if _DP_SIZE == 1:
        return [num_tokens], [1 if has_work else 0]
    dev = _DEVICE if str(_DEVICE).startswith("cuda") else "cpu"
    local = torch.tensor([num_tokens, 1 if has_work else 0],
                         dtype=torch.int64, device=dev)
    out = torch.empty(2 * _DP_SIZE, dtype=torch.int64, device=dev)
    dist.all_gather_into_tensor(out, local, group=_DP_GROUP)
    pairs = out.view(_DP_SIZE, 2).tolist()
    return [p[0] for p in pairs], [p[1] for p in pairs]


def dp_all_gather(t: torch.Tensor) -> torch.Tensor:
    """All-gather rows across the DP group (uniform row count on every
    replica — callers pad to max first): [R, ...] -> [dp*R, ...]."""
    if _DP_SIZE == 1:
        return t
    t = t.contiguous()
    out = torch.empty((_DP_SIZE * t.shape[0],) + tuple(t.shape[1:]),
                      dtype=t.dtype, device=t.device)
    dist.all_gather_into_tensor(out, t, group=_DP_GROUP)
    return out


def set_dp_forward_counts(counts: Optional[List[int]]) -> None:
    global _DP_FWD_COUNTS
    _DP_FWD_COUNTS = counts


def get_dp_forward_counts() -> Optional[List[int]]:
    return _DP_FWD_COUNTS


# ---------------------------------------------------------------- PP p2p
def send_pp_data(tensors: List[torch.Tensor], dst: int) -> None:
    """Send hidden_states (+residual) to the next stage. One xGMI link
    carries this leg; caller overlaps it on a dedicated HIP stream."""
    for t in tensors:
        dist.send(t.contiguous(), dst=dst)


def recv_pp_data(shapes, dtype, device, src: int) -> List[torch.Tensor]:
    out = []
    for shape in shapes:
        t = torch.empty(shape, dtype=dtype, device=device)
        dist.recv(t, src=src)
        out.append(t)
    return out
