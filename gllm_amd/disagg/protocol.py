"""Encoder-disaggregation wire protocol.

Parity target: reference disagg/protocol.py (EncoderJob / MmItemMeta /
notification wire format). Transport is the length-prefixed pickle
framing shared with the multi-node control plane
(engine/multinode.py send_msg/recv_msg).

Data planes:
* TCP (always available, cross-node): embeddings ride the result
  pickle.
* GPU-direct (r2, intra-node): the LM side pre-registers a hipIpc slot
  pool (disagg/gpu_plane.py); the encoder maps it once and WRITES each
  job's embeddings straight into the designated slot over xGMI, and the
  TCP result carries only (slot, n_tokens) — the reference's NIXL/UCX
  GPU-WRITE role (nixl_transfer.py:122-298) on hipIpc. Falls back to
  TCP when the mapping fails (cross-node / no GPU).
"""

import dataclasses
import hashlib
from typing import List, Optional, Tuple

import torch


def content_hash(pixel_values: torch.Tensor, grids) -> str:
    """Stable content key for encoder-side dedup (reference
    model_runner.py:161-221 MultiModalEmbeddingCache)."""
    h = hashlib.sha256()
    h.update(repr([tuple(g) for g in grids]).encode())
    h.update(pixel_values.numpy().tobytes())
    return h.hexdigest()


@dataclasses.dataclass
class EncoderJob:
    job_id: int
    content_hash: str
    grids: List[Tuple[int, int, int]]
    # None when the client believes the encoder has this hash cached
    pixel_values: Optional[torch.Tensor] = None
    # GPU-direct plane: destination slot in the client's registered pool
    slot: Optional[int] = None


@dataclasses.dataclass
class EncoderResult:
    job_id: int
    # None => cache miss on a pixel-less probe: resend with pixels
    # (unless via_pool: the payload went over the GPU-direct plane)
    embeds: Optional[torch.Tensor] = None
    cached: bool = False
    error: Optional[str] = None
    via_pool: bool = False
    n_tokens: int = 0
    embed_dim: int = 0


@dataclasses.dataclass
class PoolRegistration:
    """One-time hipIpc handshake: LM client -> encoder server."""
    handle: bytes
    n_slots: int
    slot_elems: int  # bf16 elements per slot
