"""Encoder-disaggregation wire protocol.

Parity target: reference disagg/protocol.py (EncoderJob / MmItemMeta /
notification wire format). Transport is the length-prefixed pickle
framing shared with the multi-node control plane
(engine/multinode.py send_msg/recv_msg).

Round-1 data plane: embeddings return over the same TCP channel (CPU
and single-node correct). Round-2 (ROADMAP.md): the MI355X data plane —
encoder ranks write embeddings straight into per-LM-rank hipIpc slot
pools over xGMI (the reference uses NIXL/UCX GPU WRITEs,
nixl_transfer.py:122-298), with the TCP channel carrying only readiness
notifications.
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


@dataclasses.dataclass
class EncoderResult:
    job_id: int
    # None => cache miss on a pixel-less probe: resend with pixels
    embeds: Optional[torch.Tensor] = None
    cached: bool = False
    error: Optional[str] = None
