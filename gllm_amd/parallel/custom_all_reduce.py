"""xGMI direct-write custom all-reduce (placeholder hook, round 1).

The reference uses a cudaIPC two-shot NVLink AR for <=8 MB TP messages
(distributed/custom_all_reduce.py). The MI355X equivalent is a
hipIpc-mapped two-shot AR over xGMI's all-to-all point-to-point links
(reduce-scatter by direct peer writes + all-gather) — implemented in a
later round as a HIP kernel with graph-capture buffer registration.
Until then every call falls through to RCCL.
"""

from typing import Optional

import torch

_ENABLED = False


def try_custom_all_reduce(t: torch.Tensor) -> Optional[torch.Tensor]:
    if not _ENABLED:
        return None
    return None
