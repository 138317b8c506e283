"""hipGraph-captured decode buckets (reference: CUDA-graph capture at
model_runner.py:1525-1615).

Captures the full decode forward (embed -> layers -> logits -> sample
metadata-independent part) for power-of-2 batch buckets into hipGraphs
replayed from persistent input buffers. Round-1: implemented after the
eager GPU path is validated; ``can_replay`` returns False until capture
runs.
"""

from typing import Dict, List

import torch

DECODE_BUCKETS = [1, 2, 4, 8, 16, 32, 48, 64, 96, 128, 192, 256]


class GraphRunner:
    def __init__(self, runner):
        self.runner = runner
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.captured = False

    def capture_all(self):
        # Implemented in the hipGraph pass (after eager GPU validation).
        self.captured = False

    def can_replay(self, batch_size: int) -> bool:
        return False

    def replay(self, batch, tokens, fctx):  # pragma: no cover
        raise NotImplementedError
