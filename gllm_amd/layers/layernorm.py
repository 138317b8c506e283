"""RMSNorm layers over the fused HIP kernels (reference: layers/layernorm.py)."""

from typing import Optional

import torch
import torch.nn as nn

from gllm_amd import ops


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None):
        if residual is not None:
            return ops.fused_add_rmsnorm(x, residual, self.weight, self.eps)
        return ops.rmsnorm(x, self.weight, self.eps)


class GemmaRMSNorm(nn.Module):
    """(1 + w) convention (Qwen3.5/Gemma checkpoints store w - 1).

    The effective scale (1 + w) is cached after load so the hot path
    runs the same fused add+rmsnorm kernel as RMSNorm."""

    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(hidden_size))
        self.eps = eps
        self._w1 = None

    def _scale(self):
        if self._w1 is None or self._w1.device != self.weight.device:
            self._w1 = (self.weight + 1.0).contiguous()
        return self._w1

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None):
        if residual is not None:
            return ops.fused_add_rmsnorm(x, residual, self._scale(),
                                         self.eps)
        return ops.rmsnorm(x, self._scale(), self.eps)
