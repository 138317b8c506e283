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
    """(1 + w) convention."""

    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(hidden_size))
        self.eps = eps

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None):
        w = self.weight + 1.0
        if residual is not None:
            residual.add_(x)
            return ops.rmsnorm(residual, w, self.eps), residual
        return ops.rmsnorm(x, w, self.eps)
