import torch.nn as nn

from gllm_amd import ops


class SiluAndMul(nn.Module):
    def forward(self, x):
        return ops.silu_and_mul(x)
