"""Vocab-parallel embedding and LM head.

Parity: layers/vocab_parallel_embedding.py (vocab padded to 64, shard
mask + all-reduce for the embedding; logits all-gather for the head).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from gllm_amd.parallel import (get_tp_rank, get_tp_size,
                               tensor_parallel_all_gather,
                               tensor_parallel_all_reduce)


def pad_vocab(vocab_size: int, align: int = 64) -> int:
    return -(-vocab_size // align) * align


class VocabParallelEmbedding(nn.Module):
    def __init__(self, vocab_size: int, hidden_size: int, params_dtype=None):
        super().__init__()
        tp = get_tp_size()
        self.vocab_size = vocab_size
        self.padded_vocab = pad_vocab(vocab_size, 64 * tp)
        self.shard_size = self.padded_vocab // tp
        self.vocab_start = get_tp_rank() * self.shard_size
        self.vocab_end = self.vocab_start + self.shard_size
        dtype = params_dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(
            torch.empty(self.shard_size, hidden_size, dtype=dtype),
            requires_grad=False)
        self.weight.weight_loader = self._load

    def _load(self, param, loaded):
        n = min(self.shard_size, max(0, loaded.shape[0] - self.vocab_start))
        param.data.zero_()
        if n > 0:
            param.data[:n].copy_(loaded.narrow(0, self.vocab_start, n))

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        if get_tp_size() == 1:
            return F.embedding(input_ids, self.weight)
        mask = (input_ids >= self.vocab_start) & (input_ids < self.vocab_end)
        local = (input_ids - self.vocab_start).clamp_(0, self.shard_size - 1)
        emb = F.embedding(local, self.weight)
        emb = emb * mask.unsqueeze(-1).to(emb.dtype)
        return tensor_parallel_all_reduce(emb)


class ParallelLMHead(nn.Module):
    """Logits = H @ W_shard^T, all-gathered to the full padded vocab."""

    def __init__(self, vocab_size: int, hidden_size: int, bias: bool = False,
                 params_dtype=None):
        super().__init__()
        tp = get_tp_size()
        self.vocab_size = vocab_size
        self.padded_vocab = pad_vocab(vocab_size, 64 * tp)
        self.shard_size = self.padded_vocab // tp
        self.vocab_start = get_tp_rank() * self.shard_size
        dtype = params_dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(
            torch.empty(self.shard_size, hidden_size, dtype=dtype),
            requires_grad=False)
        self.weight.weight_loader = self._load

    def _load(self, param, loaded):
        n = min(self.shard_size, max(0, loaded.shape[0] - self.vocab_start))
        param.data.zero_()
        if n > 0:
            param.data[:n].copy_(loaded.narrow(0, self.vocab_start, n))

    def tie_to(self, embedding: VocabParallelEmbedding):
        self.weight = embedding.weight

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        from gllm_amd import ops
        logits = ops.linear(hidden, self.weight)
        if get_tp_size() > 1:
            logits = tensor_parallel_all_gather(logits, dim=-1)
        return logits[..., :self.vocab_size]
