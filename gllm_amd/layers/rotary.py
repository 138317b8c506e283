"""Rotary position embedding with host-precomputed cos/sin cache.

MI355X note: on-device sinf/cosf per element turns a memory-bound op
VALU-bound (cdna_hip_programming.md App. B) — the cache is built once on
host, stored [max_pos, rot_dim] as [cos | sin] halves, and the HIP
kernel just gathers it.

Scaling variants at parity with the reference
(layers/rotary_embedding.py): linear, llama3, yarn.
"""

import math
from typing import Dict, Optional, Tuple

import torch
import torch.nn as nn

from gllm_amd import ops


class RotaryEmbedding(nn.Module):
    def __init__(self, head_dim: int, rot_dim: int, max_position: int,
                 base: float, is_neox: bool = True,
                 scaling_factor: float = 1.0):
        super().__init__()
        self.head_dim = head_dim
        self.rot_dim = rot_dim
        self.max_position = max_position
        self.base = base
        self.is_neox = is_neox
        self.scaling_factor = scaling_factor
        cache = self._build_cache()
        self.register_buffer("cos_sin_cache", cache, persistent=False)

    def _inv_freq(self) -> torch.Tensor:
        return 1.0 / (self.base ** (
            torch.arange(0, self.rot_dim, 2, dtype=torch.float32)
            / self.rot_dim))

    def _positions(self) -> torch.Tensor:
        n = int(self.max_position * max(1.0, self.scaling_factor))
        t = torch.arange(n, dtype=torch.float32)
        if self.scaling_factor != 1.0:
            t = t / self.scaling_factor
        return t

    def _build_cache(self) -> torch.Tensor:
        inv_freq = self._inv_freq()
        t = self._positions()
        freqs = torch.outer(t, inv_freq)              # [P, rot/2]
        return torch.cat([freqs.cos(), freqs.sin()], dim=-1)  # [P, rot_dim]

    def forward(self, positions: torch.Tensor, q: torch.Tensor,
                k: torch.Tensor):
        return ops.rotary_embedding(
            positions, q, k, self.head_dim,
            self.cos_sin_cache.to(device=q.device), self.is_neox)


class Llama3RotaryEmbedding(RotaryEmbedding):
    def __init__(self, head_dim, rot_dim, max_position, base, is_neox,
                 factor, low_freq_factor, high_freq_factor,
                 original_max_position):
        self.factor = factor
        self.low_freq_factor = low_freq_factor
        self.high_freq_factor = high_freq_factor
        self.original_max_position = original_max_position
        super().__init__(head_dim, rot_dim, max_position, base, is_neox)

    def _inv_freq(self):
        inv = super()._inv_freq()
        low_wl = self.original_max_position / self.low_freq_factor
        high_wl = self.original_max_position / self.high_freq_factor
        wl = 2 * math.pi / inv
        smooth = ((self.original_max_position / wl - self.low_freq_factor)
                  / (self.high_freq_factor - self.low_freq_factor))
        smooth = smooth.clamp(0.0, 1.0)
        scaled = inv / self.factor
        blended = (1 - smooth) * scaled + smooth * inv
        out = torch.where(wl > low_wl, scaled, inv)
        mid = (wl <= low_wl) & (wl >= high_wl)
        out = torch.where(mid, blended, out)
        return out


class YaRNRotaryEmbedding(RotaryEmbedding):
    def __init__(self, head_dim, rot_dim, max_position, base, is_neox,
                 factor, original_max_position, beta_fast=32, beta_slow=1,
                 attn_factor=1.0, mscale: Optional[float] = None,
                 mscale_all_dim: Optional[float] = None):
        self.factor = factor
        self.original_max_position = original_max_position
        self.beta_fast = beta_fast
        self.beta_slow = beta_slow
        self.attn_factor = attn_factor
        if mscale is not None and mscale_all_dim is not None:
            def _ms(s, m):
                return 1.0 if s <= 1 else 0.1 * m * math.log(s) + 1.0
            self.mscale = (_ms(factor, mscale)
                           / _ms(factor, mscale_all_dim) * attn_factor)
        else:
            self.mscale = (0.1 * math.log(factor) + 1.0
                           if factor > 1 else 1.0) * attn_factor
        super().__init__(head_dim, rot_dim, max_position, base, is_neox)

    def _yarn_find_dim(self, num_rot):
        return (self.rot_dim * math.log(
            self.original_max_position / (num_rot * 2 * math.pi))
            / (2 * math.log(self.base)))

    def _inv_freq(self):
        pos_freqs = self.base ** (
            torch.arange(0, self.rot_dim, 2, dtype=torch.float32)
            / self.rot_dim)
        inv_extrap = 1.0 / pos_freqs
        inv_interp = 1.0 / (self.factor * pos_freqs)
        low = max(math.floor(self._yarn_find_dim(self.beta_fast)), 0)
        high = min(math.ceil(self._yarn_find_dim(self.beta_slow)),
                   self.rot_dim - 1)
        if low == high:
            high += 0.001
        ramp = ((torch.arange(self.rot_dim // 2, dtype=torch.float32) - low)
                / (high - low)).clamp(0, 1)
        mask = 1.0 - ramp
        return inv_interp * (1 - mask) + inv_extrap * mask

    def _positions(self):
        n = int(self.max_position * self.factor)
        return torch.arange(n, dtype=torch.float32)

    def _build_cache(self):
        inv_freq = self._inv_freq()
        t = self._positions()
        freqs = torch.outer(t, inv_freq)
        return torch.cat([freqs.cos() * self.mscale,
                          freqs.sin() * self.mscale], dim=-1)


_ROPE_CACHE: Dict[Tuple, RotaryEmbedding] = {}


def get_rope(head_dim: int, rot_dim: int, max_position: int, base: float,
             is_neox: bool = True, rope_scaling: Optional[dict] = None
             ) -> RotaryEmbedding:
    def _freeze(v):
        return tuple(v) if isinstance(v, list) else v
    key = (head_dim, rot_dim, max_position, base, is_neox,
           tuple(sorted((k, _freeze(v)) for k, v in rope_scaling.items()))
           if rope_scaling else None)
    if key in _ROPE_CACHE:
        return _ROPE_CACHE[key]
    if rope_scaling is None:
        rope = RotaryEmbedding(head_dim, rot_dim, max_position, base, is_neox)
    else:
        rtype = rope_scaling.get("rope_type",
                                 rope_scaling.get("type", "linear"))
        if rtype == "linear":
            rope = RotaryEmbedding(
                head_dim, rot_dim, max_position, base, is_neox,
                scaling_factor=rope_scaling["factor"])
        elif rtype == "llama3":
            rope = Llama3RotaryEmbedding(
                head_dim, rot_dim, max_position, base, is_neox,
                rope_scaling["factor"],
                rope_scaling.get("low_freq_factor", 1.0),
                rope_scaling.get("high_freq_factor", 4.0),
                rope_scaling.get("original_max_position_embeddings", 8192))
        elif rtype == "yarn":
            rope = YaRNRotaryEmbedding(
                head_dim, rot_dim,
                rope_scaling.get("original_max_position_embeddings",
                                 max_position),
                base, is_neox, rope_scaling["factor"],
                rope_scaling.get("original_max_position_embeddings",
                                 max_position),
                beta_fast=rope_scaling.get("beta_fast", 32),
                beta_slow=rope_scaling.get("beta_slow", 1),
                attn_factor=rope_scaling.get("attention_factor", 1.0) or 1.0,
                mscale=rope_scaling.get("mscale"),
                mscale_all_dim=rope_scaling.get("mscale_all_dim"))
        elif rtype in ("mrope", "default") and \
                rope_scaling.get("mrope_section"):
            from gllm_amd.layers.mrope import MRotaryEmbedding
            rope = MRotaryEmbedding(
                head_dim, rot_dim, max_position, base,
                list(rope_scaling["mrope_section"]),
                mrope_interleaved=bool(
                    rope_scaling.get("mrope_interleaved", False)))
        elif rtype == "default":
            rope = RotaryEmbedding(head_dim, rot_dim, max_position, base,
                                   is_neox)
        else:
            raise ValueError(f"unsupported rope_type {rtype}")
    _ROPE_CACHE[key] = rope
    return rope
