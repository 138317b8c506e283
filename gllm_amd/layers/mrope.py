"""Multimodal rotary embedding (MRoPE) — Qwen2-VL family.

Parity target: reference layers/rotary_embedding.py MRotaryEmbedding
(:405-570, :607+): 3-section rope where the head-dim frequency bands are
split into (temporal, height, width) sections and each section reads its
own position stream; plus the position math that assigns (t, h, w)
coordinates to vision tokens and a shared scalar position to text.

Engine status: unit-tested building block for the round-2 multimodal
path (vision tower + embedding merge + mm scheduler plumbing).
"""

from typing import List, Tuple

import torch

from gllm_amd.layers.rotary import RotaryEmbedding


class MRotaryEmbedding(RotaryEmbedding):
    def __init__(self, head_dim: int, rot_dim: int, max_position: int,
                 base: float, mrope_section: List[int],
                 mrope_interleaved: bool = False):
        assert sum(mrope_section) * 2 == rot_dim, \
            (mrope_section, rot_dim)
        self.mrope_section = mrope_section
        # Qwen3-VL / Qwen3.5 layout: frequency pair k reads stream
        # T/H/W by k % 3 (within each section's reach) instead of
        # contiguous [T | H | W] bands (reference rotary_embedding.py
        # apply_interleaved_rope, :588-605)
        self.mrope_interleaved = mrope_interleaved
        super().__init__(head_dim, rot_dim, max_position, base,
                         is_neox=True)

    def forward(self, positions: torch.Tensor, q: torch.Tensor,
                k: torch.Tensor):
        """positions: [3, T] (t/h/w streams) or [T] (plain text fallback).

        Torch implementation (the HIP kernel variant lands with the full
        multimodal path): build per-token cos/sin by gathering each
        section's band from its stream, then apply neox rotation.
        """
        if positions.dim() == 1:
            return super().forward(positions, q, k)
        assert positions.shape[0] == 3
        cs = self.cos_sin_cache.to(q.device)      # [P, rot_dim]
        half = self.rot_dim // 2
        cos_full = cs[:, :half]
        sin_full = cs[:, half:]
        if self.mrope_interleaved:
            sec = self.mrope_section
            cos = cos_full[positions[0]].clone()     # [T, half]
            sin = sin_full[positions[0]].clone()
            for axis in (1, 2):
                sl = slice(axis, sec[axis] * 3, 3)
                cos[:, sl] = cos_full[positions[axis]][:, sl]
                sin[:, sl] = sin_full[positions[axis]][:, sl]
            cos, sin = cos.float(), sin.float()
        else:
            cos_parts, sin_parts = [], []
            off = 0
            for i, n in enumerate(self.mrope_section):
                idx = positions[i]
                cos_parts.append(cos_full[idx][:, off:off + n])
                sin_parts.append(sin_full[idx][:, off:off + n])
                off += n
            cos = torch.cat(cos_parts, dim=-1).float()   # [T, half]
            sin = torch.cat(sin_parts, dim=-1).float()
        for t in (q, k):
            T = t.shape[0]
            x = t.unflatten(-1, (-1, self.head_dim))
            rot = x[..., :self.rot_dim].float()
            x1, x2 = rot[..., :half], rot[..., half:]
            c, s = cos.unsqueeze(1), sin.unsqueeze(1)
            o1 = x1 * c - x2 * s
            o2 = x2 * c + x1 * s
            x[..., :self.rot_dim].copy_(
                torch.cat([o1, o2], dim=-1).to(t.dtype))
        return q, k

    # ------------------------------------------------------------------
    @staticmethod
    def get_input_positions(
            input_tokens: List[int], image_token_id: int,
            image_grids: List[Tuple[int, int, int]],
            spatial_merge_size: int = 2,
            ) -> Tuple[torch.Tensor, int]:
        """Build [3, T] (t, h, w) positions for a prompt whose image
        spans are already expanded to ``image_token_id`` runs, one run
        per grid (t, h, w) in patch units. Returns (positions, the delta
        to add to future decode positions).

        Text tokens advance all three streams together; each vision
        token gets (t, h/m, w/m) offsets from the span start (reference
        MRotaryEmbedding.get_input_positions).
        """
        T = len(input_tokens)
        pos = torch.zeros(3, T, dtype=torch.long)
        st = 0            # current base position
        i = 0
        img = 0
        m = spatial_merge_size
        while i < T:
            if input_tokens[i] == image_token_id and img < len(image_grids):
                t_g, h_g, w_g = image_grids[img]
                h_m, w_m = h_g // m, w_g // m
                n = t_g * h_m * w_m
                for j in range(n):
                    tt = j // (h_m * w_m)
                    hh = (j % (h_m * w_m)) // w_m
                    ww = j % w_m
                    pos[0, i + j] = st + tt
                    pos[1, i + j] = st + hh
                    pos[2, i + j] = st + ww
                st = st + max(t_g, h_m, w_m)
                i += n
                img += 1
            else:
                pos[:, i] = st
                st += 1
                i += 1
        return pos, st

    @staticmethod
    def get_next_input_positions(mrope_delta: int, pos: int
                                 ) -> torch.Tensor:
        """Decode-time positions: all three streams share one scalar."""
        return torch.full((3, 1), mrope_delta + pos, dtype=torch.long)
