"""Qwen3-VL vision tower with deepstack multiscale features.

Parity target: reference models/qwen3_vl.py Qwen3_VisionTransformer
(:193-433): Conv3d patch embed WITH bias, a learned absolute position
embedding bilinearly interpolated to the image grid, LayerNorm blocks
with a plain (non-gated) linear_fc1/act/linear_fc2 MLP, full
(per-image) attention, and DEEPSTACK: the outputs of
``deepstack_visual_indexes`` blocks pass through their own postshuffle
PatchMergers and are concatenated onto the final merger output —
[N_merged, out_hidden * (1 + len(deepstack))]. The LM side adds level d
at the image rows after decoder layer d (models/llama_family.py).
"""

from typing import List, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from gllm_amd.models.qwen2_vl_vision import (VisionAttention,
                                             VisionRotaryEmbedding,
                                             rot_pos_emb)


class Qwen3VisionMLP(nn.Module):
    def __init__(self, dim, hidden, act, dtype=None):
        super().__init__()
        self.linear_fc1 = nn.Linear(dim, hidden, bias=True, dtype=dtype)
        self.linear_fc2 = nn.Linear(hidden, dim, bias=True, dtype=dtype)
        self.act = act

    def forward(self, x):
        return self.linear_fc2(self.act(self.linear_fc1(x)))


class Qwen3VisionBlock(nn.Module):
    def __init__(self, dim, num_heads, mlp_hidden, act, dtype=None):
        super().__init__()
        self.norm1 = nn.LayerNorm(dim, eps=1e-6, dtype=dtype)
        self.norm2 = nn.LayerNorm(dim, eps=1e-6, dtype=dtype)
        self.attn = VisionAttention(dim, num_heads, dtype=dtype)
        self.mlp = Qwen3VisionMLP(dim, mlp_hidden, act, dtype=dtype)

    def forward(self, x, cu_seqlens, rotary):
        x = x + self.attn(self.norm1(x), cu_seqlens, rotary)
        x = x + self.mlp(self.norm2(x))
        return x


class Qwen3PatchMerger(nn.Module):
    """norm -> (shuffle to merge units) -> fc1/GELU/fc2. postshuffle
    variant norms AFTER flattening merge units (deepstack mergers)."""

    def __init__(self, d_model, context_dim, merge_size,
                 use_postshuffle_norm=False, dtype=None):
        super().__init__()
        self.hidden_size = context_dim * (merge_size ** 2)
        self.use_postshuffle_norm = use_postshuffle_norm
        self.norm = nn.LayerNorm(
            self.hidden_size if use_postshuffle_norm else context_dim,
            eps=1e-6, dtype=dtype)
        self.linear_fc1 = nn.Linear(self.hidden_size, self.hidden_size,
                                    bias=True, dtype=dtype)
        self.linear_fc2 = nn.Linear(self.hidden_size, d_model, bias=True,
                                    dtype=dtype)

    def forward(self, x):
        if self.use_postshuffle_norm:
            x = self.norm(x.view(-1, self.hidden_size))
        else:
            x = self.norm(x).view(-1, self.hidden_size)
        return self.linear_fc2(F.gelu(self.linear_fc1(x)))


class Qwen3VisionTransformer(nn.Module):
    """visual.* of Qwen3-VL checkpoints."""

    def __init__(self, vcfg, dtype=None):
        super().__init__()
        g = lambda k, d=None: (vcfg.get(k, d) if isinstance(vcfg, dict)
                               else getattr(vcfg, k, d))
        self.spatial_merge_size = g("spatial_merge_size", 2)
        self.patch_size = g("patch_size", 16)
        self.temporal_patch_size = g("temporal_patch_size", 2)
        dim = g("hidden_size")
        in_ch = g("in_channels", 3)
        self.deepstack_visual_indexes = list(
            g("deepstack_visual_indexes", []) or [])
        self.num_position_embeddings = g("num_position_embeddings", 2304)
        self.num_grid_per_side = int(self.num_position_embeddings ** 0.5)
        k = (self.temporal_patch_size, self.patch_size, self.patch_size)
        self.patch_embed = nn.ModuleDict()  # placeholder for name nesting
        self.patch_embed["proj"] = nn.Conv3d(in_ch, dim, kernel_size=k,
                                             stride=k, bias=True,
                                             dtype=dtype)
        self.pos_embed = nn.Embedding(self.num_position_embeddings, dim,
                                      dtype=dtype)
        num_heads = g("num_heads", 16)
        self.head_dim = dim // num_heads
        self.rotary = VisionRotaryEmbedding(self.head_dim // 2)
        act = F.silu if g("hidden_act", "silu") == "silu" \
            else lambda x: F.gelu(x, approximate="tanh")
        self.blocks = nn.ModuleList([
            Qwen3VisionBlock(dim, num_heads,
                             g("intermediate_size", dim * 4), act,
                             dtype=dtype)
            for _ in range(g("depth", 27))])
        out_h = g("out_hidden_size", dim)
        self.merger = Qwen3PatchMerger(out_h, dim,
                                       self.spatial_merge_size,
                                       dtype=dtype)
        self.deepstack_merger_list = nn.ModuleList([
            Qwen3PatchMerger(out_h, dim, self.spatial_merge_size,
                             use_postshuffle_norm=True, dtype=dtype)
            for _ in self.deepstack_visual_indexes])
        self.out_hidden_size = out_h * (
            1 + len(self.deepstack_visual_indexes))

    def _embed_patches(self, x):
        L = x.shape[0]
        proj = self.patch_embed["proj"]
        x = x.view(L, proj.in_channels, self.temporal_patch_size,
                   self.patch_size, self.patch_size)
        return proj(x.to(proj.weight.dtype)).view(L, -1)

    def _interp_pos_embed(self, grid_thw) -> torch.Tensor:
        """Bilinear interpolation of the learned grid pos-embed onto the
        image grid, reordered into merge units (reference
        fast_pos_embed_interpolate)."""
        n = self.num_grid_per_side
        m = self.spatial_merge_size
        dim = self.pos_embed.embedding_dim
        outs = []
        for t, h, w in grid_thw:
            hi = torch.linspace(0, n - 1, h)
            wi = torch.linspace(0, n - 1, w)
            h0 = hi.long()
            w0 = wi.long()
            h1 = (h0 + 1).clamp(max=n - 1)
            w1 = (w0 + 1).clamp(max=n - 1)
            dh = (hi - h0).unsqueeze(1)
            dw = (wi - w0).unsqueeze(0)
            e = self.pos_embed.weight.float()

            def at(hr, wr):
                return e[(hr.unsqueeze(1) * n + wr.unsqueeze(0))]

            emb = (at(h0, w0) * (1 - dh).unsqueeze(-1) * (1 - dw).unsqueeze(-1)
                   + at(h0, w1) * (1 - dh).unsqueeze(-1) * dw.unsqueeze(-1)
                   + at(h1, w0) * dh.unsqueeze(-1) * (1 - dw).unsqueeze(-1)
                   + at(h1, w1) * dh.unsqueeze(-1) * dw.unsqueeze(-1))
            emb = emb.reshape(h // m, m, w // m, m, dim)
            emb = emb.permute(0, 2, 1, 3, 4).reshape(-1, dim)
            outs.append(emb.repeat(t, 1))
        return torch.cat(outs, dim=0).to(self.pos_embed.weight.dtype)

    def forward(self, pixel_values: torch.Tensor,
                grid_thw: List[Tuple[int, int, int]]) -> torch.Tensor:
        x = self._embed_patches(pixel_values)
        x = x + self._interp_pos_embed(grid_thw)
        rotary = rot_pos_emb(grid_thw, self.head_dim,
                             self.spatial_merge_size, self.rotary)
        cu = [0]
        for t, h, w in grid_thw:
            for _ in range(t):
                cu.append(cu[-1] + h * w)
        deepstack = []
        for i, blk in enumerate(self.blocks):
            x = blk(x, cu, rotary)
            if i in self.deepstack_visual_indexes:
                j = self.deepstack_visual_indexes.index(i)
                deepstack.append(self.deepstack_merger_list[j](x))
        out = self.merger(x)
        return torch.cat([out] + deepstack, dim=1)
