"""Qwen2-VL vision tower (reference: models/qwen2_5_vl.py ViT side).

Faithful Qwen2-VL ViT structure so real checkpoints load unchanged:
Conv3d patch embed (temporal_patch_size x patch x patch), blocks of
[LayerNorm -> fused-qkv attention with 2D rotary -> LayerNorm ->
QuickGELU MLP], and the 2x2 PatchMerger. Full (per-image) attention via
varlen SDPA over cu_seqlens.

Round-1 status: standalone tested tower; the LM-side embedding merge +
[3, T] mrope plumbing through the batch builder is the round-2
multimodal milestone (docs/architecture.md).
"""

from typing import List, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class VisionRotaryEmbedding(nn.Module):
    def __init__(self, dim: int, theta: float = 10000.0):
        super().__init__()
        inv = 1.0 / (theta ** (torch.arange(0, dim, 2).float() / dim))
        self.register_buffer("inv_freq", inv, persistent=False)

    def forward(self, seqlen: int) -> torch.Tensor:
        seq = torch.arange(seqlen, dtype=torch.float32,
                           device=self.inv_freq.device)
        return torch.outer(seq, self.inv_freq)


def rot_pos_emb(grid_thw, head_dim: int, merge_size: int,
                rope: VisionRotaryEmbedding) -> torch.Tensor:
    """Per-patch (h, w) rotary table [L, head_dim//2] (HF layout: patches
    ordered in merge_size blocks)."""
    pos_ids = []
    for t, h, w in grid_thw:
        hpos = torch.arange(h).unsqueeze(1).expand(-1, w)
        hpos = hpos.reshape(h // merge_size, merge_size,
                            w // merge_size, merge_size)
        hpos = hpos.permute(0, 2, 1, 3).flatten()
        wpos = torch.arange(w).unsqueeze(0).expand(h, -1)
        wpos = wpos.reshape(h // merge_size, merge_size,
                            w // merge_size, merge_size)
        wpos = wpos.permute(0, 2, 1, 3).flatten()
        pos_ids.append(
            torch.stack([hpos, wpos], dim=-1).repeat(t, 1))
    pos_ids = torch.cat(pos_ids, dim=0)
    max_size = int(pos_ids.max()) + 1
    table = rope(max_size)                       # [max, head_dim//4]
    emb = table[pos_ids].flatten(1)              # [L, head_dim//2]
    return emb


def apply_rotary_vision(x: torch.Tensor, freqs: torch.Tensor):
    """x: [L, H, D]; freqs: [L, D//2]."""
    L, H, D = x.shape
    cos = freqs.cos().unsqueeze(1)
    sin = freqs.sin().unsqueeze(1)
    x1, x2 = x[..., :D // 2].float(), x[..., D // 2:].float()
    o1 = x1 * cos - x2 * sin
    o2 = x2 * cos + x1 * sin
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


class PatchEmbed(nn.Module):
    def __init__(self, patch_size, temporal_patch_size, in_channels,
                 embed_dim, dtype=None):
        super().__init__()
        self.patch_size = patch_size
        self.temporal_patch_size = temporal_patch_size
        self.in_channels = in_channels
        self.embed_dim = embed_dim
        self.proj = nn.Conv3d(in_channels, embed_dim,
                              kernel_size=(temporal_patch_size, patch_size,
                                           patch_size),
                              stride=(temporal_patch_size, patch_size,
                                      patch_size),
                              bias=False, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # x: [L, C * tps * ps * ps] flattened patches (HF processor layout)
        L = x.shape[0]
        x = x.view(L, self.in_channels, self.temporal_patch_size,
                   self.patch_size, self.patch_size)
        return self.proj(x.to(self.proj.weight.dtype)).view(L, -1)


class VisionAttention(nn.Module):
    def __init__(self, dim, num_heads, dtype=None):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.qkv = nn.Linear(dim, dim * 3, bias=True, dtype=dtype)
        self.proj = nn.Linear(dim, dim, bias=True, dtype=dtype)

    def forward(self, x, cu_seqlens, rotary):
        L, dim = x.shape
        qkv = self.qkv(x).view(L, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.unbind(1)
        q = apply_rotary_vision(q, rotary)
        k = apply_rotary_vision(k, rotary)
        out = torch.empty(L, self.num_heads, self.head_dim, dtype=x.dtype,
                          device=x.device)
        for i in range(len(cu_seqlens) - 1):
            s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
            o = F.scaled_dot_product_attention(
                q[s:e].transpose(0, 1).float(),
                k[s:e].transpose(0, 1).float(),
                v[s:e].transpose(0, 1).float())
            out[s:e] = o.transpose(0, 1).to(x.dtype)
        return self.proj(out.reshape(L, dim))


class VisionMLP(nn.Module):
    def __init__(self, dim, hidden, dtype=None):
        super().__init__()
        self.fc1 = nn.Linear(dim, hidden, dtype=dtype)
        self.fc2 = nn.Linear(hidden, dim, dtype=dtype)

    def forward(self, x):
        h = self.fc1(x)
        h = h * torch.sigmoid(1.702 * h)         # QuickGELU
        return self.fc2(h)


class VisionBlock(nn.Module):
    def __init__(self, dim, num_heads, mlp_ratio, dtype=None):
        super().__init__()
        self.norm1 = nn.LayerNorm(dim, eps=1e-6, dtype=dtype)
        self.norm2 = nn.LayerNorm(dim, eps=1e-6, dtype=dtype)
        self.attn = VisionAttention(dim, num_heads, dtype=dtype)
        self.mlp = VisionMLP(dim, int(dim * mlp_ratio), dtype=dtype)

    def forward(self, x, cu_seqlens, rotary):
        x = x + self.attn(self.norm1(x), cu_seqlens, rotary)
        x = x + self.mlp(self.norm2(x))
        return x


class PatchMerger(nn.Module):
    def __init__(self, dim, context_dim, merge_size, dtype=None):
        super().__init__()
        self.hidden_size = context_dim * (merge_size ** 2)
        self.ln_q = nn.LayerNorm(context_dim, eps=1e-6, dtype=dtype)
        self.mlp = nn.Sequential(
            nn.Linear(self.hidden_size, self.hidden_size, dtype=dtype),
            nn.GELU(),
            nn.Linear(self.hidden_size, dim, dtype=dtype))

    def forward(self, x):
        return self.mlp(self.ln_q(x).view(-1, self.hidden_size))


class Qwen2VisionTransformer(nn.Module):
    """visual.* of Qwen2-VL checkpoints."""

    def __init__(self, vcfg, dtype=None):
        super().__init__()
        g = lambda k, d=None: getattr(vcfg, k, d)
        self.spatial_merge_size = g("spatial_merge_size", 2)
        embed_dim = g("embed_dim", g("hidden_size"))
        self.patch_embed = PatchEmbed(
            g("patch_size", 14), g("temporal_patch_size", 2),
            g("in_channels", g("in_chans", 3)), embed_dim, dtype=dtype)
        num_heads = g("num_heads", 16)
        self.head_dim = embed_dim // num_heads
        self.rotary = VisionRotaryEmbedding(self.head_dim // 2)
        self.blocks = nn.ModuleList([
            VisionBlock(embed_dim, num_heads, g("mlp_ratio", 4.0),
                        dtype=dtype)
            for _ in range(g("depth", 32))])
        self.merger = PatchMerger(g("hidden_size", embed_dim) if
                                  g("embed_dim") else g("out_hidden_size",
                                                        embed_dim),
                                  embed_dim, self.spatial_merge_size,
                                  dtype=dtype)

    def forward(self, pixel_values: torch.Tensor,
                grid_thw: List[Tuple[int, int, int]]) -> torch.Tensor:
        x = self.patch_embed(pixel_values)
        rotary = rot_pos_emb(grid_thw, self.head_dim,
                             self.spatial_merge_size, self.rotary)
        lens = [t * h * w for t, h, w in grid_thw]
        cu = [0]
        for n in lens:
            cu.append(cu[-1] + n)
        for blk in self.blocks:
            x = blk(x, cu, rotary)
        return self.merger(x)
