"""Qwen2-VL vision tower (reference: models/qwen2_5_vl.py ViT side).

Faithful Qwen2-VL ViT structure so real checkpoints load unchanged:
Conv3d patch embed (temporal_patch_size x patch x patch), blocks of
[LayerNorm -> fused-qkv attention with 2D rotary -> LayerNorm ->
QuickGELU MLP], and the 2x2 PatchMerger. Full (per-image) attention via
varlen SDPA over cu_seqlens.

Wired into the full multimodal path (docs/multimodal.md): LM-side
embedding merge, [3, T] mrope through the batch builder, server-side
image processor, encoder disaggregation. The 2.5-VL windowed variant
(Qwen25VisionTransformer) and the Qwen3-VL deepstack tower
(models/qwen3_vl_vision.py) build on the same pieces.
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


# ---------------------------------------------------------------- 2.5-VL
class Vision25MLP(nn.Module):
    """SiLU-gated MLP (Qwen2.5-VL blocks; HF names gate/up/down_proj)."""

    def __init__(self, dim, hidden, dtype=None):
        super().__init__()
        self.gate_proj = nn.Linear(dim, hidden, bias=True, dtype=dtype)
        self.up_proj = nn.Linear(dim, hidden, bias=True, dtype=dtype)
        self.down_proj = nn.Linear(hidden, dim, bias=True, dtype=dtype)

    def forward(self, x):
        return self.down_proj(F.silu(self.gate_proj(x)) * self.up_proj(x))


class Vision25Block(nn.Module):
    def __init__(self, dim, num_heads, intermediate, dtype=None):
        super().__init__()
        from gllm_amd.layers.layernorm import RMSNorm
        self.norm1 = RMSNorm(dim, 1e-6)
        self.norm2 = RMSNorm(dim, 1e-6)
        self.attn = VisionAttention(dim, num_heads, dtype=dtype)
        self.mlp = Vision25MLP(dim, intermediate, dtype=dtype)

    def forward(self, x, cu_seqlens, rotary):
        x = x + self.attn(self.norm1(x), cu_seqlens, rotary)
        x = x + self.mlp(self.norm2(x))
        return x


class PatchMerger25(nn.Module):
    """RMSNorm ln_q + 2-layer MLP to out_hidden_size (HF names
    merger.ln_q, merger.mlp.{0,2})."""

    def __init__(self, d_model, context_dim, merge_size, dtype=None):
        super().__init__()
        from gllm_amd.layers.layernorm import RMSNorm
        self.hidden_size = context_dim * (merge_size ** 2)
        self.ln_q = RMSNorm(context_dim, 1e-6)
        self.mlp = nn.Sequential(
            nn.Linear(self.hidden_size, self.hidden_size, dtype=dtype),
            nn.GELU(),
            nn.Linear(self.hidden_size, d_model, dtype=dtype))

    def forward(self, x):
        return self.mlp(self.ln_q(x).view(-1, self.hidden_size))


def window_index_thw(t, h, w, merge_size, window_patches):
    """Qwen2.5-VL window partition (reference qwen2_5_vl.py:537-572):
    permutation of merge-unit indices grouping them into
    window_patches x window_patches windows, plus the per-window
    cumulative PATCH counts (x merge_unit)."""
    lh, lw = h // merge_size, w // merge_size
    idx = torch.arange(t * lh * lw).reshape(t, lh, lw)
    pad_h = (-lh) % window_patches
    pad_w = (-lw) % window_patches
    nh = (lh + pad_h) // window_patches
    nw = (lw + pad_w) // window_patches
    padded = F.pad(idx, (0, pad_w, 0, pad_h), value=-100)
    padded = padded.reshape(t, nh, window_patches, nw, window_patches)
    padded = padded.permute(0, 1, 3, 2, 4).reshape(
        t, nh * nw, window_patches, window_patches)
    seqlens = (padded != -100).sum([2, 3]).reshape(-1)
    flat = padded.reshape(-1)
    index = flat[flat != -100]
    cu = seqlens.cumsum(0) * (merge_size ** 2)
    cu = torch.unique_consecutive(cu.to(torch.int32))
    return index, cu


class Qwen25VisionTransformer(nn.Module):
    """visual.* of Qwen2.5-VL checkpoints: RMSNorm blocks, SiLU-gated
    MLP, and WINDOWED attention — all blocks attend within
    window_size x window_size pixel windows except
    ``fullatt_block_indexes``, which see the whole image. Rows are
    permuted into window order (in spatial-merge units) once, attended
    with per-window cu_seqlens, and un-permuted after the merger
    (reference qwen2_5_vl.py:440-686)."""

    def __init__(self, vcfg, dtype=None):
        super().__init__()
        g = lambda k, d=None: (vcfg.get(k, d) if isinstance(vcfg, dict)
                               else getattr(vcfg, k, d))
        self.spatial_merge_size = g("spatial_merge_size", 2)
        self.patch_size = g("patch_size", 14)
        self.window_size = g("window_size", 112)
        self.fullatt_block_indexes = list(g("fullatt_block_indexes",
                                            []) or [])
        dim = g("hidden_size")
        self.patch_embed = PatchEmbed(
            self.patch_size, g("temporal_patch_size", 2),
            g("in_channels", 3), dim, dtype=dtype)
        num_heads = g("num_heads", 16)
        self.head_dim = dim // num_heads
        self.rotary = VisionRotaryEmbedding(self.head_dim // 2)
        self.blocks = nn.ModuleList([
            Vision25Block(dim, num_heads, g("intermediate_size", dim * 4),
                          dtype=dtype)
            for _ in range(g("depth", 32))])
        self.merger = PatchMerger25(g("out_hidden_size", dim), dim,
                                    self.spatial_merge_size, dtype=dtype)

    def forward(self, pixel_values: torch.Tensor,
                grid_thw: List[Tuple[int, int, int]]) -> torch.Tensor:
        m = self.spatial_merge_size
        unit = m * m
        wp = self.window_size // m // self.patch_size
        x = self.patch_embed(pixel_values)
        rotary = rot_pos_emb(grid_thw, self.head_dim, m, self.rotary)

        # per-image window permutation over merge units
        win_index = []
        cu_window = [torch.zeros(1, dtype=torch.int32)]
        cu_full = [0]
        base = 0
        for t, h, w in grid_thw:
            idx, cu = window_index_thw(t, h, w, m, wp)
            win_index.append(idx + base)
            base += t * (h // m) * (w // m)
            cu_window.append(cu + int(cu_window[-1][-1]))
            cu_full.append(cu_full[-1] + t * h * w)
        win_index = torch.cat(win_index)
        cu_window = torch.unique_consecutive(torch.cat(cu_window)).tolist()

        L = x.shape[0]
        x = x.reshape(L // unit, unit, -1)[win_index].reshape(L, -1)
        rotary = rotary.reshape(L // unit, unit, -1)[win_index] \
            .reshape(L, -1)

        for i, blk in enumerate(self.blocks):
            cu = cu_full if i in self.fullatt_block_indexes else cu_window
            x = blk(x, cu, rotary)
        out = self.merger(x)
        reverse = torch.argsort(win_index)
        return out[reverse]
