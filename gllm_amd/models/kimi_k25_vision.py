"""Kimi-K2.5 MoonViT3d vision tower + projector.

Parity target: reference models/kimi_k25_vision.py — Conv2d per-patch
embed, learnable 2D-interpolated spatial pos-embed + fixed sincos
temporal embed, complex 2D rotary (x/y axes alternating frequency
pairs), pre-norm blocks with fused wqkv and GELU-tanh MLP, final
LayerNorm, 2x2 spatial merge with TEMPORAL MEAN-POOL (sd2_tpool), and a
LayerNorm->MLP projector to the text hidden size. Replicated on every
rank (not TP-sharded): the tower is small next to the 1T-class LM and
sharding the complex-rope wqkv packing buys little.

Input: pixel_values [N_patches, C, ps, ps] (per-patch crops), grids
[(t, h, w)] in patch units. Output: [sum (h/kh)*(w/kw), text_hidden].
"""

from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F


def _sincos_1d(dim: int, n: int) -> torch.Tensor:
    pos = torch.arange(n, dtype=torch.float64)
    omega = torch.arange(dim // 2, dtype=torch.float64) / (dim / 2.0)
    omega = 1.0 / (10000 ** omega)
    out = torch.outer(pos, omega)
    return torch.cat([out.sin(), out.cos()], dim=1).float()  # [n, dim]


class Learnable2DInterpPosEmb(nn.Module):
    def __init__(self, height, width, num_frames, dim):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(height, width, dim))
        self.register_buffer("time_weight",
                             _sincos_1d(dim, num_frames).unsqueeze(1),
                             persistent=False)

    def forward(self, x, grids):
        outs = []
        for t, h, w in grids:
            if (h, w) == tuple(self.weight.shape[:-1]):
                e2d = self.weight.flatten(end_dim=1)
            else:
                e2d = F.interpolate(
                    self.weight.permute(2, 0, 1).unsqueeze(0).float(),
                    size=(h, w), mode="bicubic").squeeze(0) \
                    .permute(1, 2, 0).flatten(end_dim=1) \
                    .to(self.weight.dtype)
            if t == 1:
                e3d = e2d
            else:
                tw = self.time_weight[:t].to(e2d.device, e2d.dtype)
                e3d = e2d.unsqueeze(0).repeat(t, 1, 1) + tw
            outs.append(e3d.reshape(-1, e3d.shape[-1]))
        return x + torch.cat(outs).to(x.dtype)


class KimiPatchEmbed(nn.Module):
    def __init__(self, out_dim, in_dim, patch_size, pos_h, pos_w, pos_t,
                 dtype=None):
        super().__init__()
        self.proj = nn.Conv2d(in_dim, out_dim, kernel_size=patch_size,
                              stride=patch_size, dtype=dtype)
        self.pos_emb = Learnable2DInterpPosEmb(pos_h, pos_w, pos_t,
                                               out_dim)

    def forward(self, x, grids):
        x = self.proj(x.to(self.proj.weight.dtype)).view(x.size(0), -1)
        return self.pos_emb(x, grids)


def rope2d_freqs_cis(head_dim: int, grids, max_h=512, max_w=512,
                     theta=10000.0) -> torch.Tensor:
    """[sum t*h*w, head_dim//2] complex: pair 2k rotates by x (width),
    pair 2k+1 by y (height)."""
    freqs = 1.0 / (theta ** (torch.arange(0, head_dim, 4)[:head_dim // 4]
                             .float() / head_dim))
    outs = []
    for t, h, w in grids:
        y = torch.arange(h).repeat_interleave(w).float()
        x = torch.arange(w).repeat(h).float()
        xf = torch.outer(x, freqs)
        yf = torch.outer(y, freqs)
        cis = torch.cat([torch.polar(torch.ones_like(xf), xf).unsqueeze(-1),
                         torch.polar(torch.ones_like(yf), yf).unsqueeze(-1)],
                        dim=-1).reshape(h * w, -1)
        outs.append(cis.repeat(t, 1))
    return torch.cat(outs, dim=0)


def apply_rope_complex(q, k, freqs_cis):
    fc = freqs_cis.unsqueeze(-2)  # [..., 1, d/2]
    q_ = torch.view_as_complex(q.float().reshape(*q.shape[:-1], -1, 2))
    k_ = torch.view_as_complex(k.float().reshape(*k.shape[:-1], -1, 2))
    q_out = torch.view_as_real(q_ * fc).flatten(-2)
    k_out = torch.view_as_real(k_ * fc).flatten(-2)
    return q_out.type_as(q), k_out.type_as(k)


class KimiVisionBlock(nn.Module):
    def __init__(self, num_heads, dim, mlp_dim, dtype=None):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.norm0 = nn.LayerNorm(dim, dtype=dtype)
        self.norm1 = nn.LayerNorm(dim, dtype=dtype)
        self.wqkv = nn.Linear(dim, dim * 3, bias=True, dtype=dtype)
        self.wo = nn.Linear(dim, dim, bias=True, dtype=dtype)
        self.mlp = nn.ModuleDict(dict(
            fc0=nn.Linear(dim, mlp_dim, bias=True, dtype=dtype),
            fc1=nn.Linear(mlp_dim, dim, bias=True, dtype=dtype)))

    def _attn(self, x, cu, freqs_cis):
        L, dim = x.shape
        qkv = self.wqkv(x).view(L, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.unbind(1)
        q, k = apply_rope_complex(q, k, freqs_cis)
        out = torch.empty(L, self.num_heads, self.head_dim, dtype=x.dtype)
        for i in range(len(cu) - 1):
            s, e = int(cu[i]), int(cu[i + 1])
            o = F.scaled_dot_product_attention(
                q[s:e].transpose(0, 1).float(),
                k[s:e].transpose(0, 1).float(),
                v[s:e].transpose(0, 1).float())
            out[s:e] = o.transpose(0, 1).to(x.dtype)
        return self.wo(out.reshape(L, dim))

    def forward(self, x, cu, freqs_cis):
        x = x + self._attn(self.norm0(x), cu, freqs_cis)
        h = self.mlp["fc0"](self.norm1(x))
        h = F.gelu(h, approximate="tanh")
        return x + self.mlp["fc1"](h)


class KimiVisionTower(nn.Module):
    def __init__(self, vcfg: dict, dtype=None):
        super().__init__()
        c = vcfg
        self.hidden_size = c["vt_hidden_size"]
        self.num_heads = c["vt_num_attention_heads"]
        self.head_dim = self.hidden_size // self.num_heads
        self.merge_kernel_size = tuple(c["merge_kernel_size"])
        self.patch_embed = KimiPatchEmbed(
            self.hidden_size, c.get("in_channels", 3), c["patch_size"],
            c.get("init_pos_emb_height", 64),
            c.get("init_pos_emb_width", 64),
            c.get("init_pos_emb_time", 4), dtype=dtype)
        self.encoder = nn.ModuleDict(dict(
            blocks=nn.ModuleList([
                KimiVisionBlock(self.num_heads, self.hidden_size,
                                c["vt_intermediate_size"], dtype=dtype)
                for _ in range(c["vt_num_hidden_layers"])]),
            final_layernorm=nn.LayerNorm(self.hidden_size, dtype=dtype)))

    def forward(self, pixel_values, grids) -> List[torch.Tensor]:
        x = self.patch_embed(pixel_values, grids)
        freqs = rope2d_freqs_cis(self.head_dim, grids)
        cu = [0]
        for t, h, w in grids:
            cu.append(cu[-1] + t * h * w)
        for blk in self.encoder["blocks"]:
            x = blk(x, cu, freqs)
        x = self.encoder["final_layernorm"](x)
        # sd2_tpool merge: 2x2 spatial groups, mean over t
        kh, kw = self.merge_kernel_size
        outs = []
        off = 0
        for t, h, w in grids:
            seq = x[off:off + t * h * w]
            nh, nw = h // kh, w // kw
            r = seq.view(t, nh, kh, nw, kw, self.hidden_size)
            r = r.permute(0, 1, 3, 2, 4, 5).contiguous().mean(dim=0)
            outs.append(r.view(nh * nw, kh * kw, -1))
            off += t * h * w
        return outs


class KimiPatchMerger(nn.Module):
    """LayerNorm -> MLP projector to the text hidden size (checkpoint
    keys pre_norm + proj.0/proj.2)."""

    def __init__(self, vcfg: dict, dtype=None):
        super().__init__()
        mm = vcfg["mm_hidden_size"]
        kh, kw = vcfg["merge_kernel_size"]
        self.hidden_size = mm * kh * kw
        self.pre_norm = nn.LayerNorm(mm, eps=vcfg.get("projector_ln_eps",
                                                      1e-5), dtype=dtype)
        self.proj = nn.Sequential(
            nn.Linear(self.hidden_size, self.hidden_size, dtype=dtype),
            nn.GELU(),
            nn.Linear(self.hidden_size, vcfg["text_hidden_size"],
                      dtype=dtype))

    def forward(self, items: List[torch.Tensor]) -> List[torch.Tensor]:
        return [self.proj(self.pre_norm(it).view(it.shape[0], -1))
                for it in items]
