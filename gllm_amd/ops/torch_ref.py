"""Pure-PyTorch reference implementations of every custom op.

These are (a) the CPU execution path (config 1: plumbing without a GPU)
and (b) the numerics oracle the HIP kernels are unit-tested against
(reference keeps `forward_native` impls for this purpose,
layers/layernorm.py:88-127). All math in fp32 for a stable oracle.
"""

import math
from typing import Optional

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * weight.float()).to(dtype)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float):
    """residual += x; x = rmsnorm(residual). In-place on both args."""
    residual.add_(x)
    x.copy_(rmsnorm(residual, weight, eps))
    return x, residual


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    return (torch.nn.functional.silu(x[..., :d].float())
            * x[..., d:].float()).to(x.dtype)



def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """gelu(x[:, :d]) * x[:, d:] (erf-based, reference gelu_and_mul)."""
    d = x.shape[-1] // 2
    a = x[..., :d].float()
    return (torch.nn.functional.gelu(a) *
            x[..., d:].float()).to(x.dtype)

def rotary_embedding(positions: torch.Tensor, q: torch.Tensor,
                     k: torch.Tensor, head_dim: int,
                     cos_sin_cache: torch.Tensor, is_neox: bool = True):
    """In-place RoPE on q [T, Hq*D] and k [T, Hk*D].

    cos_sin_cache: [max_pos, rot_dim] laid out [cos(rot/2) | sin(rot/2)].
    """
    rot_dim = cos_sin_cache.shape[-1]
    half = rot_dim // 2
    cs = cos_sin_cache[positions].float()       # [T, rot_dim]
    cos = cs[:, :half]                          # [T, half]
    sin = cs[:, half:]
    for t in (q, k):
        T = t.shape[0]
        x = t.unflatten(-1, (-1, head_dim))
        rot = x[..., :rot_dim].float()
        if is_neox:
            x1, x2 = rot[..., :half], rot[..., half:]
            c = cos.unsqueeze(1)
            s = sin.unsqueeze(1)
            o1 = x1 * c - x2 * s
            o2 = x2 * c + x1 * s
            out = torch.cat([o1, o2], dim=-1)
        else:
            x1, x2 = rot[..., 0::2], rot[..., 1::2]
            c = cos.unsqueeze(1)
            s = sin.unsqueeze(1)
            o1 = x1 * c - x2 * s
            o2 = x2 * c + x1 * s
            out = torch.stack([o1, o2], dim=-1).flatten(-2)
        x[..., :rot_dim].copy_(out.to(t.dtype))
    return q, k


def reshape_and_cache(k: torch.Tensor, v: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      slot_mapping: torch.Tensor):
    """Scatter per-token K/V [T, Hkv, D] into paged caches
    [num_pages, page_size, Hkv, D] at flat slot indices."""
    page_size = k_cache.shape[1]
    pages = torch.div(slot_mapping, page_size, rounding_mode="floor")
    offs = slot_mapping % page_size
    k_cache[pages, offs] = k.to(k_cache.dtype)
    v_cache[pages, offs] = v.to(v_cache.dtype)


def paged_attention(q: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, block_table: torch.Tensor,
                    seq_lens: torch.Tensor, query_start_loc: torch.Tensor,
                    scale: float, out: Optional[torch.Tensor] = None,
                    sliding_window: int = 0) -> torch.Tensor:
    """Varlen causal attention over the paged KV cache.

    q:            [T, Hq, D] — all new tokens of the batch, ragged by seq
    k/v_cache:    [num_pages, page_size, Hkv, D]
    block_table:  [B, max_pages] int — page ids per seq
    seq_lens:     [B] — TOTAL context length per seq (past + new)
    query_start_loc: [B+1] — ragged boundaries into q
    Causal offset: query token i of seq b attends to cache positions
    [0, seq_len - q_len + i].
    """
    T, Hq, D = q.shape
    Hkv = k_cache.shape[2]
    page_size = k_cache.shape[1]
    group = Hq // Hkv
    if out is None:
        out = torch.empty_like(q)
    B = seq_lens.shape[0]
    for b in range(B):
        qs, qe = int(query_start_loc[b]), int(query_start_loc[b + 1])
        q_len = qe - qs
        if q_len == 0:
            continue
        s_len = int(seq_lens[b])
        n_pages = -(-s_len // page_size)
        pages = block_table[b, :n_pages].long()
        k = k_cache[pages].reshape(-1, Hkv, D)[:s_len].float()  # [S, Hkv, D]
        v = v_cache[pages].reshape(-1, Hkv, D)[:s_len].float()
        qq = q[qs:qe].float()                                   # [L, Hq, D]
        # [Hq, L, S]
        scores = torch.einsum("lhd,shd->hls", qq,
                              k.repeat_interleave(group, dim=1)) * scale
        # causal mask with past offset
        past = s_len - q_len
        pos_q = torch.arange(q_len, device=q.device).unsqueeze(1) + past
        pos_k = torch.arange(s_len, device=q.device).unsqueeze(0)
        mask = pos_k <= pos_q                                   # [L, S]
        if sliding_window > 0:
            mask &= pos_k > (pos_q - sliding_window)
        scores.masked_fill_(~mask.unsqueeze(0), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hls,shd->lhd", p,
                         v.repeat_interleave(group, dim=1))
        out[qs:qe] = o.to(out.dtype)
    return out


def mla_paged_attention(q: torch.Tensor, k_cache: torch.Tensor,
                        v_cache: torch.Tensor, block_table: torch.Tensor,
                        seq_lens: torch.Tensor,
                        query_start_loc: torch.Tensor,
                        scale: float,
                        topk_positions: Optional[torch.Tensor] = None,
                        seq_lens_cpu: Optional[list] = None,
                        query_start_loc_cpu: Optional[list] = None
                        ) -> torch.Tensor:
    """Varlen causal attention with asymmetric head dims (MLA):
    q/k have Dk (nope+rope), v has Dv. Same paged layout as
    paged_attention; returns [T, H, Dv].

    ``topk_positions`` (DeepSeek Sparse Attention, reference
    deepseek_v32.py:235-792): [T, topk] int per-QUERY token POSITIONS
    within the query's own sequence (-1 padded). When given, each query
    row attends ONLY to its selected positions (intersected with the
    causal mask). For topk >= seq_len the selection covers every key,
    making sparse == dense — the DSA correctness oracle."""
    T, H, Dk = q.shape
    Dv = v_cache.shape[3]
    page_size = k_cache.shape[1]
    out = torch.empty(T, H, Dv, dtype=q.dtype, device=q.device)
    B = seq_lens.shape[0]
    # host geometry (when the caller has it) avoids 2B+1 device syncs
    qsl_h = query_start_loc_cpu or query_start_loc.tolist()
    lens_h = seq_lens_cpu or seq_lens.tolist()
    for b in range(B):
        qs, qe = qsl_h[b], qsl_h[b + 1]
        q_len = qe - qs
        if q_len == 0:
            continue
        s_len = lens_h[b]
        n_pages = -(-s_len // page_size)
        pages = block_table[b, :n_pages].long()
        Hkv = k_cache.shape[2]
        k = k_cache[pages].reshape(-1, Hkv, Dk)[:s_len].float()
        v = v_cache[pages].reshape(-1, Hkv, Dv)[:s_len].float()
        if Hkv == 1 and H > 1:
            # absorbed MLA: MQA over the shared latent row
            k = k.expand(s_len, H, Dk)
            v = v.expand(s_len, H, Dv)
        qq = q[qs:qe].float()
        dev = q.device
        scores = torch.einsum("lhd,shd->hls", qq, k) * scale
        past = s_len - q_len
        pos_q = torch.arange(q_len, device=dev).unsqueeze(1) + past
        pos_k = torch.arange(s_len, device=dev).unsqueeze(0)
        mask = pos_k <= pos_q                          # [q_len, s_len]
        if topk_positions is not None:
            sel = topk_positions[qs:qe].long().to(dev)  # [q_len, topk]
            allowed = torch.zeros(q_len, s_len, dtype=torch.bool,
                                  device=dev)
            valid = (sel >= 0) & (sel < s_len)
            rows = torch.arange(q_len,
                                device=dev).unsqueeze(1).expand_as(sel)
            allowed[rows[valid], sel[valid]] = True
            mask = mask & allowed
        scores.masked_fill_(~mask.unsqueeze(0), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hls,shd->lhd", p, v)
        out[qs:qe] = o.to(out.dtype)
    return out


def grouped_topk(scores: torch.Tensor, topk: int, n_group: int,
                 topk_group: int, renormalize: bool = True,
                 scoring: str = "softmax",
                 e_bias: Optional[torch.Tensor] = None):
    """DeepSeek grouped routing (reference: layers/moe/topk.py
    group_limited_greedy / noaux_tc). ``e_bias`` set => noaux_tc: sigmoid
    scores + bias for SELECTION, original sigmoid scores as weights."""
    T, E = scores.shape
    if scoring == "sigmoid":
        probs = scores.float().sigmoid()
    else:
        probs = torch.softmax(scores.float(), dim=-1)
    sel = probs + e_bias.float() if e_bias is not None else probs
    g = sel.view(T, n_group, E // n_group)
    if e_bias is not None:
        group_scores = g.topk(2, dim=-1)[0].sum(-1)     # noaux_tc
    else:
        group_scores = g.max(dim=-1).values
    grp_idx = group_scores.topk(topk_group, dim=-1)[1]  # [T, topk_group]
    mask = torch.zeros(T, n_group, dtype=torch.bool, device=scores.device)
    mask.scatter_(1, grp_idx, True)
    mask = mask.unsqueeze(-1).expand(T, n_group, E // n_group)
    sel = sel.masked_fill(~mask.reshape(T, E), float("-inf"))
    ids = sel.topk(topk, dim=-1)[1]
    weights = probs.gather(1, ids)
    if renormalize:
        weights = weights / weights.sum(-1, keepdim=True).clamp_min(1e-20)
    return weights, ids.to(torch.int32)


def topk_softmax(gating: torch.Tensor, topk: int, renormalize: bool = True):
    """MoE routing: softmax then top-k. Returns (weights [T,K], ids [T,K])."""
    probs = torch.softmax(gating.float(), dim=-1)
    weights, ids = torch.topk(probs, topk, dim=-1)
    if renormalize:
        weights = weights / weights.sum(-1, keepdim=True)
    return weights, ids.to(torch.int32)


def apply_repetition_penalty(logits: torch.Tensor, token_ids_per_row,
                             penalties: torch.Tensor) -> torch.Tensor:
    """logits [B, V]; token_ids_per_row: list of LongTensors; penalties [B]."""
    for i, toks in enumerate(token_ids_per_row):
        p = float(penalties[i])
        if p == 1.0 or toks.numel() == 0:
            continue
        row = logits[i]
        vals = row[toks]
        row[toks] = torch.where(vals > 0, vals / p, vals * p)
    return logits
