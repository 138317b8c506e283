"""Custom-op facade — the single switch point between backends.

Mirrors the role of the reference's ``_custom_ops.py`` ("single point
where we can swap backends", _custom_ops.py:1-10): every layer calls
through here. Dispatch rule:

* CUDA (= HIP/ROCm) tensors -> the in-tree gfx950 extension
  ``gllm_amd._kernels``. If the extension is missing on a GPU box this
  RAISES — no silent eager fallback (a GPU run must exercise the native
  kernels).
* CPU tensors -> the PyTorch reference implementations (ops.torch_ref),
  which double as the numerics oracle for the HIP kernels.
"""

from typing import Optional

import torch

from gllm_amd.ops import torch_ref

_K = None
_K_ERR: Optional[str] = None


def _load_kernels():
    global _K, _K_ERR
    if _K is not None or _K_ERR is not None:
        return _K
    try:
        from gllm_amd import _kernels  # built in-tree by setup.py
        _K = _kernels
    except ImportError as e:  # pragma: no cover
        _K_ERR = str(e)
    return _K


def _gpu_kernels():
    k = _load_kernels()
    if k is None:
        raise RuntimeError(
            "gllm_amd._kernels HIP extension not built but a CUDA tensor "
            "was passed — build with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_K_ERR}")
    return k


def has_kernels() -> bool:
    return _load_kernels() is not None


# --------------------------------------------------------------- norm
def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _gpu_kernels().rmsnorm(out, x, weight, eps)
        return out
    return torch_ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float):
    if x.is_cuda:
        _gpu_kernels().fused_add_rmsnorm(x, residual, weight, eps)
        return x, residual
    return torch_ref.fused_add_rmsnorm(x, residual, weight, eps)


# --------------------------------------------------------------- activation
def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        d = x.shape[-1] // 2
        out = torch.empty(x.shape[:-1] + (d,), dtype=x.dtype, device=x.device)
        _gpu_kernels().silu_and_mul(out, x)
        return out
    return torch_ref.silu_and_mul(x)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        d = x.shape[-1] // 2
        out = torch.empty(x.shape[:-1] + (d,), dtype=x.dtype, device=x.device)
        _gpu_kernels().gelu_and_mul(out, x)
        return out
    return torch_ref.gelu_and_mul(x)


# --------------------------------------------------------------- rope
def rotary_embedding(positions: torch.Tensor, q: torch.Tensor,
                     k: torch.Tensor, head_dim: int,
                     cos_sin_cache: torch.Tensor, is_neox: bool = True):
    if q.is_cuda:
        _gpu_kernels().rotary_embedding(positions, q, k, head_dim,
                                        cos_sin_cache, is_neox)
        return q, k
    return torch_ref.rotary_embedding(positions, q, k, head_dim,
                                      cos_sin_cache, is_neox)


# --------------------------------------------------------------- kv cache
def reshape_and_cache(k: torch.Tensor, v: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      slot_mapping: torch.Tensor):
    if k.is_cuda:
        _gpu_kernels().reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    torch_ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


# --------------------------------------------------------------- attention
def paged_attention(q: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, block_table: torch.Tensor,
                    seq_lens: torch.Tensor, query_start_loc: torch.Tensor,
                    scale: float, max_query_len: int = 1,
                    out: Optional[torch.Tensor] = None,
                    sliding_window: int = 0) -> torch.Tensor:
    if q.is_cuda:
        kern = _gpu_kernels()
        if out is None:
            out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        if max_query_len == 1:
            kern.paged_attention_decode(
                out, q, k_cache, v_cache, block_table, seq_lens, scale,
                sliding_window)
        else:
            kern.paged_attention_prefill(
                out, q, k_cache, v_cache, block_table, seq_lens,
                query_start_loc.int(), max_query_len, scale, sliding_window)
        return out
    return torch_ref.paged_attention(q, k_cache, v_cache, block_table,
                                     seq_lens, query_start_loc, scale,
                                     out=out, sliding_window=sliding_window)


# --------------------------------------------------------------- moe
def topk_softmax(gating: torch.Tensor, topk: int, renormalize: bool = True):
    return torch_ref.topk_softmax(gating, topk, renormalize)


def grouped_topk(scores, topk, n_group, topk_group, renormalize=True,
                 scoring="softmax", e_bias=None):
    # device-resident (a CPU round-trip here would sync every MoE
    # layer of every DeepSeek step)
    return torch_ref.grouped_topk(scores, topk, n_group, topk_group,
                                  renormalize, scoring, e_bias)


def fused_moe(x: torch.Tensor, w13: torch.Tensor, w2: torch.Tensor,
              topk_weights: torch.Tensor, topk_ids: torch.Tensor,
              expert_start: int = 0,
              num_global_experts: Optional[int] = None) -> torch.Tensor:
    """Grouped-GEMM fused MoE on the gfx950 kernels (GPU, bf16).

    Reference fused_experts_impl (fused_moe_triton/fused_moe.py:779):
    align -> GEMM1 -> silu_and_mul -> GEMM2 (weighted scatter) -> sum.
    Fully device-resident and hipGraph-safe (no host syncs; grids are
    sized by upper bound and gated on the device-side block count).
    Returns the PARTIAL output for this rank's expert shard (callers
    run their TP/EP all-reduce on top).
    """
    T, K = x.shape
    E_local, two_i, _ = w13.shape
    inter = two_i // 2
    topk = topk_ids.shape[1]
    n_pairs = T * topk
    # tile heuristic: match block_m to the expected rows per expert so
    # decode doesn't re-stream expert weights per m-block; large dense
    # batches use the wide-BN prefill variant (gemm code 164 = BM 64 x
    # BN 256 — quarters the A-tile re-reads that dominate prefill)
    rpe = n_pairs / max(1, E_local)
    if n_pairs >= 4096:
        align_m, gemm_m = 64, 164
    elif rpe < 12:
        align_m = gemm_m = 16
    elif rpe < 40:
        align_m = gemm_m = 32
    else:
        align_m = gemm_m = 64
    cap = n_pairs + E_local * (align_m - 1) + 1
    max_blocks = (n_pairs + align_m - 1) // align_m + E_local
    dev = x.device
    k = _gpu_kernels()
    ids32 = topk_ids.int().contiguous()
    sorted_ids = torch.empty(cap, dtype=torch.int32, device=dev)
    expert_blocks = torch.empty(max_blocks, dtype=torch.int32, device=dev)
    n_post = torch.empty(1, dtype=torch.int32, device=dev)
    k.moe_align(ids32, E_local, expert_start, align_m, sorted_ids,
                expert_blocks, n_post)
    rows_pad = cap - 1 + align_m  # >= n_post_pad upper bound
    inter1 = torch.empty(rows_pad, two_i, dtype=x.dtype, device=dev)
    k.moe_gemm(inter1, x, w13, sorted_ids, expert_blocks, n_post, None,
               n_pairs, topk, gemm_m, False)
    act = silu_and_mul(inter1)
    # zero-filled: pairs routed to non-local experts (EP shards) or to
    # the DP padding id -1 are dropped by the align kernel and must
    # contribute zeros
    pair_out = x.new_zeros(n_pairs, K)
    k.moe_gemm(pair_out, act, w2, sorted_ids, expert_blocks, n_post,
               topk_weights.float().contiguous(), n_pairs, topk, gemm_m,
               True)
    out = torch.empty(T, K, dtype=x.dtype, device=dev)
    k.moe_sum(out, pair_out, topk)
    return out


# --------------------------------------------------------------- fp8
def per_token_group_quant_fp8(x: torch.Tensor, group: int = 128,
                              ue8m0: bool = False,
                              transposed: bool = False):
    """[T, K] bf16 -> (e4m3 [T, K], fp32 scales [T, K/group]).
    GPU kernel; CPU callers use the torch math in quantization/fp8.py.
    ``transposed=True`` additionally returns scales_t [K/group, T]
    (column-major) — the skinny GEMM stages a k-group's scales as ONE
    contiguous run instead of a strided 64-line gather."""
    assert x.is_cuda and group == 128
    T, K = x.shape
    q = torch.empty(T, K, dtype=torch.float8_e4m3fn, device=x.device)
    scales = torch.empty(T, K // group, dtype=torch.float32,
                         device=x.device)
    st = torch.empty(K // group, T, dtype=torch.float32,
                     device=x.device) if transposed else None
    _gpu_kernels().per_token_group_quant_fp8(
        x.contiguous(), q.view(torch.uint8), scales, st, ue8m0)
    if transposed:
        return q, scales, st
    return q, scales


_FP8_WS: dict = {}


def fp8_linear(x: torch.Tensor, w_q: torch.Tensor, w_scale: torch.Tensor,
               bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = x @ dequant(w_q).T via the fp8 weight-streaming kernel
    (M <= 256): per-token-group activation quant + block-scale fp8
    MFMA. Weights stream at 1 B/elem — ~2x the bf16 decode rate."""
    M, K = x.shape
    N = w_q.shape[0]
    aq, _, ast = per_token_group_quant_fp8(x.contiguous(),
                                           transposed=True)
    splitk = _quant_splitk(N, K, 128)
    need = splitk * M * N
    key = x.device.index or 0
    ws = _FP8_WS.get(key)
    if ws is None or ws.numel() < need:
        if torch.cuda.is_current_stream_capturing():
            raise RuntimeError("fp8 workspace grown during graph capture")
        ws = torch.empty(need, dtype=torch.float32, device=x.device)
        _FP8_WS[key] = ws
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    _gpu_kernels().fp8_skinny_gemm(
        out, aq.view(torch.uint8), ast, w_q.view(torch.uint8),
        w_scale.contiguous(),
        bias.float() if bias is not None else None, ws, splitk)
    return out


_IDENT: dict = {}


def _identity_tables(device, m_pad: int, block: int):
    """Routing tables that make the grouped MoE GEMM a plain dense
    GEMM: sorted_ids = arange (row i of the batch is 'pair' i, topk=1),
    every block belongs to expert 0. Cached per (device, m_pad) — the
    graph warmup pass populates the cache so capture never allocates."""
    key = (device.index or 0, m_pad, block)
    ent = _IDENT.get(key)
    if ent is None:
        if torch.cuda.is_current_stream_capturing():
            raise RuntimeError("identity tables built during graph capture")
        ids = torch.arange(m_pad, dtype=torch.int32, device=device)
        blocks = torch.zeros(m_pad // block, dtype=torch.int32,
                             device=device)
        n_post = torch.full((1,), m_pad, dtype=torch.int32, device=device)
        ent = (ids, blocks, n_post)
        _IDENT[key] = ent
    return ent


def fp8_prefill_linear(x: torch.Tensor, w_q: torch.Tensor,
                       w_scale: torch.Tensor,
                       bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Dense fp8 GEMM for M > 256: the grouped MoE fp8 kernel (wide
    BM64xBN256 prefill tile) with identity routing — weights stay e4m3,
    no per-call dequant materialization."""
    M, K = x.shape
    N = w_q.shape[0]
    aq, as_ = per_token_group_quant_fp8(x.contiguous())
    m_pad = (M + 63) // 64 * 64
    ids, blocks, n_post = _identity_tables(x.device, m_pad, 64)
    out_pad = torch.empty(m_pad, N, dtype=x.dtype, device=x.device)
    _gpu_kernels().moe_gemm_fp8(
        out_pad, aq.view(torch.uint8), as_,
        w_q.view(torch.uint8).view(1, N, K),
        w_scale.contiguous().view(1, w_scale.shape[0], w_scale.shape[1]),
        ids, blocks, n_post, None, M, 1, 164, False)
    out = out_pad.narrow(0, 0, M)
    if bias is not None:
        out = out.add_(bias)
    return out


def int4_prefill_linear(x: torch.Tensor, wq4: torch.Tensor,
                        sb: torch.Tensor,
                        bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Dense int4 (w4a16) GEMM for M > 256 via the grouped MoE int4
    kernel with identity routing — packed nibbles stream directly."""
    M, K2 = x.shape[0], wq4.shape[1]
    N = wq4.shape[0]
    m_pad = (M + 63) // 64 * 64
    ids, blocks, n_post = _identity_tables(x.device, m_pad, 64)
    out_pad = torch.empty(m_pad, N, dtype=x.dtype, device=x.device)
    _gpu_kernels().moe_gemm_int4(
        out_pad, x.contiguous(), wq4.view(1, N, K2),
        sb.view(1, N, -1, 2).contiguous() if sb.dim() == 3 else sb,
        ids, blocks, n_post, None, M, 1, 64, False)
    out = out_pad.narrow(0, 0, M)
    if bias is not None:
        out = out.add_(bias)
    return out


def fused_moe_fp8(x: torch.Tensor, w13: torch.Tensor, w13_scale,
                  w2: torch.Tensor, w2_scale,
                  topk_weights: torch.Tensor, topk_ids: torch.Tensor,
                  expert_start: int = 0) -> torch.Tensor:
    """fp8 grouped-GEMM MoE: same align/sum pipeline as fused_moe with
    block-scale fp8 GEMMs; expert weights stay e4m3-resident."""
    T, K = x.shape
    E_local, two_i, _ = w13.shape
    inter = two_i // 2
    topk = topk_ids.shape[1]
    n_pairs = T * topk
    # same tile heuristic as the bf16 path (164 = BM64 x BN256 prefill)
    rpe = n_pairs / max(1, E_local)
    if n_pairs >= 4096:
        align_m, gemm_m = 64, 164
    elif rpe < 12:
        align_m = gemm_m = 16
    elif rpe < 40:
        align_m = gemm_m = 32
    else:
        align_m = gemm_m = 64
    cap = n_pairs + E_local * (align_m - 1) + 1
    max_blocks = (n_pairs + align_m - 1) // align_m + E_local
    dev = x.device
    k = _gpu_kernels()
    ids32 = topk_ids.int().contiguous()
    sorted_ids = torch.empty(cap, dtype=torch.int32, device=dev)
    expert_blocks = torch.empty(max_blocks, dtype=torch.int32, device=dev)
    n_post = torch.empty(1, dtype=torch.int32, device=dev)
    k.moe_align(ids32, E_local, expert_start, align_m, sorted_ids,
                expert_blocks, n_post)
    rows_pad = cap - 1 + align_m
    aq, as_ = per_token_group_quant_fp8(x.contiguous())
    inter1 = torch.empty(rows_pad, two_i, dtype=x.dtype, device=dev)
    k.moe_gemm_fp8(inter1, aq.view(torch.uint8), as_,
                   w13.view(torch.uint8), w13_scale.contiguous(),
                   sorted_ids, expert_blocks, n_post, None, n_pairs, topk,
                   gemm_m, False)
    act = silu_and_mul(inter1)
    actq, act_s = per_token_group_quant_fp8(act)
    pair_out = x.new_zeros(n_pairs, K)
    k.moe_gemm_fp8(pair_out, actq.view(torch.uint8), act_s,
                   w2.view(torch.uint8), w2_scale.contiguous(),
                   sorted_ids, expert_blocks, n_post,
                   topk_weights.float().contiguous(), n_pairs, topk,
                   gemm_m, True)
    out = torch.empty(T, K, dtype=x.dtype, device=dev)
    k.moe_sum(out, pair_out, topk)
    return out


# --------------------------------------------------------------- gdn
def gdn_conv_update(x: torch.Tensor, weight: torch.Tensor,
                    conv_state: torch.Tensor,
                    slots: torch.Tensor) -> torch.Tensor:
    """Batched causal-conv1d decode step: x [B, C] one token per seq,
    conv_state pool rows indexed by ``slots`` roll in place."""
    out = torch.empty_like(x)
    _gpu_kernels().gdn_conv_update(out, x, weight, conv_state, slots)
    return out


def gdn_decode(qn: torch.Tensor, kn: torch.Tensor, v: torch.Tensor,
               g: torch.Tensor, beta: torch.Tensor, state: torch.Tensor,
               slots: torch.Tensor) -> torch.Tensor:
    """Batched fused recurrent gated-delta-rule step (one token per
    seq). qn/kn [B, Hv, 128] fp32 (normalized, q pre-scaled), v [B, Hv,
    Dv] fp32; state pool [slots, Hv, Dv, 128] fp32 updated in place."""
    B, Hv, Dv = v.shape
    o = torch.empty(B, Hv, Dv, dtype=torch.bfloat16, device=v.device)
    _gpu_kernels().gdn_decode(o, qn.contiguous(), kn.contiguous(),
                              v.contiguous(), g.contiguous(),
                              beta.contiguous(), state, slots)
    return o


def gdn_chunk_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      g: torch.Tensor, beta: torch.Tensor,
                      states: torch.Tensor, scale: float) -> torch.Tensor:
    """Fused WY chunk-parallel gated delta rule over a padded batch
    (one launch per layer; replaces ~25 torch launches per 256-token
    chunk). q/k [B, T, Hk, 128], v [B, T, Hv, 128], g/beta [B, T, Hv]
    fp32, states [B, Hv, 128, 128] fp32 updated IN PLACE. Falls back
    to the torch oracle off-GPU or for non-128 head dims."""
    B, T, Hk, Dk = q.shape
    Hv, Dv = v.shape[2], v.shape[3]
    if not (q.is_cuda and has_kernels() and Dk == 128 and Dv == 128
            and Hv % Hk == 0):
        from gllm_amd.ops import gdn_ref
        return gdn_ref.gated_delta_rule_chunked_batched(
            q, k, v, g, beta, scale, states)
    Tp = (T + 63) // 64 * 64
    if Tp != T:  # zero padding rows carry g = 0, beta = 0 -> inert
        pad = (0, 0, 0, 0, 0, Tp - T)
        q = torch.nn.functional.pad(q, pad)
        k = torch.nn.functional.pad(k, pad)
        v = torch.nn.functional.pad(v, pad)
        g = torch.nn.functional.pad(g, (0, 0, 0, Tp - T))
        beta = torch.nn.functional.pad(beta, (0, 0, 0, Tp - T))
    o = torch.empty(B, Tp, Hv, Dv, dtype=torch.bfloat16, device=q.device)
    _gpu_kernels().gdn_chunk_prefill(
        o, q.to(torch.bfloat16).contiguous(),
        k.to(torch.bfloat16).contiguous(),
        v.to(torch.bfloat16).contiguous(),
        g.float().contiguous(), beta.float().contiguous(),
        states, float(scale))
    return o[:, :T]


def rmsnorm_gated(x: torch.Tensor, z: torch.Tensor, weight: torch.Tensor,
                  eps: float) -> torch.Tensor:
    """out = rmsnorm(x) * w * silu(z) (fused, GPU)."""
    if not x.is_cuda:
        from gllm_amd.ops import gdn_ref
        return gdn_ref.rmsnorm_gated(x, z, weight, eps)
    out = torch.empty_like(x)
    _gpu_kernels().rmsnorm_gated(out, x.contiguous(), z.contiguous(),
                                 weight, eps)
    return out


def int4_linear(x: torch.Tensor, wq4: torch.Tensor, sbt: torch.Tensor,
                group: int,
                bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = x @ dequant(wq4).T via the fused-dequant int4 skinny GEMM
    (M <= 256): packed nibbles stream at 0.5 B/elem. ``sbt`` is the
    TRANSPOSED (scale, bias) bank [K/group, 2, N] (lane-contiguous
    staging; repack_canonical builds it)."""
    M, K = x.shape
    N = wq4.shape[0]
    splitk = _quant_splitk(N, K, 256)
    need = splitk * M * N
    key = x.device.index or 0
    ws = _FP8_WS.get(key)  # shares the fp32 partial workspace pool
    if ws is None or ws.numel() < need:
        if torch.cuda.is_current_stream_capturing():
            raise RuntimeError("int4 workspace grown during graph capture")
        ws = torch.empty(need, dtype=torch.float32, device=x.device)
        _FP8_WS[key] = ws
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    _gpu_kernels().int4_skinny_gemm(
        out, x.contiguous(), wq4, sbt,
        bias.float() if bias is not None else None, ws, group)
    return out


def fused_moe_int4(x: torch.Tensor, w13_c, w13_sb, w2_c, w2_sb,
                   topk_weights: torch.Tensor, topk_ids: torch.Tensor,
                   expert_start: int = 0) -> torch.Tensor:
    """int4 (w4a16) grouped-GEMM MoE: packed-nibble expert banks with
    in-fragment dequant — no bf16 materialization at all."""
    T, K = x.shape
    E_local, two_i, _ = w13_c.shape
    topk = topk_ids.shape[1]
    n_pairs = T * topk
    rpe = n_pairs / max(1, E_local)
    block_m = 16 if rpe < 12 else (32 if rpe < 40 else 64)
    cap = n_pairs + E_local * (block_m - 1) + 1
    max_blocks = (n_pairs + block_m - 1) // block_m + E_local
    dev = x.device
    k = _gpu_kernels()
    ids32 = topk_ids.int().contiguous()
    sorted_ids = torch.empty(cap, dtype=torch.int32, device=dev)
    expert_blocks = torch.empty(max_blocks, dtype=torch.int32, device=dev)
    n_post = torch.empty(1, dtype=torch.int32, device=dev)
    k.moe_align(ids32, E_local, expert_start, block_m, sorted_ids,
                expert_blocks, n_post)
    rows_pad = cap - 1 + block_m
    inter1 = torch.empty(rows_pad, two_i, dtype=x.dtype, device=dev)
    k.moe_gemm_int4(inter1, x.contiguous(), w13_c, w13_sb, sorted_ids,
                    expert_blocks, n_post, None, n_pairs, topk, block_m,
                    False)
    act = silu_and_mul(inter1)
    pair_out = x.new_zeros(n_pairs, K)
    k.moe_gemm_int4(pair_out, act, w2_c, w2_sb, sorted_ids,
                    expert_blocks, n_post,
                    topk_weights.float().contiguous(), n_pairs, topk,
                    block_m, True)
    out = torch.empty(T, K, dtype=x.dtype, device=dev)
    k.moe_sum(out, pair_out, topk)
    return out


# --------------------------------------------------------------- sampling
def topk_topp_filter(probs: torch.Tensor, top_ks: torch.Tensor,
                     top_ps: torch.Tensor,
                     min_ps: Optional[torch.Tensor] = None):
    """In-place fused top-k/top-p/min-p filter + renormalize on [B, V]
    fp32 probs (GPU kernel; sorting-free 3-level radix select). CPU
    callers use the torch sort composite in layers/sampler.py."""
    assert probs.is_cuda
    _gpu_kernels().topk_topp_filter(
        probs, top_ks.int(), top_ps.float(),
        min_ps.float() if min_ps is not None else None)
    return probs


# --------------------------------------------------------------- mla
def apply_penalty_pool(logits, mask_pool, slots, penalties):
    """In-place repetition penalty against the persistent uint8 mask
    pool (GPU kernel; CPU callers use the inline torch math in
    layers/sampler.py)."""
    assert logits.is_cuda
    _gpu_kernels().apply_repetition_penalty(logits, mask_pool, slots,
                                            penalties)
    return logits


def cache_latent(k, k_cache, slot_mapping):
    """Scatter the absorbed-MLA latent row [T, 1, lora+rope] into the
    paged latent cache (the v 'cache' is a view of its first lora dims,
    so one write covers both). Reference concat_and_cache_mla
    (cache_kernels.py:161)."""
    if k.is_cuda:
        if k_cache.shape[-1] == 576:
            _gpu_kernels().cache_latent(
                k.reshape(k.shape[0], -1), k_cache, slot_mapping)
        else:  # non-standard latent width: plain K/V scatter, twice
            _gpu_kernels().reshape_and_cache(k, k, k_cache, k_cache,
                                             slot_mapping)
        return
    page_size = k_cache.shape[1]
    pages = torch.div(slot_mapping, page_size, rounding_mode="floor")
    offs = slot_mapping % page_size
    k_cache[pages, offs] = k.to(k_cache.dtype)


_MLA_GPU_WARNED = False


def mla_paged_attention(q, k_cache, v_cache, block_table, seq_lens,
                        query_start_loc, scale, topk_positions=None,
                        seq_lens_cpu=None, query_start_loc_cpu=None):
    """Varlen causal attention with asymmetric head dims (MLA).

    GPU + absorbed layout (Hkv=1, 576-dim latent) -> the gfx950 MQA
    MFMA kernel (one kernel serves decode w/ split-KV, chunked prefill
    and mixed batches — reference attention.py:366-446,653-925 roles).
    DSA sparse selection (topk_positions) and the non-absorbed debug
    layout stay on the torch path."""
    global _MLA_GPU_WARNED
    if (q.is_cuda and topk_positions is None and k_cache.shape[2] == 1
            and q.shape[-1] == 576 and v_cache.shape[-1] == 512
            and q.dtype == torch.bfloat16):
        qsl_h = query_start_loc_cpu or query_start_loc.tolist()
        lens_h = seq_lens_cpu or seq_lens.tolist()
        max_q = max((qsl_h[i + 1] - qsl_h[i] for i in range(len(lens_h))),
                    default=1)
        max_s = max(lens_h, default=1)
        T, H = q.shape[0], q.shape[1]
        out = torch.empty(T, H, 512, dtype=q.dtype, device=q.device)
        qsl_i = query_start_loc.int() \
            if query_start_loc.dtype != torch.int32 else query_start_loc
        _gpu_kernels().mla_paged_attention(
            out, q.contiguous(), k_cache, block_table, seq_lens, qsl_i,
            max_q, scale, max_s)
        return out
    if q.is_cuda and not _MLA_GPU_WARNED:
        from gllm_amd.logger import logger
        logger.warning("MLA attention falls back to the torch path on "
                       "GPU (sparse/non-absorbed layout)")
        _MLA_GPU_WARNED = True
    return torch_ref.mla_paged_attention(
        q, k_cache, v_cache, block_table, seq_lens, query_start_loc,
        scale, topk_positions=topk_positions, seq_lens_cpu=seq_lens_cpu,
        query_start_loc_cpu=query_start_loc_cpu)


# --------------------------------------------------------------- gemm
_SKINNY_WS: dict = {}
SKINNY_MAX_M = 256


def _skinny_splitk(M: int, N: int, K: int) -> int:
    # v2 kernel grid is (N/64, splitk) — M-blocks live INSIDE one
    # workgroup, so the fill target counts only N tiles
    n_wg = -(-N // 64)
    s = 1
    while s < 16 and n_wg * s < 512 and (K // (s * 2)) >= 64:
        s *= 2
    k_slice = ((-(-K // s)) + 63) // 64 * 64
    return -(-K // k_slice)


def _quant_splitk(N: int, K: int, bk: int) -> int:
    """splitk for the quant skinny GEMMs (fp8 bk=128, int4 bk=256):
    deepest split that still leaves >= 8 ring stages per block, with
    the grid capped near chip size (measured sweep, profiles/)."""
    n_wg = -(-N // 64)
    s = 1
    while (s < 16 and n_wg * (s * 2) <= 1024
           and (K // bk) // (s * 2) >= 8):
        s *= 2
    k_slice = ((-(-K // s)) + bk - 1) // bk * bk
    return -(-K // k_slice)


def skinny_gemm(x: torch.Tensor, w: torch.Tensor,
                bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = x @ w.T via the gfx950 weight-streaming kernel (M<=256-ish)."""
    import os
    M, K = x.shape
    N = w.shape[0]
    splitk = int(os.environ.get("SK_SPLITK", "0")) or \
        _skinny_splitk(M, N, K)
    need = splitk * M * N
    key = x.device.index or 0
    ws = _SKINNY_WS.get(key)
    if ws is None or ws.numel() < need:
        if torch.cuda.is_current_stream_capturing():
            # growing the cached workspace mid-capture would free the
            # buffer an earlier-captured graph writes (the r1 replay
            # fault class); warmup must size it first
            raise RuntimeError(
                "skinny_gemm workspace grown during graph capture — "
                "run an eager warmup at the largest bucket first")
        ws = torch.empty(need, dtype=torch.float32, device=x.device)
        _SKINNY_WS[key] = ws
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    _gpu_kernels().skinny_gemm(out, x, w, None, ws, splitk)
    if bias is not None:
        out += bias
    return out


def _use_skinny(M: int, N: int, K: int) -> bool:
    """Measured dispatch rule (scripts/gemm_sweep.py on MI355X r01):
    the skinny kernel beats hipBLASLt on K-deep reductions at any decode
    M (down_proj: 1.9 -> 4.1 TB/s) and on every N<=8k shape once M>64
    (library tiles collapse to ~1.1-1.6 TB/s there); the library keeps
    wide-N shapes (gate_up, lm_head) and small-M small-K projections."""
    import os
    if os.environ.get("GLLM_DISABLE_SKINNY"):
        return False
    max_m = int(os.environ.get("GLLM_SKINNY_MAX_M", "0")) or SKINNY_MAX_M
    if K % 64 != 0 or N < 1024 or M > max_m:
        return False
    # MB>=2 under capture: r1 faulted with dynamic LDS; the kernel now
    # uses static LDS and captures cleanly (scripts/graph_skinny_repro)
    if K >= 2 * N:
        return True
    return M > 64 and N <= 8192


def linear(x: torch.Tensor, w: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """GEMM dispatch: the skinny weight-streaming kernel where measured
    faster, hipBLASLt otherwise."""
    import torch.nn.functional as F
    if (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 2
            and x.is_contiguous()
            and _use_skinny(x.shape[0], w.shape[0], w.shape[1])):
        return skinny_gemm(x, w, bias)
    return F.linear(x, w, bias)
