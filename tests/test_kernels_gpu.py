"""HIP kernel numerics vs the torch fp32 oracle (run on MI355X)."""

import math

import pytest
import torch

from gllm_amd.ops import torch_ref as R

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def kernels():
    from gllm_amd import ops
    assert ops.has_kernels(), "HIP extension must be built"
    return ops


def assert_close_bf16(out, ref, atol=2e-2, rtol=2e-2, frac=1e-3):
    """bf16 kernel vs fp32 oracle: allow bf16 rounding noise."""
    out = out.float().cpu()
    ref = ref.float().cpu()
    bad = (~torch.isclose(out, ref, atol=atol, rtol=rtol)).float().mean()
    assert bad < frac, (
        f"{bad*100:.3f}% mismatched; max abs err "
        f"{(out-ref).abs().max():.4f}")


def test_rmsnorm(kernels):
    torch.manual_seed(0)
    x = torch.randn(512, 4096, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
    out = kernels.rmsnorm(x, w, 1e-6)
    ref = R.rmsnorm(x.float().cpu(), w.float().cpu(), 1e-6)
    assert_close_bf16(out, ref)


def test_fused_add_rmsnorm(kernels):
    torch.manual_seed(1)
    x = torch.randn(256, 4096, dtype=torch.bfloat16, device="cuda")
    r = torch.randn(256, 4096, dtype=torch.bfloat16, device="cuda")
    x_ref = x.float().cpu().clone()
    r_ref = r.float().cpu().clone()
    w = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
    kernels.fused_add_rmsnorm(x, r, w, 1e-6)
    r2 = (x_ref + r_ref)
    x2 = R.rmsnorm(r2.to(torch.bfloat16).float(), w.float().cpu(), 1e-6)
    assert_close_bf16(r, r2)
    assert_close_bf16(x, x2)


def test_silu_and_mul(kernels):
    torch.manual_seed(2)
    x = torch.randn(333, 2 * 1536, dtype=torch.bfloat16, device="cuda")
    out = kernels.silu_and_mul(x)
    ref = R.silu_and_mul(x.float().cpu())
    assert_close_bf16(out, ref)


@pytest.mark.parametrize("is_neox", [True, False])
def test_rope(kernels, is_neox):
    torch.manual_seed(3)
    T, Hq, Hk, D = 100, 8, 2, 128
    inv = 1.0 / (10000 ** (torch.arange(0, D, 2).float() / D))
    t = torch.arange(4096).float()
    freqs = torch.outer(t, inv)
    cache = torch.cat([freqs.cos(), freqs.sin()], -1).cuda()
    q = torch.randn(T, Hq * D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(T, Hk * D, dtype=torch.bfloat16, device="cuda")
    pos = torch.randint(0, 4000, (T,), device="cuda")
    q_ref, k_ref = R.rotary_embedding(
        pos.cpu(), q.float().cpu().clone(), k.float().cpu().clone(), D,
        cache.cpu(), is_neox)
    kernels.rotary_embedding(pos, q, k, D, cache, is_neox)
    assert_close_bf16(q, q_ref)
    assert_close_bf16(k, k_ref)


def test_reshape_and_cache(kernels):
    torch.manual_seed(4)
    T, H, D, ps, P = 64, 8, 128, 16, 32
    k = torch.randn(T, H, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(T, H, D, dtype=torch.bfloat16, device="cuda")
    kc = torch.zeros(P, ps, H, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros(P, ps, H, D, dtype=torch.bfloat16, device="cuda")
    slots = torch.randperm(P * ps, device="cuda")[:T]
    kernels.reshape_and_cache(k, v, kc, vc, slots)
    kr = torch.zeros_like(kc).cpu()
    vr = torch.zeros_like(vc).cpu()
    R.reshape_and_cache(k.cpu(), v.cpu(), kr, vr, slots.cpu())
    assert torch.equal(kc.cpu(), kr)
    assert torch.equal(vc.cpu(), vr)


def _mk_paged(B, Hkv, D, ps, ctx_lens, seed=0):
    torch.manual_seed(seed)
    max_pages = max(-(-c // ps) for c in ctx_lens)
    total_pages = sum(-(-c // ps) for c in ctx_lens) + 1
    k_cache = torch.randn(total_pages, ps, Hkv, D, dtype=torch.bfloat16,
                          device="cuda")
    v_cache = torch.randn(total_pages, ps, Hkv, D, dtype=torch.bfloat16,
                          device="cuda")
    bt = torch.zeros(B, max_pages, dtype=torch.int32, device="cuda")
    next_page = 1
    for b, c in enumerate(ctx_lens):
        n = -(-c // ps)
        bt[b, :n] = torch.arange(next_page, next_page + n)
        next_page += n
    return k_cache, v_cache, bt


@pytest.mark.parametrize("G", [4, 5, 8])
def test_decode_attention(kernels, G):
    B, Hkv, D, ps = 5, 4, 128, 16
    Hq = G * Hkv
    ctx = [1, 17, 160, 1000, 333]
    k_cache, v_cache, bt = _mk_paged(B, Hkv, D, ps, ctx, seed=G)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl = torch.arange(B + 1, dtype=torch.int32, device="cuda")
    from gllm_amd import ops
    out = ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl,
                              1.0 / math.sqrt(D), max_query_len=1)
    ref = R.paged_attention(q.float().cpu(), k_cache.float().cpu(),
                            v_cache.float().cpu(), bt.cpu(), seq_lens.cpu(),
                            qsl.cpu(), 1.0 / math.sqrt(D))
    assert_close_bf16(out, ref, frac=2e-3)


@pytest.mark.parametrize("case", [
    # (q_lens, ctx_lens): ctx >= q (past = ctx - q)
    ([64], [64]),
    ([1, 64, 17], [100, 64, 333]),       # mixed decode + prefill + chunk
    ([200, 1], [200, 77]),
    ([130], [1030]),                     # long past (chunked context)
])
def test_prefill_attention(kernels, case):
    q_lens, ctx = case
    B, Hkv, D, ps = len(q_lens), 2, 128, 16
    G = 4
    Hq = G * Hkv
    k_cache, v_cache, bt = _mk_paged(B, Hkv, D, ps, ctx, seed=7)
    T = sum(q_lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl = torch.tensor([0] + list(torch.cumsum(
        torch.tensor(q_lens), 0)), dtype=torch.int32, device="cuda")
    from gllm_amd import ops
    out = ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl,
                              1.0 / math.sqrt(D),
                              max_query_len=max(q_lens))
    ref = R.paged_attention(q.float().cpu(), k_cache.float().cpu(),
                            v_cache.float().cpu(), bt.cpu(), seq_lens.cpu(),
                            qsl.cpu(), 1.0 / math.sqrt(D))
    assert_close_bf16(out, ref, frac=2e-3)


def test_prefill_attention_head_dim_64(kernels):
    B, Hkv, D, ps, G = 2, 2, 64, 16, 2
    Hq = G * Hkv
    ctx = [96, 40]
    q_lens = [96, 40]
    k_cache, v_cache, bt = _mk_paged(B, Hkv, D, ps, ctx, seed=9)
    T = sum(q_lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl = torch.tensor([0, 96, 136], dtype=torch.int32, device="cuda")
    from gllm_amd import ops
    out = ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl,
                              1.0 / math.sqrt(D), max_query_len=96)
    ref = R.paged_attention(q.float().cpu(), k_cache.float().cpu(),
                            v_cache.float().cpu(), bt.cpu(), seq_lens.cpu(),
                            qsl.cpu(), 1.0 / math.sqrt(D))
    assert_close_bf16(out, ref, frac=2e-3)


def test_sliding_window_prefill_and_decode(kernels):
    B, Hkv, D, ps, G, W = 2, 2, 128, 16, 4, 32
    Hq = G * Hkv
    ctx = [200, 77]
    q_lens = [200, 1]
    k_cache, v_cache, bt = _mk_paged(B, Hkv, D, ps, ctx, seed=11)
    T = sum(q_lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl = torch.tensor([0, 200, 201], dtype=torch.int32, device="cuda")
    from gllm_amd import ops
    out = ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl,
                              1.0 / math.sqrt(D), max_query_len=200,
                              sliding_window=W)
    ref = R.paged_attention(q.float().cpu(), k_cache.float().cpu(),
                            v_cache.float().cpu(), bt.cpu(), seq_lens.cpu(),
                            qsl.cpu(), 1.0 / math.sqrt(D), sliding_window=W)
    assert_close_bf16(out, ref, frac=2e-3)
    # decode-only path
    q1 = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    qsl1 = torch.arange(B + 1, dtype=torch.int32, device="cuda")
    out1 = ops.paged_attention(q1, k_cache, v_cache, bt, seq_lens, qsl1,
                               1.0 / math.sqrt(D), max_query_len=1,
                               sliding_window=W)
    ref1 = R.paged_attention(q1.float().cpu(), k_cache.float().cpu(),
                             v_cache.float().cpu(), bt.cpu(),
                             seq_lens.cpu(), qsl1.cpu(),
                             1.0 / math.sqrt(D), sliding_window=W)
    assert_close_bf16(out1, ref1, frac=2e-3)


# ---------------------------------------------------------------- MLA
def _mk_latent_paged(B, ps, ctx_lens, seed=0, DK=576):
    torch.manual_seed(seed)
    max_pages = max(-(-c // ps) for c in ctx_lens)
    total_pages = sum(-(-c // ps) for c in ctx_lens) + 1
    k_cache = torch.randn(total_pages, ps, 1, DK, dtype=torch.bfloat16,
                          device="cuda")
    bt = torch.zeros(B, max_pages, dtype=torch.int32, device="cuda")
    next_page = 1
    for b, c in enumerate(ctx_lens):
        n = -(-c // ps)
        bt[b, :n] = torch.arange(next_page, next_page + n)
        next_page += n
    return k_cache, bt


@pytest.mark.parametrize("case", [
    # (q_lens, ctx_lens, H)
    ([1, 1, 1], [5, 900, 333], 128),        # decode, split-KV path
    ([1], [4096], 128),                     # long decode, deep splits
    ([64, 17], [100, 333], 128),            # chunked prefill w/ past
    ([1, 130, 1], [77, 1030, 16], 16),      # mixed batch, TP=8 heads
    ([33], [33], 8),                        # fresh prefill, small H
])
def test_mla_attention(kernels, case):
    q_lens, ctx, H = case
    B, ps, DK, DV = len(q_lens), 16, 576, 512
    k_cache, bt = _mk_latent_paged(B, ps, ctx, seed=11)
    v_cache = k_cache[..., :DV]
    T = sum(q_lens)
    q = torch.randn(T, H, DK, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl = torch.tensor([0] + list(torch.cumsum(
        torch.tensor(q_lens), 0)), dtype=torch.int32, device="cuda")
    from gllm_amd import ops
    scale = 1.0 / math.sqrt(DK)
    out = ops.mla_paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl,
                                  scale, seq_lens_cpu=ctx,
                                  query_start_loc_cpu=[0] + list(
                                      torch.cumsum(torch.tensor(q_lens),
                                                   0).tolist()))
    ref = R.mla_paged_attention(q.float().cpu(), k_cache.float().cpu(),
                                v_cache.float().cpu(), bt.cpu(),
                                seq_lens.cpu(), qsl.cpu(), scale)
    assert_close_bf16(out, ref, frac=2e-3)


def test_cache_latent(kernels):
    torch.manual_seed(3)
    T, ps, P, DK = 33, 16, 12, 576
    k = torch.randn(T, 1, DK, dtype=torch.bfloat16, device="cuda")
    kc = torch.zeros(P, ps, 1, DK, dtype=torch.bfloat16, device="cuda")
    slots = torch.randperm(P * ps, device="cuda")[:T]
    from gllm_amd import ops
    ops.cache_latent(k, kc, slots)
    kr = torch.zeros(P, ps, 1, DK, dtype=torch.bfloat16)
    R.reshape_and_cache(k.cpu(), k.cpu(), kr, kr, slots.cpu())
    assert torch.equal(kc.cpu(), kr)


# ---------------------------------------------------------------- MoE
def _moe_ref(x, w13, w2, weights, ids, expert_start=0):
    """Per-pair gather loop in fp32 (oracle)."""
    T, H = x.shape
    E, two_i, _ = w13.shape
    out = torch.zeros(T, H, dtype=torch.float32)
    xf = x.float().cpu()
    w13f, w2f = w13.float().cpu(), w2.float().cpu()
    wf, idc = weights.float().cpu(), ids.cpu()
    for t in range(T):
        for k in range(idc.shape[1]):
            e = int(idc[t, k]) - expert_start
            if not 0 <= e < E:
                continue
            h = xf[t] @ w13f[e].T
            d = two_i // 2
            a = torch.nn.functional.silu(h[:d]) * h[d:]
            out[t] += wf[t, k] * (a @ w2f[e].T)
    return out


@pytest.mark.parametrize("case", [
    # (T, E, topk, hidden, inter, expert_start, E_local)
    (64, 8, 2, 512, 1024, 0, 8),       # Mixtral-ish dense routing
    (7, 64, 6, 256, 128, 0, 64),       # sparse DeepSeek-ish
    (33, 16, 4, 256, 512, 4, 8),       # EP shard (partial output)
    (300, 8, 2, 512, 256, 0, 8),       # prefill block_m=64 path
])
def test_fused_moe(kernels, case):
    T, E, topk, H, I, start, E_local = case
    torch.manual_seed(T + E)
    from gllm_amd import ops
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E_local, 2 * I, H, dtype=torch.bfloat16,
                      device="cuda") / math.sqrt(H)
    w2 = torch.randn(E_local, H, I, dtype=torch.bfloat16,
                     device="cuda") / math.sqrt(I)
    logits = torch.randn(T, E, device="cuda")
    weights, ids = ops.topk_softmax(logits, topk, True)
    out = ops.fused_moe(x, w13, w2, weights, ids.int(),
                        expert_start=start, num_global_experts=E)
    ref = _moe_ref(x, w13, w2, weights, ids, expert_start=start)
    assert_close_bf16(out, ref, atol=3e-2, rtol=3e-2, frac=2e-3)


def test_fused_moe_graph_capture(kernels):
    """The MoE pipeline must be hipGraph-capturable (no host syncs)."""
    from gllm_amd import ops
    T, E, topk, H, I = 16, 8, 2, 256, 512
    torch.manual_seed(0)
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * I, H, dtype=torch.bfloat16,
                      device="cuda") / math.sqrt(H)
    w2 = torch.randn(E, H, I, dtype=torch.bfloat16,
                     device="cuda") / math.sqrt(I)
    logits = torch.randn(T, E, device="cuda")
    weights, ids = ops.topk_softmax(logits, topk, True)
    ids = ids.int()
    eager = ops.fused_moe(x, w13, w2, weights, ids, 0, E)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = ops.fused_moe(x, w13, w2, weights, ids, 0, E)
    out.zero_()
    g.replay()
    torch.cuda.synchronize()
    assert torch.equal(out, eager)


# ------------------------------------------------------------ sampling
@pytest.mark.parametrize("case", [
    # (B, V, k, p, min_p)
    (8, 32000, 50, 0.9, 0.0),
    (4, 152064, -1, 0.8, 0.0),      # top-p only, big vocab
    (16, 32000, 5, 1.0, 0.0),       # top-k only
    (8, 32000, -1, 1.0, 0.05),      # min-p only
    (8, 32000, 200, 0.95, 0.02),    # all three
    (3, 32000, -1, 1.0, 0.0),       # disabled rows: untouched
])
def test_topk_topp_filter(kernels, case):
    B, V, k, p, mp = case
    torch.manual_seed(B * V + k)
    from gllm_amd import ops
    from gllm_amd.layers.sampler import Sampler
    logits = torch.randn(B, V, device="cuda") * 3
    probs = torch.softmax(logits, dim=-1)
    ks = torch.full((B,), k, dtype=torch.int32, device="cuda")
    ps = torch.full((B,), p, dtype=torch.float32, device="cuda")
    mps = torch.full((B,), mp, dtype=torch.float32, device="cuda")
    ref = Sampler._apply_top_k_top_p(probs.clone(), ks, ps)
    ref = Sampler._apply_min_p(ref, mps)
    out = probs.clone()
    ops.topk_topp_filter(out, ks, ps, mps)
    # exact same kept set (no ties in random fp32) and same renorm
    assert ((out > 0) == (ref > 0)).all(), (
        f"kept-set mismatch: {int((out > 0).sum())} vs "
        f"{int((ref > 0).sum())}")
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-4)


def test_topk_topp_filter_mixed_rows(kernels):
    """Per-row parameters differ (the serving case)."""
    torch.manual_seed(0)
    from gllm_amd import ops
    from gllm_amd.layers.sampler import Sampler
    B, V = 6, 50000
    probs = torch.softmax(torch.randn(B, V, device="cuda") * 2, dim=-1)
    ks = torch.tensor([1, 50, -1, 3000, -1, 10], dtype=torch.int32,
                      device="cuda")
    ps = torch.tensor([1.0, 0.9, 0.5, 0.99, 1.0, 1.0],
                      dtype=torch.float32, device="cuda")
    mps = torch.tensor([0.0, 0.0, 0.0, 0.01, 0.0, 0.0],
                       dtype=torch.float32, device="cuda")
    ref = Sampler._apply_top_k_top_p(probs.clone(), ks, ps)
    ref = Sampler._apply_min_p(ref, mps)
    out = probs.clone()
    ops.topk_topp_filter(out, ks, ps, mps)
    assert ((out > 0) == (ref > 0)).all()
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-4)


# ------------------------------------------------------------ fp8
def test_per_token_group_quant(kernels):
    torch.manual_seed(5)
    from gllm_amd import ops
    from gllm_amd.layers.quantization import fp8 as qfp8
    x = torch.randn(37, 512, dtype=torch.bfloat16, device="cuda") * 3
    q, s = ops.per_token_group_quant_fp8(x)
    qr, sr = qfp8.per_token_group_quant_fp8(x.cpu())
    assert torch.allclose(s.cpu(), sr, rtol=1e-4)
    # transposed aux output (skinny GEMM staging layout)
    q2, s2, st = ops.per_token_group_quant_fp8(x, transposed=True)
    assert torch.equal(st, s2.t().contiguous())
    assert torch.equal(q2.view(torch.uint8), q.view(torch.uint8))
    # round trip within e4m3 quantization error (3-bit mantissa: half a
    # quantum = amax/448 * 16 near the top of the range); the raw codes
    # may differ by one step where mul-by-1/s vs div-by-s rounds apart
    dq = (q.float().view(37, 4, 128) * s.unsqueeze(-1)).view(37, 512)
    err = (dq.cpu() - x.float().cpu()).abs()
    tol = (sr * 16.5).repeat_interleave(128, dim=1)
    assert (err <= tol).all(), float((err - tol).max())


@pytest.mark.parametrize("case", [
    # (M, N, K, bias?) — bias exercises the fused splitk==1 epilogue
    # and the reduce-kernel bias path
    (1, 1024, 512, False), (8, 896, 1024, True), (64, 2048, 896, False),
    (64, 55296 // 9, 1152, True), (100, 768, 1024, True),
    (200, 512, 1280, True),
])
def test_fp8_linear(kernels, case):
    M, N, K, with_bias = case
    torch.manual_seed(M + N)
    from gllm_amd import ops
    from gllm_amd.layers.quantization import fp8 as qfp8
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16) / math.sqrt(K)
    bias = torch.randn(N, dtype=torch.float32, device="cuda")         if with_bias else None
    wq, ws = qfp8.block_quant_fp8(w)
    out = ops.fp8_linear(x, wq.cuda(), ws.cuda(), bias)
    # oracle: dequantized weights x quantized activations in fp32
    aq, as_ = qfp8.per_token_group_quant_fp8(x.cpu())
    adq = aq.float().view(M, K // 128, 128) * as_.unsqueeze(-1)
    wdq = qfp8.dequant_block_fp8(wq, ws, (128, 128), torch.float32)
    ref = adq.view(M, K).float() @ wdq.T
    if bias is not None:
        ref = ref + bias.cpu().float()
    assert_close_bf16(out, ref, atol=5e-2, rtol=5e-2, frac=2e-3)


def test_fused_moe_fp8(kernels):
    T, E, topk, H, I = 45, 8, 2, 512, 256
    torch.manual_seed(9)
    from gllm_amd import ops
    from gllm_amd.layers.quantization import fp8 as qfp8
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13l, w2l, w13q, w2q, w13s, w2s = [], [], [], [], [], []
    for e in range(E):
        a = torch.randn(2 * I, H, dtype=torch.bfloat16) / math.sqrt(H)
        b = torch.randn(H, I, dtype=torch.bfloat16) / math.sqrt(I)
        qa, sa = qfp8.block_quant_fp8(a)
        qb, sb = qfp8.block_quant_fp8(b)
        w13q.append(qa); w2q.append(qb); w13s.append(sa); w2s.append(sb)
        w13l.append(qfp8.dequant_block_fp8(qa, sa, (128, 128),
                                           torch.bfloat16))
        w2l.append(qfp8.dequant_block_fp8(qb, sb, (128, 128),
                                          torch.bfloat16))
    logits = torch.randn(T, E, device="cuda")
    weights, ids = ops.topk_softmax(logits, topk, True)
    out = ops.fused_moe_fp8(
        x, torch.stack(w13q).cuda(), torch.stack(w13s).cuda(),
        torch.stack(w2q).cuda(), torch.stack(w2s).cuda(), weights,
        ids.int())
    ref = _moe_ref(x, torch.stack(w13l).cuda(), torch.stack(w2l).cuda(),
                   weights, ids)
    # activation quant noise on top of weight quant: loose tolerance
    assert_close_bf16(out, ref, atol=8e-2, rtol=8e-2, frac=5e-3)


# ------------------------------------------------------------ gdn
def test_gdn_decode_kernels(kernels):
    """Batched conv-update + fused recurrent step vs the sequential
    torch oracle (gdn_ref), including state roll parity."""
    from gllm_amd import ops
    from gllm_amd.ops import gdn_ref
    torch.manual_seed(4)
    B, Hk, Hv, Dk, Dv, K = 5, 2, 4, 128, 128, 4
    C = 2 * Hk * Dk + Hv * Dv
    slots = torch.tensor([3, 0, 7, 1, 5], dtype=torch.long, device="cuda")
    n_slots = 8
    x = torch.randn(B, C, dtype=torch.bfloat16, device="cuda") / 2
    w = torch.randn(C, K, dtype=torch.bfloat16, device="cuda") / 2
    conv_pool = torch.randn(n_slots, C, K - 1, dtype=torch.bfloat16,
                            device="cuda") / 2
    conv_ref = conv_pool.clone()
    out = ops.gdn_conv_update(x, w, conv_pool, slots)
    for i in range(B):
        ref = gdn_ref.causal_conv1d_update(
            x[i].cpu(), w.cpu(), conv_ref[slots[i]].cpu().clone())
        st = conv_ref[slots[i]].cpu()
        ctx = torch.cat([st.float(), x[i].cpu().float().unsqueeze(1)], 1)
        conv_ref[slots[i]] = ctx[:, 1:].to(torch.bfloat16).cuda()
        assert_close_bf16(out[i], ref.float(), atol=3e-2, rtol=3e-2)
    assert torch.equal(conv_pool, conv_ref)

    # recurrent step
    state_pool = torch.randn(n_slots, Hv, Dv, Dk, device="cuda") / 8
    state_ref = state_pool.clone()
    q = torch.randn(B, Hk, Dk, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, Hk, Dk, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, Hv, Dv, dtype=torch.bfloat16, device="cuda")
    g = -torch.rand(B, Hv, device="cuda") * 0.1
    beta = torch.rand(B, Hv, device="cuda")
    scale = Dk ** -0.5
    G = Hv // Hk
    qn = gdn_ref.l2norm(q.float()) * scale
    kn = gdn_ref.l2norm(k.float())
    o = ops.gdn_decode(qn.repeat_interleave(G, 1),
                       kn.repeat_interleave(G, 1), v.float(), g, beta,
                       state_pool, slots)
    for i in range(B):
        st = state_ref[slots[i]].cpu()
        oref = gdn_ref.gated_delta_rule(
            q[i:i + 1].cpu(), k[i:i + 1].cpu(), v[i:i + 1].cpu(),
            g[i:i + 1].cpu(), beta[i:i + 1].cpu(), scale, st)
        state_ref[slots[i]] = st.cuda()
        assert_close_bf16(o[i], oref[0].float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(state_pool, state_ref, atol=1e-4, rtol=1e-4)


def test_rmsnorm_gated(kernels):
    from gllm_amd import ops
    from gllm_amd.ops import gdn_ref
    torch.manual_seed(6)
    x = torch.randn(300, 128, dtype=torch.bfloat16, device="cuda")
    z = torch.randn(300, 128, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(128, dtype=torch.bfloat16, device="cuda")
    out = ops.rmsnorm_gated(x, z, w, 1e-6)
    ref = gdn_ref.rmsnorm_gated(x.cpu(), z.cpu(), w.cpu(), 1e-6)
    assert_close_bf16(out, ref.float())


# ------------------------------------------------------------ int4
@pytest.mark.parametrize("case", [(8, 512, 1024, False),
                                  (64, 896, 2048, True),
                                  (64, 4096, 1280, True),
                                  (100, 768, 1536, True),
                                  (200, 1024, 512, True)])
def test_int4_linear(kernels, case):
    M, N, K, with_bias = case
    torch.manual_seed(M)
    from types import SimpleNamespace
    from gllm_amd import ops
    from gllm_amd.layers.quantization import int4 as qi4
    w = torch.randn(N, K) / math.sqrt(K)
    qweight, qzeros, scales = qi4.pack_gptq(w, group_size=128)
    layer = SimpleNamespace(int4_cfg=("gptq", 128), qweight=qweight,
                            qzeros=qzeros, scales=scales)
    wq4, sb, grp = qi4.repack_canonical(layer)
    sbt = sb.permute(1, 2, 0).contiguous()
    ref_w = qi4.dequant_gptq(qweight, qzeros, scales, 128, torch.float32)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    bias = torch.randn(N, dtype=torch.float32, device="cuda")         if with_bias else None
    out = ops.int4_linear(x, wq4.cuda(), sbt.cuda(), grp, bias)
    ref = x.float().cpu() @ ref_w.T
    if bias is not None:
        ref = ref + bias.cpu().float()
    assert_close_bf16(out, ref, atol=5e-2, rtol=5e-2, frac=2e-3)


def test_int4_prefill_linear(kernels):
    """Identity-routed grouped int4 GEMM (dense M > 256 path)."""
    M, N, K = 300, 896, 1024
    torch.manual_seed(4)
    from types import SimpleNamespace
    from gllm_amd import ops
    from gllm_amd.layers.quantization import int4 as qi4
    w = torch.randn(N, K) / math.sqrt(K)
    qweight, qzeros, scales = qi4.pack_gptq(w, group_size=128)
    layer = SimpleNamespace(int4_cfg=("gptq", 128), qweight=qweight,
                            qzeros=qzeros, scales=scales)
    wq4, sb, _ = qi4.repack_canonical(layer)
    ref_w = qi4.dequant_gptq(qweight, qzeros, scales, 128, torch.float32)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    bias = torch.randn(N, dtype=torch.bfloat16, device="cuda")
    out = ops.int4_prefill_linear(x, wq4.cuda(), sb.cuda(), bias)
    ref = x.float().cpu() @ ref_w.T + bias.float().cpu()
    assert_close_bf16(out, ref, atol=5e-2, rtol=5e-2, frac=2e-3)


def test_fp8_prefill_linear(kernels):
    """Identity-routed grouped fp8 GEMM (dense M > 256 path)."""
    M, N, K = 300, 1024, 512
    torch.manual_seed(6)
    from gllm_amd import ops
    from gllm_amd.layers.quantization import fp8 as qfp8
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16) / math.sqrt(K)
    wq, ws = qfp8.block_quant_fp8(w)
    out = ops.fp8_prefill_linear(x, wq.cuda(), ws.cuda())
    aq, as_ = qfp8.per_token_group_quant_fp8(x.cpu())
    adq = aq.float().view(M, K // 128, 128) * as_.unsqueeze(-1)
    wdq = qfp8.dequant_block_fp8(wq, ws, (128, 128), torch.float32)
    ref = adq.view(M, K).float() @ wdq.T
    assert_close_bf16(out, ref, atol=5e-2, rtol=5e-2, frac=2e-3)


def test_fused_moe_int4(kernels):
    """w4a16 grouped MoE vs the dequantized bf16 reference."""
    from types import SimpleNamespace
    from gllm_amd import ops
    from gllm_amd.layers.quantization import int4 as qi4
    torch.manual_seed(13)
    T, E, topk, H, I = 37, 4, 2, 256, 256
    w13q, w2q, w13d, w2d = [], [], [], []
    for e in range(E):
        a = torch.randn(2 * I, H) / math.sqrt(H)
        b = torch.randn(H, I) / math.sqrt(I)
        qa = qi4.pack_gptq(a, group_size=128)
        qb = qi4.pack_gptq(b, group_size=128)
        w13q.append(qa)
        w2q.append(qb)
        w13d.append(qi4.dequant_gptq(*qa, 128, torch.bfloat16))
        w2d.append(qi4.dequant_gptq(*qb, 128, torch.bfloat16))
    moe = SimpleNamespace(
        int4_cfg=("gptq", 128), num_local_experts=E,
        w13_qweight=[q[0] for q in w13q], w13_qzeros=[q[1] for q in w13q],
        w13_scales=[q[2] for q in w13q],
        w2_qweight=[q[0] for q in w2q], w2_qzeros=[q[1] for q in w2q],
        w2_scales=[q[2] for q in w2q])
    w13c, w13sb, w2c, w2sb, grp = qi4.repack_canonical_moe(moe)
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    logits = torch.randn(T, E, device="cuda")
    weights, ids = ops.topk_softmax(logits, topk, True)
    out = ops.fused_moe_int4(x, w13c.cuda(), w13sb.cuda(), w2c.cuda(),
                             w2sb.cuda(), weights, ids.int())
    ref = _moe_ref(x, torch.stack(w13d).cuda(), torch.stack(w2d).cuda(),
                   weights, ids)
    assert_close_bf16(out, ref, atol=5e-2, rtol=5e-2, frac=3e-3)


@pytest.mark.parametrize("heads", [(2, 4), (4, 4)])
def test_gdn_chunk_prefill(kernels, heads):
    """Fused WY chunk prefill vs the fp32 torch oracle: padded batch,
    multi-chunk sequences, ragged lengths (inert padding), GQA repeat
    (and the G=1 Hk==Hv case), and in-place state update."""
    from gllm_amd import ops
    from gllm_amd.ops import gdn_ref
    torch.manual_seed(11)
    Hk, Hv = heads
    B, T, D = 3, 192, 128
    lens = [192, 130, 64]
    q = torch.randn(B, T, Hk, D, device="cuda") * 0.5
    k = torch.randn(B, T, Hk, D, device="cuda") * 0.5
    v = torch.randn(B, T, Hv, D, device="cuda") * 0.5
    g = -torch.rand(B, T, Hv, device="cuda") * 0.1
    beta = torch.rand(B, T, Hv, device="cuda") * 0.9 + 0.05
    for b, n in enumerate(lens):  # padding rows must be inert
        g[b, n:] = 0
        beta[b, n:] = 0
        q[b, n:] = 0
        k[b, n:] = 0
        v[b, n:] = 0
    st = torch.randn(B, Hv, D, D, device="cuda") * 0.05
    st_hip = st.clone()
    st_ref = st.clone()
    # oracle consumes the same bf16-rounded inputs the kernel reads
    qh = q.to(torch.bfloat16).float()
    kh = k.to(torch.bfloat16).float()
    vh = v.to(torch.bfloat16).float()
    out = ops.gdn_chunk_prefill(q, k, v, g, beta, st_hip, 0.5)
    ref = gdn_ref.gated_delta_rule_chunked_batched(
        qh, kh, vh, g, beta, 0.5, st_ref, chunk=64)
    for b, n in enumerate(lens):
        assert_close_bf16(out[b, :n].float(), ref[b, :n].float(),
                          atol=2e-2, rtol=2e-2, frac=1e-3)
    assert torch.allclose(st_hip, st_ref, atol=2e-2, rtol=2e-2)


def test_gelu_and_mul(kernels):
    torch.manual_seed(6)
    x = torch.randn(97, 2 * 1024, dtype=torch.bfloat16, device="cuda")
    out = kernels.gelu_and_mul(x)
    ref = R.gelu_and_mul(x.float().cpu())
    assert_close_bf16(out, ref)
