"""int4 weight-only quantization (AWQ / GPTQ checkpoint formats).

Parity target: reference layers/moe/fused_moe_triton/layer.py
Int4MarlinMoEMethod + the gptq-marlin dense path (_custom_ops.py:
gptq_marlin_repack / moe_wna16_marlin_gemm). MI355X stance: the round-2
path is an LDS-resident dequant-fused MFMA GEMM (weights stay packed in
registers/LDS, unpacked on the fly — int4 is memory-bound decode's best
friend at 4x bf16 density); round-1 executes like fp8: weights load in
the checkpoint's packed layout and are dequantized lazily (cached) into
the compute dtype, so numerics are exact w.r.t. the format. Packed
tensors shard across TP at load time: column-family layers slice the
stored N axis (checkpoints keep weights K-major, so the output dim is
dim 1 of every packed tensor), row-parallel slices the stored K axis —
qweight by packed rows, qzeros/scales by whole quant groups, which
requires K/tp to be a multiple of group_size (the standard wna16
restriction; group 128 divides every real K/tp).

Formats (public conventions):
  GPTQ (bits=4, no act-order): qweight int32 [K/8, N], K packed
    little-nibble-first; qzeros int32 [G, N/8] (stored zero is z-1: add
    1 when unpacking — the autogptq legacy quirk); scales fp16 [G, N];
    w[k, n] = s[g, n] * (q[k, n] - (z[g, n] + 1)),  g = k // group.
  AWQ (bits=4): qweight int32 [K, N/8] with the interleaved nibble
    order [0, 2, 4, 6, 1, 3, 5, 7]; qzeros int32 [G, N/8] same order;
    scales fp16 [G, N]; w[k, n] = s * (q - z).
Both store weights K-major ([K, N]); torch linear wants [N, K], so
dequant transposes at the end.
"""


import torch
import torch.nn as nn

AWQ_ORDER = (0, 2, 4, 6, 1, 3, 5, 7)


def _unpack_nibbles_k(packed: torch.Tensor) -> torch.Tensor:
    """GPTQ qweight [K/8, N] int32 -> [K, N] int, nibble i of word w is
    row 8*w + i."""
    Kp, N = packed.shape
    out = torch.empty(Kp * 8, N, dtype=torch.int32, device=packed.device)
    p = packed.to(torch.int64) & 0xFFFFFFFF
    for i in range(8):
        # word w's nibble i is row 8*w + i == out[i::8][w]
        out[i::8] = ((p >> (4 * i)) & 0xF).to(torch.int32)
    return out


def _unpack_nibbles_n(packed: torch.Tensor, order=None) -> torch.Tensor:
    """[R, N/8] int32 -> [R, N] int; nibble i of a word is column
    8*w + order[i] (AWQ interleave) or 8*w + i (GPTQ zeros)."""
    R, Np = packed.shape
    order = order or tuple(range(8))
    out = torch.empty(R, Np * 8, dtype=torch.int32, device=packed.device)
    p = packed.to(torch.int64) & 0xFFFFFFFF
    for i, col in enumerate(order):
        out[:, col::8] = ((p >> (4 * i)) & 0xF).to(torch.int32)
    return out


def dequant_gptq(qweight, qzeros, scales, group_size: int,
                 dtype: torch.dtype) -> torch.Tensor:
    """-> [N, K] dense weight."""
    wq = _unpack_nibbles_k(qweight)                   # [K, N]
    zeros = _unpack_nibbles_n(qzeros) + 1             # [G, N] (legacy +1)
    K = wq.shape[0]
    g = torch.arange(K, device=wq.device) // group_size
    w = (wq.float() - zeros.float()[g]) * scales.float()[g]
    return w.t().contiguous().to(dtype)


def dequant_awq(qweight, qzeros, scales, group_size: int,
                dtype: torch.dtype) -> torch.Tensor:
    """-> [N, K] dense weight."""
    wq = _unpack_nibbles_n(qweight, AWQ_ORDER)        # [K, N]
    zeros = _unpack_nibbles_n(qzeros, AWQ_ORDER)      # [G, N]
    K = wq.shape[0]
    g = torch.arange(K, device=wq.device) // group_size
    w = (wq.float() - zeros.float()[g]) * scales.float()[g]
    return w.t().contiguous().to(dtype)


# --------------------------------------------------------- test packing
def pack_gptq(w: torch.Tensor, group_size: int):
    """[N, K] fp -> (qweight, qzeros, scales); round-trip oracle for
    tests (asymmetric per-group min/max quant)."""
    N, K = w.shape
    assert K % group_size == 0 and K % 8 == 0 and N % 8 == 0
    wk = w.t().float()                                # [K, N]
    G = K // group_size
    wg = wk.view(G, group_size, N)
    mn, mx = wg.min(1).values, wg.max(1).values       # [G, N]
    scales = ((mx - mn) / 15.0).clamp_min(1e-8)
    # the stored-z-1 convention cannot represent z = 0 (real GPTQ
    # packers keep z in [1, 15]); clamping low keeps dequant's +1 exact
    zeros = (-mn / scales).round().clamp(1, 15)       # [G, N]
    g = torch.arange(K) // group_size
    q = (wk / scales[g] + zeros[g]).round().clamp(0, 15).to(torch.int64)
    qweight = torch.zeros(K // 8, N, dtype=torch.int64)
    for i in range(8):
        qweight |= q[i::8] << (4 * i)
    zstore = zeros.to(torch.int64) - 1                 # legacy z-1
    qzeros = torch.zeros(G, N // 8, dtype=torch.int64)
    for i in range(8):
        qzeros |= zstore[:, i::8] << (4 * i)
    return (qweight.to(torch.int32), qzeros.to(torch.int32),
            scales.to(torch.float16))


def pack_awq(w: torch.Tensor, group_size: int):
    N, K = w.shape
    assert K % group_size == 0 and N % 8 == 0
    wk = w.t().float()
    G = K // group_size
    wg = wk.view(G, group_size, N)
    mn, mx = wg.min(1).values, wg.max(1).values
    scales = ((mx - mn) / 15.0).clamp_min(1e-8)
    zeros = (-mn / scales).round().clamp(0, 15)
    g = torch.arange(K) // group_size
    q = (wk / scales[g] + zeros[g]).round().clamp(0, 15).to(torch.int64)
    qweight = torch.zeros(K, N // 8, dtype=torch.int64)
    zq = zeros.to(torch.int64)
    qzeros = torch.zeros(G, N // 8, dtype=torch.int64)
    for i, col in enumerate(AWQ_ORDER):
        qweight |= q[:, col::8] << (4 * i)
        qzeros |= zq[:, col::8] << (4 * i)
    return (qweight.to(torch.int32), qzeros.to(torch.int32),
            scales.to(torch.float16))


# ------------------------------------------------------ layer conversion
_QUANT_SUFFIXES = (
    "q_proj", "k_proj", "v_proj", "o_proj", "qkv_proj",
    "gate_proj", "up_proj", "down_proj", "gate_up_proj",
)


def convert_linear_to_int4(layer, method: str, group_size: int) -> None:
    """Replace the dense weight with packed int4 params. Checkpoint
    names (qweight/qzeros/scales) route to them directly through the
    models' substring-mapped loaders. Under TP each loader takes this
    rank's slice of the full checkpoint tensor: dim 1 (the stored N
    axis) for the column family, dim 0 (packed K rows / quant groups)
    for row-parallel."""
    from gllm_amd.layers.linear import (MergedColumnParallelLinear,
                                        QKVParallelLinear,
                                        RowParallelLinear)
    from gllm_amd.parallel import get_tp_rank, get_tp_size
    tp, tp_rank = get_tp_size(), get_tp_rank()
    N, K = layer.weight.shape  # already this rank's shard
    row = isinstance(layer, RowParallelLinear)
    if tp > 1:
        if row:
            assert K % group_size == 0 and K % 8 == 0, (
                f"int4 row-parallel needs K/tp divisible by the quant "
                f"group ({group_size}) and the pack factor: K/tp={K}")
        elif isinstance(layer, QKVParallelLinear):
            assert layer.q_size % 8 == 0 and layer.kv_size % 8 == 0
        elif isinstance(layer, MergedColumnParallelLinear):
            assert all(s // tp % 8 == 0 for s in layer.output_sizes)
        else:
            assert N % 8 == 0, f"int4 column shard N/tp={N} not 8-aligned"
    G = K // group_size

    def _mk_loader(div: int):
        """Build the loader for one packed tensor. ``div`` is that
        tensor's N packing factor (8 for qzeros / AWQ qweight, 1 for
        scales / GPTQ qweight); stored dim 1 holds N/div columns, so a
        rank's slice of a full tensor is loaded.shape[axis] // tp wide
        on the shard axis. Merged/QKV additionally place the slice at
        the sub-projection's offset inside the fused param."""
        if isinstance(layer, MergedColumnParallelLinear):
            def load(param, loaded, shard_id: int):
                off = sum(layer.output_sizes[:shard_id]) // tp // div
                size = layer.output_sizes[shard_id] // tp // div
                param.data.narrow(1, off, size).copy_(
                    loaded.narrow(1, tp_rank * size, size))
        elif isinstance(layer, QKVParallelLinear):
            def load(param, loaded, shard_id: str):
                if shard_id == "q":
                    off, size, src = 0, layer.q_size, tp_rank
                elif shard_id == "k":
                    off, size = layer.q_size, layer.kv_size
                    src = tp_rank // layer.kv_replication
                else:
                    off, size = layer.q_size + layer.kv_size, layer.kv_size
                    src = tp_rank // layer.kv_replication
                param.data.narrow(1, off // div, size // div).copy_(
                    loaded.narrow(1, src * (size // div), size // div))
        elif row:
            def load(param, loaded, *a):
                shard = loaded.shape[0] // tp
                param.data.copy_(loaded.narrow(0, tp_rank * shard, shard))
        else:
            def load(param, loaded, *a):
                shard = loaded.shape[1] // tp
                param.data.copy_(loaded.narrow(1, tp_rank * shard, shard))
        return load

    if method == "gptq":
        qw_shape, qw_div = (K // 8, N), 1
    else:
        qw_shape, qw_div = (K, N // 8), 8
    layer.qweight = nn.Parameter(
        torch.zeros(qw_shape, dtype=torch.int32), requires_grad=False)
    layer.qweight.weight_loader = _mk_loader(qw_div)
    layer.qzeros = nn.Parameter(
        torch.zeros(G, N // 8, dtype=torch.int32), requires_grad=False)
    layer.qzeros.weight_loader = _mk_loader(8)
    layer.scales = nn.Parameter(
        torch.zeros(G, N, dtype=torch.float16), requires_grad=False)
    layer.scales.weight_loader = _mk_loader(1)
    # drop the dense weight (it has no checkpoint tensor)
    layer.weight = nn.Parameter(torch.zeros(1, dtype=torch.float32),
                                requires_grad=False)
    layer.weight.weight_loader = lambda param, loaded, *a: None
    layer.int4_cfg = (method, group_size)
    layer._w_dq = None


def convert_moe_to_int4(moe, method: str, group_size: int) -> None:
    """Swap a FusedMoE's stacked expert weights for packed int4 banks
    (AWQ Mixtral / GPTQ Qwen-MoE releases quantize every routed expert;
    the router gate stays dense). Checkpoint tensors are per-expert in
    the dense formats' K-major layout (w13: K=hidden, N=intermediate —
    column-like, TP slices stored N; w2: K=intermediate — row-like, TP
    slices packed K rows and whole quant groups), so stacking offsets
    and shards must be pack- (8) and group-aligned."""
    from gllm_amd.parallel import get_tp_rank, get_tp_size
    tp_rank = get_tp_rank()
    tp = 1 if moe.use_ep else get_tp_size()
    E = moe.num_local_experts
    I, H = moe.intermediate_per_rank, moe.hidden_size
    g = group_size
    assert H % g == 0 and H % 8 == 0 and I % g == 0 and I % 8 == 0, (
        f"int4 MoE needs hidden ({H}) and per-rank intermediate ({I}) "
        f"aligned to the quant group ({g}) and the pack factor")

    def mk(shape, dtype, loader):
        p = nn.Parameter(torch.zeros(shape, dtype=dtype),
                         requires_grad=False)
        p.weight_loader = loader
        return p

    def col_loader(param, loaded, expert_id: int, shard_id: int):
        # w13 tensors: place this rank's stored-N slice at the
        # sub-projection's offset (dim 1 of the per-expert tensor)
        lid = moe._local_expert(expert_id)
        if lid is None:
            return
        size = loaded.shape[1] // tp
        shard = loaded.narrow(1, tp_rank * size, size) if tp > 1 else loaded
        param.data[lid].narrow(1, shard_id * size, size).copy_(shard)

    def row_loader(param, loaded, expert_id: int):
        # w2 tensors: this rank's stored-K row slice (packed rows for
        # qweight, whole quant groups for qzeros/scales)
        lid = moe._local_expert(expert_id)
        if lid is None:
            return
        size = loaded.shape[0] // tp
        shard = loaded.narrow(0, tp_rank * size, size) if tp > 1 else loaded
        param.data[lid].copy_(shard)

    if method == "gptq":
        w13_qw, w2_qw = (E, H // 8, 2 * I), (E, I // 8, H)
    else:
        w13_qw, w2_qw = (E, H, 2 * I // 8), (E, I, H // 8)
    moe.w13_qweight = mk(w13_qw, torch.int32, col_loader)
    moe.w13_qzeros = mk((E, H // g, 2 * I // 8), torch.int32, col_loader)
    moe.w13_scales = mk((E, H // g, 2 * I), torch.float16, col_loader)
    moe.w2_qweight = mk(w2_qw, torch.int32, row_loader)
    moe.w2_qzeros = mk((E, I // g, H // 8), torch.int32, row_loader)
    moe.w2_scales = mk((E, I // g, H), torch.float16, row_loader)
    # drop the dense banks (no checkpoint tensor feeds them)
    for wname in ("w13_weight", "w2_weight"):
        dead = nn.Parameter(torch.zeros(1), requires_grad=False)
        dead.weight_loader = lambda param, loaded, *a: None
        setattr(moe, wname, dead)
    moe.int4_cfg = (method, group_size)
    moe._dq_cache = None


def convert_model_to_int4(model, quant_config: dict) -> int:
    from gllm_amd.layers.linear import LinearBase
    from gllm_amd.layers.moe.layer import FusedMoE
    method = quant_config["quant_method"]
    group = int(quant_config.get("group_size", 128))
    n = 0
    for name, mod in model.named_modules():
        if isinstance(mod, FusedMoE):
            convert_moe_to_int4(mod, method, group)
            n += 1
            continue
        if not isinstance(mod, LinearBase):
            continue
        if name.rsplit(".", 1)[-1] in _QUANT_SUFFIXES:
            convert_linear_to_int4(mod, method, group)
            n += 1
    return n


def dequant_layer(layer, dtype: torch.dtype) -> torch.Tensor:
    method, group = layer.int4_cfg
    fn = dequant_gptq if method == "gptq" else dequant_awq
    return fn(layer.qweight, layer.qzeros, layer.scales, group, dtype)


# ------------------------------------------------- canonical repack (r2)
def repack_canonical(layer, dtype=torch.bfloat16):
    """One-time repack of a converted int4 linear's checkpoint tensors
    into the gfx950 fused-dequant GEMM layout (ops/csrc/int4.hip):

      wq4  uint8 [N, K/2]      (low nibble = even k, high = odd k)
      sb   fp32  [N, K/g, 2]   ({scale, -zero*scale} per group)

    Returns (wq4, sb, group). Runs in torch once (the reference's
    gptq_marlin_repack role); caller caches the result and FREES the
    original packed tensors.
    """
    method, group = layer.int4_cfg
    if method == "gptq":
        wq = _unpack_nibbles_k(layer.qweight)              # [K, N]
        zeros = _unpack_nibbles_n(layer.qzeros) + 1        # [G, N]
    else:
        wq = _unpack_nibbles_n(layer.qweight, AWQ_ORDER)   # [K, N]
        zeros = _unpack_nibbles_n(layer.qzeros, AWQ_ORDER)
    scales = layer.scales.float()                          # [G, N]
    wq = wq.t().contiguous().to(torch.uint8)               # [N, K]
    N, K = wq.shape
    wq4 = (wq[:, 0::2] | (wq[:, 1::2] << 4)).contiguous()  # [N, K/2]
    s = scales.t().contiguous()                            # [N, G]
    b = (-zeros.float() * scales).t().contiguous()         # [N, G]
    sb = torch.stack([s, b], dim=-1).contiguous()          # [N, G, 2]
    return wq4, sb, group


def repack_canonical_moe(moe):
    """Per-expert canonical repack of a FusedMoE's int4 banks for the
    grouped w4a16 GEMM (ops/csrc/int4.hip::moe_gemm_int4):
    returns (w13_c [E, 2I, H/2] u8, w13_sb [E, 2I, H/g, 2] f32,
             w2_c [E, H, I/2], w2_sb [E, H, I/g, 2], group)."""
    from types import SimpleNamespace
    method, group = moe.int4_cfg
    banks = []
    for name in ("w13", "w2"):
        cs, sbs = [], []
        for e in range(moe.num_local_experts):
            shim = SimpleNamespace(
                int4_cfg=moe.int4_cfg,
                qweight=getattr(moe, f"{name}_qweight")[e],
                qzeros=getattr(moe, f"{name}_qzeros")[e],
                scales=getattr(moe, f"{name}_scales")[e])
            wq4, sb, _ = repack_canonical(shim)
            cs.append(wq4)
            sbs.append(sb)
        banks.append((torch.stack(cs).contiguous(),
                      torch.stack(sbs).contiguous()))
    return banks[0][0], banks[0][1], banks[1][0], banks[1][1], group


# ------------------------------------------------------------------
# compressed-tensors "pack-quantized" int4 (Kimi-K2.5 routed experts;
# reference model_loader.py:538-591 _normalize_kimi_quant_config +
# fused_moe_triton/layer.py:229-545 Int4MarlinMoEMethod). Layout per
# expert: weight_packed int32 [N, K/8] (8 x int4 along K, nibble j at
# bits 4j), SYMMETRIC with offset 8 (uint4b8: value = q - 8), and
# weight_scale [N, K/group] group scales. No zero points.

def pack_ct_int4(w: torch.Tensor, group_size: int = 32):
    """Quantize [N, K] -> (weight_packed int32 [N, K/8],
    weight_scale [N, K/group]) in the compressed-tensors symmetric
    int4 layout (test/tooling helper; checkpoints arrive packed)."""
    N, K = w.shape
    assert K % group_size == 0 and K % 8 == 0
    wg = w.float().view(N, K // group_size, group_size)
    scale = wg.abs().amax(-1).clamp(min=1e-8) / 7.0
    q = torch.clamp(torch.round(wg / scale.unsqueeze(-1)), -8, 7)
    qu = (q + 8).to(torch.int32).view(N, K // 8, 8)
    packed = torch.zeros(N, K // 8, dtype=torch.int32)
    for j in range(8):
        packed |= qu[:, :, j] << (4 * j)
    return packed, scale.to(w.dtype)


def dequant_ct_int4(packed: torch.Tensor, scale: torch.Tensor,
                    group_size: int, dtype=torch.bfloat16) -> torch.Tensor:
    """(int32 [N, K/8], scales [N, K/group]) -> [N, K] dtype."""
    N, Kp = packed.shape
    K = Kp * 8
    shifts = torch.arange(8, device=packed.device) * 4
    q = (packed.unsqueeze(-1) >> shifts) & 0xF      # [N, K/8, 8]
    q = q.view(N, K).float() - 8.0
    s = scale.float().repeat_interleave(group_size, dim=1)
    return (q * s).to(dtype)


def convert_moe_to_int4_packed(model, mq: dict) -> int:
    """Swap every FusedMoE's bf16 expert banks for compressed-tensors
    packed-int4 parameters (+ group scales) with TP/EP-aware loaders.
    Execution: dequantized bf16 banks feed the grouped MFMA MoE GEMM
    (memory for the shadow bank comes out of the 288 GB pool — the
    group-32 native kernel variant is a ROADMAP item)."""
    import torch.nn as nn
    from gllm_amd.layers.moe.layer import FusedMoE
    group = int(mq.get("group_size", 32))
    n = 0
    for moe in model.modules():
        if not isinstance(moe, FusedMoE):
            continue
        E = moe.num_local_experts
        I = moe.intermediate_per_rank
        H = moe.hidden_size
        assert H % 8 == 0 and I % 8 == 0 and H % group == 0 \
            and I % group == 0
        dt = moe.w13_weight.dtype
        dev = moe.w13_weight.device
        del moe.w13_weight, moe.w2_weight

        def mk(shape, dtype):
            return nn.Parameter(torch.empty(*shape, dtype=dtype,
                                            device=dev),
                                requires_grad=False)

        moe.w13_weight_packed = mk((E, 2 * I, H // 8), torch.int32)
        moe.w13_weight_scale = mk((E, 2 * I, H // group), dt)
        moe.w2_weight_packed = mk((E, H, I // 8), torch.int32)
        moe.w2_weight_scale = mk((E, H, I // group), dt)
        # w13: rows are the (gate|up) output dim — the existing row
        # loaders shard dim 0, packing is along K (dim 1): reuse as-is
        moe.w13_weight_packed.weight_loader = moe._load_w13
        moe.w13_weight_scale.weight_loader = moe._load_w13

        def load_w2_cols(param, loaded, expert_id, div, _moe=moe):
            lid = _moe._local_expert(expert_id)
            if lid is None:
                return
            cols = _moe.intermediate_per_rank // div
            if _moe.use_ep:
                shard = loaded
            else:
                from gllm_amd.parallel import get_tp_rank
                shard = loaded.narrow(1, get_tp_rank() * cols, cols)
            param.data[lid].copy_(shard)

        moe.w2_weight_packed.weight_loader = \
            lambda p, w, e, _f=load_w2_cols: _f(p, w, e, 8)
        moe.w2_weight_scale.weight_loader = \
            lambda p, w, e, _f=load_w2_cols: _f(p, w, e, group)
        moe.int4_packed = (4, group)
        n += 1
    return n
