"""fp8 (OCP e4m3) block-quantized weights + per-token-group activation
quantization.

Parity target: reference layers/quantization/fp8.py (Fp8LinearMethod
backend chain + per-token-group quant kernels + UE8M0 power-of-2 scale
mode for DeepSeek-V3.2). MI355X stance: CDNA4 has native fp8 MFMA
(V_MFMA_F32_16X16X32_FP8) at 2x the bf16 rate — the round-2 item is a
block-scale fp8 MFMA GEMM kernel consuming exactly the (weight fp8,
weight_scale_inv, per-token-group activation scales) layout prepared
here. Round-1 execution path: weights are stored fp8-blocked exactly as
the checkpoint ships them, and DEQUANTIZED ONCE (lazily, cached) into
the compute dtype for hipBLASLt/torch GEMMs — numerically identical to
the reference's Triton w8a8 path up to the activation quant it skips,
and bit-exact with any correctly sharded load (the TP parity test in
tests/test_fp8_cpu.py).

Checkpoint format (DeepSeek / Qwen fp8 releases):
  quantization_config = {"quant_method": "fp8",
                         "weight_block_size": [128, 128], ...}
  per quantized linear:  <name>.weight            float8_e4m3fn [N, K]
                         <name>.weight_scale_inv  fp32 [ceil(N/bs0),
                                                        ceil(K/bs1)]
  dequant: w[i, j] = fp8[i, j] * scale_inv[i // bs0, j // bs1]
"""

import math
from typing import Tuple

import torch
import torch.nn as nn

FP8_MAX = 448.0  # e4m3 finite max
FP8_DTYPE = torch.float8_e4m3fn


# ------------------------------------------------------------ quant math
def per_token_group_quant_fp8(x: torch.Tensor, group_size: int = 128,
                              ue8m0: bool = False
                              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Activation quant (reference fp8.py:556-676): per (token, K-group)
    e4m3 + fp32 scales [T, K/group]. ``ue8m0`` rounds scales UP to a
    power of two (DeepSeek-V3.2 scale_fmt)."""
    T, K = x.shape
    assert K % group_size == 0, (K, group_size)
    g = x.float().view(T, K // group_size, group_size)
    amax = g.abs().amax(dim=-1, keepdim=True).clamp_min(1e-4)
    scale = amax / FP8_MAX
    if ue8m0:
        scale = torch.exp2(torch.ceil(torch.log2(scale)))
    q = (g / scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return q.view(T, K), scale.squeeze(-1)


def block_quant_fp8(w: torch.Tensor, block: Tuple[int, int] = (128, 128),
                    ue8m0: bool = False
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Weight quant: per (bs0 x bs1) block e4m3 + fp32 scale_inv grid.
    Used to produce checkpoints in tests; real checkpoints ship
    pre-quantized."""
    N, K = w.shape
    bs0, bs1 = block
    n_blk, k_blk = math.ceil(N / bs0), math.ceil(K / bs1)
    q = torch.empty(N, K, dtype=FP8_DTYPE)
    scale_inv = torch.empty(n_blk, k_blk, dtype=torch.float32)
    wf = w.float()
    for i in range(n_blk):
        for j in range(k_blk):
            blk = wf[i * bs0:(i + 1) * bs0, j * bs1:(j + 1) * bs1]
            amax = blk.abs().max().clamp_min(1e-4)
            s = amax / FP8_MAX
            if ue8m0:
                s = torch.exp2(torch.ceil(torch.log2(s)))
            scale_inv[i, j] = s
            q[i * bs0:(i + 1) * bs0, j * bs1:(j + 1) * bs1] = \
                (blk / s).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return q, scale_inv


def dequant_block_fp8(wq: torch.Tensor, scale_inv: torch.Tensor,
                      block: Tuple[int, int],
                      dtype: torch.dtype) -> torch.Tensor:
    """w[i, j] = fp8[i, j] * scale_inv[i // bs0, j // bs1]."""
    N, K = wq.shape
    bs0, bs1 = block
    s = scale_inv.repeat_interleave(bs0, dim=0)[:N] \
        .repeat_interleave(bs1, dim=1)[:, :K]
    return (wq.float() * s).to(dtype)


# ----------------------------------------------------- layer conversion
# Linear projections that fp8 checkpoints quantize (embeddings, lm_head,
# router gates, norms and the DSA weights_proj stay bf16). MoE expert
# tensors (w13/w2) convert through convert_moe_to_fp8 below.
_QUANT_SUFFIXES = (
    "q_proj", "k_proj", "v_proj", "o_proj", "qkv_proj",
    "gate_proj", "up_proj", "down_proj", "gate_up_proj",
    "q_a_proj", "q_b_proj", "kv_a_proj_with_mqa", "kv_b_proj",
    "fused_qkv_a_proj", "wq_b", "wk",
)


def _scale_shape(out_partition: int, in_partition: int, block):
    return (math.ceil(out_partition / block[0]),
            math.ceil(in_partition / block[1]))


def _make_scale_loader(layer, block):
    """weight_scale_inv loader mirroring the layer's weight sharding in
    BLOCK space. Requires per-rank shard sizes to be block-aligned when
    tp > 1 (true for all real fp8 checkpoints: dims are multiples of
    128)."""
    from gllm_amd.layers.linear import (ColumnParallelLinear,
                                        MergedColumnParallelLinear,
                                        QKVParallelLinear,
                                        ReplicatedLinear,
                                        RowParallelLinear)
    from gllm_amd.parallel import get_tp_rank, get_tp_size
    bs0, bs1 = block

    def _blocks(n, bs):
        assert n % bs == 0, \
            f"fp8 TP shard ({n}) not aligned to block size {bs}"
        return n // bs

    if isinstance(layer, MergedColumnParallelLinear):
        def load(param, loaded, shard_id: int):
            tp_rank, tp = get_tp_rank(), get_tp_size()
            size = layer.output_sizes[shard_id] // tp
            off_b = _blocks(sum(layer.output_sizes[:shard_id]) // tp, bs0)
            size_b = _blocks(size, bs0)
            param.data.narrow(0, off_b, size_b).copy_(
                loaded.narrow(0, tp_rank * size_b, size_b))
    elif isinstance(layer, QKVParallelLinear):
        def load(param, loaded, shard_id: str):
            tp_rank = get_tp_rank()
            if shard_id == "q":
                off, size, src = 0, layer.q_size, tp_rank
            elif shard_id == "k":
                off, size = layer.q_size, layer.kv_size
                src = tp_rank // layer.kv_replication
            else:
                off, size = layer.q_size + layer.kv_size, layer.kv_size
                src = tp_rank // layer.kv_replication
            off_b, size_b = _blocks(off, bs0) if off else 0, \
                _blocks(size, bs0)
            param.data.narrow(0, off_b, size_b).copy_(
                loaded.narrow(0, src * size_b, size_b))
    elif isinstance(layer, RowParallelLinear):
        def load(param, loaded):
            tp_rank, tp = get_tp_rank(), get_tp_size()
            if tp == 1:
                param.data.copy_(loaded)
                return
            _blocks(layer.weight.shape[1], bs1)  # shard must align
            size_b = param.shape[1]
            param.data.copy_(loaded.narrow(1, tp_rank * size_b, size_b))
    elif isinstance(layer, ColumnParallelLinear):
        def load(param, loaded):
            tp_rank, tp = get_tp_rank(), get_tp_size()
            if tp == 1:
                param.data.copy_(loaded)
                return
            _blocks(layer.weight.shape[0], bs0)  # shard must align
            size_b = param.shape[0]
            param.data.copy_(loaded.narrow(0, tp_rank * size_b, size_b))
    else:
        assert isinstance(layer, ReplicatedLinear), type(layer)

        def load(param, loaded):
            param.data.copy_(loaded)
    return load


def convert_linear_to_fp8(layer, block: Tuple[int, int]) -> None:
    """Swap a LinearBase's bf16 weight for (fp8 weight, fp32
    weight_scale_inv). The existing weight_loader keeps working — fp8
    checkpoint tensors shard through the same narrow/copy paths."""
    old = layer.weight
    neww = nn.Parameter(torch.empty(old.shape, dtype=FP8_DTYPE),
                        requires_grad=False)
    neww.weight_loader = old.weight_loader
    layer.weight = neww
    scale = nn.Parameter(
        torch.ones(_scale_shape(old.shape[0], old.shape[1], block),
                   dtype=torch.float32), requires_grad=False)
    scale.weight_loader = _make_scale_loader(layer, block)
    layer.weight_scale_inv = scale
    layer.fp8_block = tuple(block)
    layer._w_dq = None


def convert_moe_to_fp8(moe, block: Tuple[int, int]) -> None:
    """Swap a FusedMoE's stacked expert weights for fp8 + per-expert
    block scale grids (DeepSeek-V3 style checkpoints quantize every
    routed expert; the router gate stays high precision). Checkpoint
    tensors stay per-expert ([I, H] gate/up, [H, I] down; scale grids in
    128-block space), so the scale loaders mirror the weight loaders'
    stacking and TP narrowing in BLOCK space — which needs the per-rank
    intermediate dim to be block-aligned (true for every real fp8 MoE
    release: moe_intermediate_size is a multiple of 128)."""
    from gllm_amd.parallel import get_tp_rank
    bs0, bs1 = block
    E = moe.num_local_experts
    I, H = moe.intermediate_per_rank, moe.hidden_size
    assert I % bs0 == 0 and I % bs1 == 0, (
        f"fp8 MoE needs the per-rank intermediate dim ({I}) aligned to "
        f"the quant block {block}")

    for wname in ("w13_weight", "w2_weight"):
        old = getattr(moe, wname)
        neww = nn.Parameter(torch.empty(old.shape, dtype=FP8_DTYPE),
                            requires_grad=False)
        neww.weight_loader = old.weight_loader
        # keep registration order/name: assign through setattr
        setattr(moe, wname, neww)

    w13_scale = nn.Parameter(
        torch.ones(E, 2 * (I // bs0), math.ceil(H / bs1),
                   dtype=torch.float32), requires_grad=False)
    w2_scale = nn.Parameter(
        torch.ones(E, math.ceil(H / bs0), I // bs1,
                   dtype=torch.float32), requires_grad=False)

    def load_w13_scale(param, loaded, expert_id: int, shard_id: int):
        lid = moe._local_expert(expert_id)
        if lid is None:
            return
        ib = I // bs0
        shard = loaded if moe.use_ep else \
            loaded.narrow(0, get_tp_rank() * ib, ib)
        param.data[lid].narrow(0, shard_id * ib, ib).copy_(shard)

    def load_w2_scale(param, loaded, expert_id: int):
        lid = moe._local_expert(expert_id)
        if lid is None:
            return
        ib = I // bs1
        shard = loaded if moe.use_ep else \
            loaded.narrow(1, get_tp_rank() * ib, ib)
        param.data[lid].copy_(shard)

    w13_scale.weight_loader = load_w13_scale
    w2_scale.weight_loader = load_w2_scale
    moe.w13_weight_scale_inv = w13_scale
    moe.w2_weight_scale_inv = w2_scale
    moe.fp8_block = tuple(block)
    moe._dq_cache = None


def convert_model_to_fp8(model, quant_config: dict) -> int:
    """Walk the model and convert every checkpoint-quantized linear and
    MoE expert bank. Returns the number of converted modules."""
    from gllm_amd.layers.linear import LinearBase
    from gllm_amd.layers.moe.layer import FusedMoE
    block = tuple(quant_config.get("weight_block_size") or (128, 128))
    n = 0
    for name, mod in model.named_modules():
        if isinstance(mod, FusedMoE):
            convert_moe_to_fp8(mod, block)
            n += 1
            continue
        if not isinstance(mod, LinearBase):
            continue
        leaf = name.rsplit(".", 1)[-1]
        if leaf in _QUANT_SUFFIXES:
            convert_linear_to_fp8(mod, block)
            n += 1
    return n
