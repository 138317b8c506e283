"""Megatron-style tensor-parallel linear layers.

Parity with the reference's layers/linear.py (Row/Column/Merged/QKV/
Replicated). GEMMs go through torch.nn.functional.linear → hipBLASLt on
ROCm (plain library GEMMs; the fused hot ops are the hand-written HIP
kernels in ops/). Weight sharding happens at load time via a
``weight_loader`` attribute attached to each parameter.
"""

from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from gllm_amd import ops
from gllm_amd.parallel import (get_tp_rank, get_tp_size,
                               tensor_parallel_all_reduce)


def _narrow_copy(param_data: torch.Tensor, loaded: torch.Tensor,
                 dim: int, rank: int, size: int):
    shard = loaded.shape[dim] // size
    param_data.copy_(loaded.narrow(dim, rank * shard, shard))


class LinearBase(nn.Module):
    def __init__(self, input_size: int, output_size: int, bias: bool,
                 params_dtype=None):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        dtype = params_dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(
            torch.empty(output_size, input_size, dtype=dtype),
            requires_grad=False)
        if bias:
            self.bias = nn.Parameter(torch.empty(output_size, dtype=dtype),
                                     requires_grad=False)
        else:
            self.register_parameter("bias", None)
        # set by quantization/{fp8,int4}.py converters
        self.fp8_block = None
        self.int4_cfg = None
        self._w_dq = None

    def _gemm(self, x, bias=None):
        """GEMM entry for all subclasses.

        fp8-block weights on GPU run NATIVE: per-token-group activation
        quant + the block-scale fp8 MFMA weight-streaming kernel at
        decode sizes (weights stay e4m3-resident, halving the decode
        weight stream); large-M prefill calls dequantize per call (no
        resident cache — prefill is compute-bound and bursty). CPU and
        int4 keep the dequant path."""
        w = self.weight
        if self.fp8_block is not None:
            native = (x.is_cuda and x.dim() == 2
                      and x.dtype == torch.bfloat16
                      and x.shape[1] % 128 == 0
                      and self.fp8_block == (128, 128)
                      and ops.has_kernels())
            if native:
                # measured crossover (profiles/): the weight-streaming
                # skinny wins only at M <= 64; above that the bf16
                # shadow + library GEMM is faster (M=256 gate_up:
                # 329 us skinny vs ~103 us library)
                if x.shape[0] <= 64:
                    return ops.fp8_linear(x, w, self.weight_scale_inv,
                                          self._bias32())
                return self._prefill_quant(x, bias)
            from gllm_amd.layers.quantization.fp8 import dequant_block_fp8
            if x.is_cuda:
                wd = dequant_block_fp8(w, self.weight_scale_inv,
                                       self.fp8_block, x.dtype)
                return ops.linear(x, wd, bias)
            if self._w_dq is None:
                self._w_dq = dequant_block_fp8(
                    w, self.weight_scale_inv, self.fp8_block,
                    x.dtype).to(x.device)
            w = self._w_dq
        elif self.int4_cfg is not None:
            native = (x.is_cuda and x.dim() == 2
                      and x.dtype == torch.bfloat16
                      and x.shape[1] % 256 == 0
                      and self.int4_cfg[1] == 128 and ops.has_kernels())
            if native:
                # packed-nibble streaming GEMMs (one-time canonical
                # repack; sbt = transposed (scale,bias) for the skinny)
                if getattr(self, "_i4_canon", None) is None:
                    from gllm_amd.layers.quantization.int4 import \
                        repack_canonical
                    wq4, sb, grp = repack_canonical(self)
                    sbt = sb.permute(1, 2, 0).contiguous()
                    self._i4_canon = (wq4.to(x.device), sb.to(x.device),
                                      sbt.to(x.device), grp)
                wq4, sb, sbt, grp = self._i4_canon
                if x.shape[0] <= 64:
                    return ops.int4_linear(x, wq4, sbt, grp,
                                           self._bias32())
                return self._prefill_quant(x, bias)
            if x.is_cuda:
                from gllm_amd.layers.quantization.int4 import dequant_layer
                return ops.linear(x, dequant_layer(self, x.dtype), bias)
            if self._w_dq is None:
                from gllm_amd.layers.quantization.int4 import dequant_layer
                self._w_dq = dequant_layer(self, x.dtype).to(x.device)
            w = self._w_dq
        return ops.linear(x, w, bias)

    def _bias32(self):
        """fp32 bias cached once (the skinny epilogues add fp32)."""
        if self.bias is None:
            return None
        if getattr(self, "_bias_f32", None) is None:
            self._bias_f32 = self.bias.float()
        return self._bias_f32

    def _prefill_quant(self, x, bias):
        """Quantized prefill (M > 256), three modes via
        GLLM_QUANT_PREFILL:

        - "cache" (default): dequantize ONCE into a resident bf16 copy
          and run the library GEMM. MI355X carries 288 GB HBM3E — for a
          32B model the bf16 shadow costs 64 GB and buys bf16-identical
          prefill speed, while decode keeps streaming the quantized
          weights (where the bandwidth win lives). Built during the
          profile run, so KV sizing accounts for it.
        - "stream": identity-routed grouped quant GEMM — no shadow
          copy; slower (dense M/64 weight re-reads) but memory-lean.
        - "dequant": dequantize per call, nothing cached.
        """
        import os
        from gllm_amd import ops
        mode = os.environ.get("GLLM_QUANT_PREFILL", "cache")
        if mode == "stream" and ops.has_kernels():
            if self.fp8_block is not None:
                return ops.fp8_prefill_linear(x, self.weight,
                                              self.weight_scale_inv,
                                              self.bias)
            wq4, sb, _, _ = self._i4_canon
            return ops.int4_prefill_linear(x, wq4, sb, self.bias)
        if mode == "cache":
            if self._w_dq is None:
                if torch.cuda.is_current_stream_capturing():
                    raise RuntimeError(
                        "quant prefill cache built during graph capture")
                self._w_dq = self._dequant_full(x.dtype).to(x.device)
            return ops.linear(x, self._w_dq, bias)
        return ops.linear(x, self._dequant_full(x.dtype), bias)

    def _dequant_full(self, dtype):
        if self.fp8_block is not None:
            from gllm_amd.layers.quantization.fp8 import dequant_block_fp8
            return dequant_block_fp8(self.weight, self.weight_scale_inv,
                                     self.fp8_block, dtype)
        from gllm_amd.layers.quantization.int4 import dequant_layer
        return dequant_layer(self, dtype)


class ReplicatedLinear(LinearBase):
    def __init__(self, input_size, output_size, bias=False, params_dtype=None):
        super().__init__(input_size, output_size, bias, params_dtype)
        self.weight.weight_loader = self._load
        if self.bias is not None:
            self.bias.weight_loader = self._load

    @staticmethod
    def _load(param, loaded):
        param.data.copy_(loaded)

    def forward(self, x):
        return self._gemm(x, self.bias)


class ColumnParallelLinear(LinearBase):
    """Y = XW^T with W sharded along its output dim; no collective."""

    def __init__(self, input_size, output_size, bias=False, params_dtype=None,
                 gather_output: bool = False):
        tp = get_tp_size()
        assert output_size % tp == 0, (output_size, tp)
        super().__init__(input_size, output_size // tp, bias, params_dtype)
        self.full_output_size = output_size
        self.gather_output = gather_output
        self.weight.weight_loader = self._load
        if self.bias is not None:
            self.bias.weight_loader = self._load

    @staticmethod
    def _load(param, loaded):
        _narrow_copy(param.data, loaded, 0, get_tp_rank(), get_tp_size())

    def forward(self, x):
        out = self._gemm(x, self.bias)
        if self.gather_output:
            from gllm_amd.parallel import tensor_parallel_all_gather
            out = tensor_parallel_all_gather(out, dim=-1)
        return out


class MergedColumnParallelLinear(LinearBase):
    """Several column-parallel projections fused into one GEMM
    (gate_proj + up_proj). Checkpoints hold them separately; the loader
    places each sub-weight's TP shard into its slice."""

    def __init__(self, input_size, output_sizes: List[int], bias=False,
                 params_dtype=None):
        tp = get_tp_size()
        for s in output_sizes:
            assert s % tp == 0
        self.output_sizes = output_sizes
        super().__init__(input_size, sum(output_sizes) // tp, bias,
                         params_dtype)
        self.weight.weight_loader = self._load
        if self.bias is not None:
            self.bias.weight_loader = self._load

    def _load(self, param, loaded, shard_id: int):
        tp_rank, tp = get_tp_rank(), get_tp_size()
        offset = sum(self.output_sizes[:shard_id]) // tp
        size = self.output_sizes[shard_id] // tp
        param.data.narrow(0, offset, size).copy_(
            loaded.narrow(0, tp_rank * size, size))

    def forward(self, x):
        return self._gemm(x, self.bias)


class QKVParallelLinear(LinearBase):
    """Fused qkv projection, sharded by head. Handles GQA where kv heads
    replicate across TP ranks when tp > num_kv_heads."""

    def __init__(self, hidden_size, head_dim, num_heads, num_kv_heads,
                 bias=False, params_dtype=None):
        tp = get_tp_size()
        assert num_heads % tp == 0
        self.head_dim = head_dim
        self.total_num_heads = num_heads
        self.total_num_kv_heads = num_kv_heads
        self.num_heads = num_heads // tp
        if num_kv_heads >= tp:
            assert num_kv_heads % tp == 0
            self.num_kv_heads = num_kv_heads // tp
            self.kv_replication = 1
        else:
            assert tp % num_kv_heads == 0
            self.num_kv_heads = 1
            self.kv_replication = tp // num_kv_heads
        out_per_rank = (self.num_heads + 2 * self.num_kv_heads) * head_dim
        super().__init__(hidden_size, out_per_rank, bias, params_dtype)
        self.q_size = self.num_heads * head_dim
        self.kv_size = self.num_kv_heads * head_dim
        self.weight.weight_loader = self._load
        if self.bias is not None:
            self.bias.weight_loader = self._load

    def _load(self, param, loaded, shard_id: str):
        tp_rank = get_tp_rank()
        if shard_id == "q":
            offset, size, src_rank = 0, self.q_size, tp_rank
        elif shard_id == "k":
            offset, size = self.q_size, self.kv_size
            src_rank = tp_rank // self.kv_replication
        else:
            offset, size = self.q_size + self.kv_size, self.kv_size
            src_rank = tp_rank // self.kv_replication
        param.data.narrow(0, offset, size).copy_(
            loaded.narrow(0, src_rank * size, size))

    def forward(self, x):
        out = self._gemm(x, self.bias)
        return out.split([self.q_size, self.kv_size, self.kv_size], dim=-1)


class RowParallelLinear(LinearBase):
    """Y = XW^T with W sharded along its input dim; output all-reduce
    across TP (2x per decoder layer: o_proj + down_proj)."""

    def __init__(self, input_size, output_size, bias=False, params_dtype=None,
                 reduce_results: bool = True):
        tp = get_tp_size()
        assert input_size % tp == 0
        super().__init__(input_size // tp, output_size, bias, params_dtype)
        self.reduce_results = reduce_results
        self.weight.weight_loader = self._load_w
        if self.bias is not None:
            self.bias.weight_loader = self._load_b

    @staticmethod
    def _load_w(param, loaded):
        _narrow_copy(param.data, loaded, 1, get_tp_rank(), get_tp_size())

    @staticmethod
    def _load_b(param, loaded):
        param.data.copy_(loaded)

    def forward(self, x):
        out = self._gemm(x)
        if self.reduce_results:
            out = tensor_parallel_all_reduce(out)
        # bias applied once, after the reduction
        if self.bias is not None:
            out = out + self.bias
        return out
