"""Dense decoder family: Llama-2/3, Qwen2/2.5, Qwen3 (qk-norm).

Parity targets: reference models/{llama,qwen2,qwen3}.py. One
parameterized implementation: the three architectures differ only in
attention bias, qk-norm, tied embeddings and rope scaling.

PP contract: stage 0 embeds input_ids; stages >0 receive
(hidden_states, residual) from the previous stage; the last stage norms
and computes logits via ``compute_logits`` (last-token rows selected by
``fctx.logits_indices``, the reference's query_start_loc-1 gather,
models/qwen2.py:211-214).
"""

from typing import Iterable, List, Optional, Tuple

import torch
import torch.nn as nn

from gllm_amd.layers.activation import SiluAndMul
from gllm_amd.layers.attention import Attention
from gllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from gllm_amd.layers.layernorm import RMSNorm
from gllm_amd.layers.linear import (MergedColumnParallelLinear,
                                    QKVParallelLinear, RowParallelLinear)
from gllm_amd.layers.rotary import get_rope
from gllm_amd.runtime.forward_context import ForwardContext


class DenseMLP(nn.Module):
    def __init__(self, hidden_size: int, intermediate_size: int, dtype=None):
        super().__init__()
        self.gate_up_proj = MergedColumnParallelLinear(
            hidden_size, [intermediate_size, intermediate_size],
            params_dtype=dtype)
        self.down_proj = RowParallelLinear(intermediate_size, hidden_size,
                                           params_dtype=dtype)
        self.act_fn = SiluAndMul()

    def forward(self, x):
        return self.down_proj(self.act_fn(self.gate_up_proj(x)))


class DenseAttention(nn.Module):
    def __init__(self, cfg, layer_idx: int, *, qkv_bias: bool, qk_norm: bool,
                 dtype=None):
        super().__init__()
        hidden = cfg.hidden_size
        sw = getattr(cfg, "sliding_window", None)
        use_sw = getattr(cfg, "use_sliding_window", sw is not None)
        self.sliding_window = int(sw) if (sw and use_sw) else 0
        self.total_heads = cfg.num_attention_heads
        self.total_kv_heads = getattr(cfg, "num_key_value_heads",
                                      self.total_heads)
        self.head_dim = getattr(cfg, "head_dim",
                                hidden // self.total_heads) or \
            hidden // self.total_heads
        self.qkv_proj = QKVParallelLinear(
            hidden, self.head_dim, self.total_heads, self.total_kv_heads,
            bias=qkv_bias, params_dtype=dtype)
        self.o_proj = RowParallelLinear(
            self.total_heads * self.head_dim, hidden, params_dtype=dtype)
        rope_theta = getattr(cfg, "rope_theta", 10000.0)
        max_pos = getattr(cfg, "max_position_embeddings", 32768)
        rope_scaling = getattr(cfg, "rope_scaling", None)
        self.rotary_emb = get_rope(self.head_dim, self.head_dim, max_pos,
                                   rope_theta, is_neox=True,
                                   rope_scaling=rope_scaling)
        self.qk_norm = qk_norm
        if qk_norm:
            eps = getattr(cfg, "rms_norm_eps", 1e-6)
            self.q_norm = RMSNorm(self.head_dim, eps)
            self.k_norm = RMSNorm(self.head_dim, eps)
        self.attn = Attention(
            layer_idx, self.qkv_proj.num_heads, self.qkv_proj.num_kv_heads,
            self.head_dim, self.head_dim ** -0.5,
            sliding_window=self.sliding_window)

    def forward(self, positions, hidden, fctx: ForwardContext):
        q, k, v = self.qkv_proj(hidden)
        if self.qk_norm:
            T = q.shape[0]
            q = self.q_norm(
                q.contiguous().view(T, -1, self.head_dim)).view(T, -1)
            k = self.k_norm(
                k.contiguous().view(T, -1, self.head_dim)).view(T, -1)
        q, k = self.rotary_emb(positions, q, k)
        o = self.attn(q, k, v, fctx)
        return self.o_proj(o)


class DenseDecoderLayer(nn.Module):
    def __init__(self, cfg, layer_idx: int, *, qkv_bias: bool, qk_norm: bool,
                 dtype=None):
        super().__init__()
        eps = getattr(cfg, "rms_norm_eps", 1e-6)
        self.self_attn = DenseAttention(cfg, layer_idx, qkv_bias=qkv_bias,
                                        qk_norm=qk_norm, dtype=dtype)
        self.mlp = DenseMLP(cfg.hidden_size, cfg.intermediate_size,
                            dtype=dtype)
        self.input_layernorm = RMSNorm(cfg.hidden_size, eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, eps)

    def forward(self, positions, hidden, residual, fctx):
        if residual is None:
            residual = hidden
            hidden = self.input_layernorm(hidden)
        else:
            hidden, residual = self.input_layernorm(hidden, residual)
        hidden = self.self_attn(positions, hidden, fctx)
        hidden, residual = self.post_attention_layernorm(hidden, residual)
        hidden = self.mlp(hidden)
        return hidden, residual


class LlamaFamilyForCausalLM(nn.Module):
    # subclass knobs
    qkv_bias = False
    qk_norm = False

    # stacked-weight mapping: (fused param name, ckpt name, shard id)
    stacked_params_mapping = [
        ("qkv_proj", "q_proj", "q"),
        ("qkv_proj", "k_proj", "k"),
        ("qkv_proj", "v_proj", "v"),
        ("gate_up_proj", "gate_proj", 0),
        ("gate_up_proj", "up_proj", 1),
    ]

    def __init__(self, cfg, engine_config):
        super().__init__()
        self.cfg = cfg
        self.engine_config = engine_config
        dtype = engine_config.torch_dtype()
        num_layers = cfg.num_hidden_layers
        from gllm_amd.parallel import get_pp_rank, is_first_pp_rank, \
            is_last_pp_rank
        self.layer_start, self.layer_end = engine_config.pp_layer_range(
            get_pp_rank(), num_layers)
        self.is_first_stage = is_first_pp_rank()
        self.is_last_stage = is_last_pp_rank()

        if self.is_first_stage:
            self.embed_tokens = VocabParallelEmbedding(
                cfg.vocab_size, cfg.hidden_size, params_dtype=dtype)
        self.layers = nn.ModuleList([
            DenseDecoderLayer(cfg, local_idx, qkv_bias=self.qkv_bias,
                              qk_norm=self.qk_norm, dtype=dtype)
            for local_idx in range(self.layer_end - self.layer_start)])
        if self.is_last_stage:
            self.norm = RMSNorm(cfg.hidden_size,
                                getattr(cfg, "rms_norm_eps", 1e-6))
            self.lm_head = ParallelLMHead(cfg.vocab_size, cfg.hidden_size,
                                          params_dtype=dtype)
            if getattr(cfg, "tie_word_embeddings", False):
                if self.is_first_stage:
                    self.lm_head.tie_to(self.embed_tokens)
                # PP>1 with tied embeddings: the last stage loads
                # embed_tokens.weight into lm_head (handled in load_weights)

    @property
    def num_local_layers(self) -> int:
        return self.layer_end - self.layer_start

    def forward(self, input_ids: Optional[torch.Tensor],
                positions: torch.Tensor, fctx: ForwardContext,
                hidden_states: Optional[torch.Tensor] = None,
                residual: Optional[torch.Tensor] = None):
        if self.is_first_stage:
            hidden_states = self.embed_tokens(input_ids)
            if fctx.mm_rows is not None:
                # vision-embedding merge (reference embed_input_ids /
                # disagg_set_embedding, model_runner.py:931-994)
                hidden_states = hidden_states.index_copy(
                    0, fctx.mm_rows,
                    fctx.mm_embeds.to(hidden_states.dtype))
            residual = None
        ds = getattr(fctx, "mm_deepstack", None)
        for li, layer in enumerate(self.layers):
            hidden_states, residual = layer(positions, hidden_states,
                                            residual, fctx)
            if ds is not None:
                # Qwen3-VL deepstack: level li of the multiscale vision
                # features is ADDED at the image rows after decoder
                # layer li (reference qwen3_vl.py Qwen3LLMModel.forward)
                g = self.layer_start + li
                H = hidden_states.shape[-1]
                if g * H < ds.shape[1]:
                    hidden_states = hidden_states.index_add(
                        0, fctx.mm_rows,
                        ds[:, g * H:(g + 1) * H].to(hidden_states.dtype))
        if self.is_last_stage:
            hidden_states, _ = self.norm(hidden_states, residual)
            return hidden_states, None
        return hidden_states, residual

    def compute_logits(self, hidden_states: torch.Tensor,
                       fctx: ForwardContext) -> torch.Tensor:
        rows = hidden_states
        if fctx.logits_indices is not None:
            rows = hidden_states.index_select(0, fctx.logits_indices)
        return self.lm_head(rows)

    # ------------------------------------------------------------------
    def load_weights(self, weights: Iterable[Tuple[str, torch.Tensor]]):
        params = dict(self.named_parameters())
        tied = getattr(self.cfg, "tie_word_embeddings", False)
        loaded = set()
        for name, w in weights:
            name = name.replace("model.", "", 1) if name.startswith("model.") \
                else name
            if name.startswith("layers."):
                parts = name.split(".")
                g_idx = int(parts[1])
                if not (self.layer_start <= g_idx < self.layer_end):
                    continue
                parts[1] = str(g_idx - self.layer_start)
                name = ".".join(parts)
            elif name.startswith("embed_tokens"):
                if self.is_first_stage:
                    p = params["embed_tokens.weight"]
                    p.weight_loader(p, w)
                    loaded.add("embed_tokens.weight")
                if tied and self.is_last_stage and not self.is_first_stage:
                    p = params["lm_head.weight"]
                    p.weight_loader(p, w)
                    loaded.add("lm_head.weight")
                continue
            elif name.startswith("lm_head"):
                if self.is_last_stage and not tied:
                    p = params["lm_head.weight"]
                    p.weight_loader(p, w)
                    loaded.add("lm_head.weight")
                continue
            elif name.startswith("norm."):
                if not self.is_last_stage:
                    continue
            # stacked params
            hit = False
            for fused, ckpt, shard in self.stacked_params_mapping:
                if ckpt in name:
                    tgt = name.replace(ckpt, fused)
                    if tgt in params:
                        p = params[tgt]
                        p.weight_loader(p, w, shard)
                        loaded.add(tgt)
                        hit = True
                    break
            if hit:
                continue
            if name in params:
                p = params[name]
                if hasattr(p, "weight_loader"):
                    p.weight_loader(p, w)
                else:
                    p.data.copy_(w)
                loaded.add(name)
        if tied and self.is_first_stage and self.is_last_stage:
            loaded.add("lm_head.weight")
        missing = set(params) - loaded
        if missing:
            from gllm_amd.logger import logger
            logger.warning("weights not found in checkpoint: %s",
                           sorted(missing)[:8])


class LlamaForCausalLM(LlamaFamilyForCausalLM):
    qkv_bias = False
    qk_norm = False


class Qwen2ForCausalLM(LlamaFamilyForCausalLM):
    qkv_bias = True
    qk_norm = False


class Qwen3ForCausalLM(LlamaFamilyForCausalLM):
    qkv_bias = False
    qk_norm = True


class MistralForCausalLM(LlamaFamilyForCausalLM):
    """Llama structure + sliding-window attention (read from config)."""
    qkv_bias = False
    qk_norm = False
