"""ChatGLM3 / GLM-4 (reference: models/chatglm.py).

Differences from the llama family: fused query_key_value checkpoint
weight (+bias), partial rotary (rot_dim = head_dim/2, interleaved),
fused dense_h_to_4h (gate|up already concatenated), "transformer.*"
checkpoint naming.
"""

from typing import Iterable, Tuple

import torch
import torch.nn as nn

from gllm_amd.layers.activation import SiluAndMul
from gllm_amd.layers.attention import Attention
from gllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from gllm_amd.layers.layernorm import RMSNorm
from gllm_amd.layers.linear import (MergedColumnParallelLinear,
                                    QKVParallelLinear, RowParallelLinear)
from gllm_amd.layers.rotary import get_rope
from gllm_amd.parallel import get_tp_rank, get_tp_size
from gllm_amd.runtime.forward_context import ForwardContext


class GLMAttention(nn.Module):
    def __init__(self, cfg, layer_idx: int, dtype=None):
        super().__init__()
        hidden = cfg.hidden_size
        self.total_heads = cfg.num_attention_heads
        self.total_kv_heads = getattr(cfg, "multi_query_group_num",
                                      self.total_heads) \
            if getattr(cfg, "multi_query_attention", False) \
            else self.total_heads
        self.head_dim = getattr(cfg, "kv_channels",
                                hidden // self.total_heads)
        bias = getattr(cfg, "add_qkv_bias",
                       getattr(cfg, "add_bias_linear", False))
        self.qkv_proj = QKVParallelLinear(
            hidden, self.head_dim, self.total_heads, self.total_kv_heads,
            bias=bias, params_dtype=dtype)
        self.o_proj = RowParallelLinear(
            self.total_heads * self.head_dim, hidden,
            bias=getattr(cfg, "add_bias_linear", False), params_dtype=dtype)
        rope_ratio = getattr(cfg, "rope_ratio", 1.0)
        max_pos = getattr(cfg, "seq_length", 8192)
        # GLM rope: half the head dim, interleaved pairs
        self.rotary_emb = get_rope(self.head_dim, self.head_dim // 2,
                                   max_pos, 10000.0 * rope_ratio,
                                   is_neox=False)
        self.attn = Attention(
            layer_idx, self.qkv_proj.num_heads, self.qkv_proj.num_kv_heads,
            self.head_dim, self.head_dim ** -0.5)

    def forward(self, positions, hidden, fctx):
        q, k, v = self.qkv_proj(hidden)
        q, k = self.rotary_emb(positions, q, k)
        o = self.attn(q, k, v, fctx)
        return self.o_proj(o)


class GLMMLP(nn.Module):
    def __init__(self, cfg, dtype=None):
        super().__init__()
        ffn = cfg.ffn_hidden_size
        # dense_h_to_4h holds [gate | up] pre-concatenated
        self.gate_up_proj = MergedColumnParallelLinear(
            cfg.hidden_size, [ffn, ffn],
            bias=getattr(cfg, "add_bias_linear", False), params_dtype=dtype)
        self.down_proj = RowParallelLinear(
            ffn, cfg.hidden_size,
            bias=getattr(cfg, "add_bias_linear", False), params_dtype=dtype)
        self.act_fn = SiluAndMul()

    def forward(self, x):
        return self.down_proj(self.act_fn(self.gate_up_proj(x)))


class GLMDecoderLayer(nn.Module):
    def __init__(self, cfg, layer_idx: int, dtype=None):
        super().__init__()
        eps = getattr(cfg, "layernorm_epsilon", 1e-5)
        self.self_attn = GLMAttention(cfg, layer_idx, dtype=dtype)
        self.mlp = GLMMLP(cfg, dtype=dtype)
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


class ChatGLMForCausalLM(nn.Module):
    def __init__(self, cfg, engine_config):
        super().__init__()
        self.cfg = cfg
        self.engine_config = engine_config
        dtype = engine_config.torch_dtype()
        num_layers = getattr(cfg, "num_layers", None) or \
            cfg.num_hidden_layers
        from gllm_amd.parallel import get_pp_rank, is_first_pp_rank, \
            is_last_pp_rank
        self.layer_start, self.layer_end = engine_config.pp_layer_range(
            get_pp_rank(), num_layers)
        self.is_first_stage = is_first_pp_rank()
        self.is_last_stage = is_last_pp_rank()
        vocab = getattr(cfg, "padded_vocab_size", None) or cfg.vocab_size
        self.vocab_size = vocab
        if self.is_first_stage:
            self.embed_tokens = VocabParallelEmbedding(
                vocab, cfg.hidden_size, params_dtype=dtype)
        self.layers = nn.ModuleList([
            GLMDecoderLayer(cfg, local, dtype=dtype)
            for local in range(self.layer_end - self.layer_start)])
        if self.is_last_stage:
            self.norm = RMSNorm(cfg.hidden_size,
                                getattr(cfg, "layernorm_epsilon", 1e-5))
            self.lm_head = ParallelLMHead(vocab, cfg.hidden_size,
                                          params_dtype=dtype)

    @property
    def num_local_layers(self):
        return self.layer_end - self.layer_start

    @property
    def kv_geometry(self):
        a = self.layers[0].self_attn
        return a.qkv_proj.num_kv_heads, a.head_dim

    def forward(self, input_ids, positions, fctx: ForwardContext,
                hidden_states=None, residual=None):
        if self.is_first_stage:
            hidden_states = self.embed_tokens(input_ids)
            residual = None
        for layer in self.layers:
            hidden_states, residual = layer(positions, hidden_states,
                                            residual, fctx)
        if self.is_last_stage:
            hidden_states, _ = self.norm(hidden_states, residual)
            return hidden_states, None
        return hidden_states, residual

    def compute_logits(self, hidden_states, fctx):
        rows = hidden_states
        if fctx.logits_indices is not None:
            rows = hidden_states.index_select(0, fctx.logits_indices)
        return self.lm_head(rows)

    # ------------------------------------------------------------------
    def load_weights(self, weights: Iterable[Tuple[str, torch.Tensor]]):
        params = dict(self.named_parameters())
        head_dim = self.layers[0].self_attn.head_dim if self.layers else 0
        q_total = self.cfg.num_attention_heads * head_dim
        kv_total = (self.layers[0].self_attn.total_kv_heads * head_dim
                    if self.layers else 0)
        for name, w in weights:
            name = name.replace("transformer.", "", 1)
            if name.startswith("embedding."):
                if self.is_first_stage:
                    p = params["embed_tokens.weight"]
                    p.weight_loader(p, w)
                continue
            if name.startswith("output_layer"):
                if self.is_last_stage:
                    p = params["lm_head.weight"]
                    p.weight_loader(p, w)
                continue
            if name.startswith("encoder.final_layernorm"):
                if self.is_last_stage:
                    params["norm.weight"].data.copy_(w)
                continue
            if name == "rotary_pos_emb.inv_freq":
                continue
            if not name.startswith("encoder.layers."):
                continue
            parts = name.split(".")
            g_idx = int(parts[2])
            if not (self.layer_start <= g_idx < self.layer_end):
                continue
            local = g_idx - self.layer_start
            rest = ".".join(parts[3:])
            pre = f"layers.{local}"
            if rest.startswith("self_attention.query_key_value"):
                kind = rest.split(".")[-1]       # weight | bias
                p = params[f"{pre}.self_attn.qkv_proj.{kind}"]
                qw = w.narrow(0, 0, q_total)
                kw = w.narrow(0, q_total, kv_total)
                vw = w.narrow(0, q_total + kv_total, kv_total)
                p.weight_loader(p, qw, "q")
                p.weight_loader(p, kw, "k")
                p.weight_loader(p, vw, "v")
            elif rest.startswith("self_attention.dense"):
                kind = rest.split(".")[-1]
                p = params[f"{pre}.self_attn.o_proj.{kind}"]
                p.weight_loader(p, w)
            elif rest.startswith("mlp.dense_h_to_4h"):
                kind = rest.split(".")[-1]
                p = params[f"{pre}.mlp.gate_up_proj.{kind}"]
                half = w.shape[0] // 2
                p.weight_loader(p, w.narrow(0, 0, half), 0)
                p.weight_loader(p, w.narrow(0, half, half), 1)
            elif rest.startswith("mlp.dense_4h_to_h"):
                kind = rest.split(".")[-1]
                p = params[f"{pre}.mlp.down_proj.{kind}"]
                p.weight_loader(p, w)
            elif rest.startswith("input_layernorm"):
                params[f"{pre}.input_layernorm.weight"].data.copy_(w)
            elif rest.startswith("post_attention_layernorm"):
                params[f"{pre}.post_attention_layernorm.weight"].data.copy_(w)
