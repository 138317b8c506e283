"""MoE decoder family: Mixtral, Qwen2-MoE, Qwen3-MoE.

Parity: reference models/{mixtral,qwen2_moe,qwen3_moe}.py.
Reuses the dense attention/norm stack from llama_family; swaps the MLP
for FusedMoE (+ shared expert for Qwen2-MoE).
"""

from typing import Iterable, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from gllm_amd.layers.activation import SiluAndMul
from gllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from gllm_amd.layers.layernorm import RMSNorm
from gllm_amd.layers.linear import (MergedColumnParallelLinear,
                                    ReplicatedLinear, RowParallelLinear)
from gllm_amd.layers.moe.layer import FusedMoE
from gllm_amd.models.llama_family import (DenseAttention, DenseMLP,
                                          LlamaFamilyForCausalLM)


class MoEBlock(nn.Module):
    """Router + experts (+ optional shared expert)."""

    def __init__(self, cfg, engine_config, *, num_experts, top_k,
                 moe_intermediate, shared_intermediate=0,
                 norm_topk_prob=True, dtype=None):
        super().__init__()
        self.gate = ReplicatedLinear(cfg.hidden_size, num_experts,
                                     params_dtype=dtype)
        self.experts = FusedMoE(num_experts, top_k, cfg.hidden_size,
                                moe_intermediate,
                                renormalize=norm_topk_prob,
                                use_ep=engine_config.use_ep,
                                params_dtype=dtype)
        if shared_intermediate:
            self.shared_expert = DenseMLP(cfg.hidden_size,
                                          shared_intermediate, dtype=dtype)
            self.shared_expert_gate = ReplicatedLinear(cfg.hidden_size, 1,
                                                       params_dtype=dtype)
        else:
            self.shared_expert = None

    def forward(self, x):
        router_logits = self.gate(x)
        out = self.experts(x, router_logits)
        if self.shared_expert is not None:
            shared = self.shared_expert(x)
            g = torch.sigmoid(self.shared_expert_gate(x).float()).to(x.dtype)
            out = out + shared * g
        return out


class MoEDecoderLayer(nn.Module):
    def __init__(self, cfg, engine_config, layer_idx, *, qkv_bias, qk_norm,
                 moe_kwargs, use_dense_mlp=False, dtype=None):
        super().__init__()
        eps = getattr(cfg, "rms_norm_eps", 1e-6)
        self.self_attn = DenseAttention(cfg, layer_idx, qkv_bias=qkv_bias,
                                        qk_norm=qk_norm, dtype=dtype)
        if use_dense_mlp:
            self.mlp = DenseMLP(cfg.hidden_size, cfg.intermediate_size,
                                dtype=dtype)
        else:
            self.mlp = MoEBlock(cfg, engine_config, dtype=dtype, **moe_kwargs)
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


class MoEFamilyForCausalLM(LlamaFamilyForCausalLM):
    """Shares PP-stage plumbing + dense weight paths with llama_family;
    adds expert weight routing."""

    qkv_bias = False
    qk_norm = False

    # (ckpt expert weight suffix, fused param, shard)
    expert_params_mapping = [
        ("gate_proj", "w13_weight", 0),
        ("up_proj", "w13_weight", 1),
        ("down_proj", "w2_weight", None),
        ("w1", "w13_weight", 0),
        ("w3", "w13_weight", 1),
        ("w2", "w2_weight", None),
    ]

    def moe_kwargs(self, cfg) -> dict:
        raise NotImplementedError

    def is_dense_layer(self, cfg, global_idx: int) -> bool:
        return False

    def __init__(self, cfg, engine_config):
        # build the dense skeleton then replace layers with MoE layers
        nn.Module.__init__(self)
        self.cfg = cfg
        self.engine_config = engine_config
        dtype = engine_config.torch_dtype()
        from gllm_amd.parallel import get_pp_rank, is_first_pp_rank, \
            is_last_pp_rank
        num_layers = cfg.num_hidden_layers
        self.layer_start, self.layer_end = engine_config.pp_layer_range(
            get_pp_rank(), num_layers)
        self.is_first_stage = is_first_pp_rank()
        self.is_last_stage = is_last_pp_rank()
        if self.is_first_stage:
            self.embed_tokens = VocabParallelEmbedding(
                cfg.vocab_size, cfg.hidden_size, params_dtype=dtype)
        mk = self.moe_kwargs(cfg)
        self.layers = nn.ModuleList([
            MoEDecoderLayer(cfg, engine_config, local,
                            qkv_bias=self.qkv_bias, qk_norm=self.qk_norm,
                            moe_kwargs=mk,
                            use_dense_mlp=self.is_dense_layer(
                                cfg, self.layer_start + local),
                            dtype=dtype)
            for local in range(self.layer_end - self.layer_start)])
        if self.is_last_stage:
            self.norm = RMSNorm(cfg.hidden_size,
                                getattr(cfg, "rms_norm_eps", 1e-6))
            self.lm_head = ParallelLMHead(cfg.vocab_size, cfg.hidden_size,
                                          params_dtype=dtype)
            if getattr(cfg, "tie_word_embeddings", False) and \
                    self.is_first_stage:
                self.lm_head.tie_to(self.embed_tokens)

    # ------------------------------------------------------------------
    def load_weights(self, weights: Iterable[Tuple[str, torch.Tensor]]):
        expert_weights = []

        def split_stream():
            for name, w in weights:
                if ".experts." in name:
                    expert_weights.append((name, w))
                else:
                    yield self._canonical_moe_name(name), w

        super().load_weights(split_stream())
        params = dict(self.named_parameters())
        for name, w in expert_weights:
            name = name.replace("model.", "", 1) \
                if name.startswith("model.") else name
            parts = name.split(".")
            g_idx = int(parts[1])
            if not (self.layer_start <= g_idx < self.layer_end):
                continue
            parts[1] = str(g_idx - self.layer_start)
            # .../experts.<eid>.<wname>.weight
            eidx = parts.index("experts")
            expert_id = int(parts[eidx + 1])
            wname = parts[eidx + 2]
            suffix = parts[eidx + 3] if len(parts) > eidx + 3 else "weight"
            prefix = ".".join(parts[:eidx + 1])
            prefix = self._canonical_moe_name(prefix)
            for ckpt, fused, shard in self.expert_params_mapping:
                if wname == ckpt:
                    if suffix == "weight_scale_inv":  # fp8 expert scales
                        fused = fused + "_scale_inv"
                    elif suffix == "weight_packed":  # ct int4 experts
                        fused = fused + "_packed"
                    elif suffix == "weight_scale":
                        fused = fused + "_scale"
                    elif suffix in ("qweight", "qzeros", "scales"):
                        fused = fused.replace("weight", suffix)  # int4
                    p = params[f"{prefix}.{fused}"]
                    if shard is None:
                        p.weight_loader(p, w, expert_id)
                    else:
                        p.weight_loader(p, w, expert_id, shard)
                    break

    @staticmethod
    def _canonical_moe_name(name: str) -> str:
        return name.replace("block_sparse_moe", "mlp")


class MixtralForCausalLM(MoEFamilyForCausalLM):
    def moe_kwargs(self, cfg):
        return dict(num_experts=cfg.num_local_experts,
                    top_k=cfg.num_experts_per_tok,
                    moe_intermediate=cfg.intermediate_size,
                    norm_topk_prob=True)


class Qwen2MoeForCausalLM(MoEFamilyForCausalLM):
    qkv_bias = True

    def moe_kwargs(self, cfg):
        return dict(num_experts=cfg.num_experts,
                    top_k=cfg.num_experts_per_tok,
                    moe_intermediate=cfg.moe_intermediate_size,
                    shared_intermediate=getattr(
                        cfg, "shared_expert_intermediate_size", 0),
                    norm_topk_prob=getattr(cfg, "norm_topk_prob", False))

    def is_dense_layer(self, cfg, global_idx):
        step = getattr(cfg, "decoder_sparse_step", 1)
        mlp_only = getattr(cfg, "mlp_only_layers", []) or []
        return global_idx in mlp_only or (step > 1 and
                                          (global_idx + 1) % step != 0)


class Qwen3MoeForCausalLM(MoEFamilyForCausalLM):
    qk_norm = True

    def moe_kwargs(self, cfg):
        return dict(num_experts=cfg.num_experts,
                    top_k=cfg.num_experts_per_tok,
                    moe_intermediate=cfg.moe_intermediate_size,
                    norm_topk_prob=getattr(cfg, "norm_topk_prob", True))

    def is_dense_layer(self, cfg, global_idx):
        mlp_only = getattr(cfg, "mlp_only_layers", []) or []
        return global_idx in mlp_only
