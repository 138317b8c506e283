"""DeepSeek-V2 / V3 family: MLA attention + grouped-topk MoE.

Parity target: reference models/deepseek_v2.py (MLA with
q_lora/kv_lora, YaRN with mscale, routed-scaling MoE with shared
experts, first_k_dense_replace). Round-1 compute path: the
"decompressed" MLA form — per-head K (nope|rope, Dk = 192) and V
(Dv = 128) cached in the paged pool and attended by
ops.mla_paged_attention (CPU reference; the absorbed-latent gfx950
decode kernel over the 576-dim latent cache is the round-2 item, see
docs/kernels.md).
"""

import math
from typing import Iterable, Tuple

import torch
import torch.nn as nn

from gllm_amd import ops
from gllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from gllm_amd.layers.layernorm import RMSNorm
from gllm_amd.layers.linear import (ColumnParallelLinear,
                                    MergedColumnParallelLinear,
                                    ReplicatedLinear, RowParallelLinear)
from gllm_amd.layers.moe.layer import FusedMoE
from gllm_amd.layers.rotary import get_rope
from gllm_amd.models.llama_family import DenseMLP
from gllm_amd.runtime.forward_context import ForwardContext


def _yarn_mscale(scale: float, mscale: float) -> float:
    if scale <= 1.0:
        return 1.0
    return 0.1 * mscale * math.log(scale) + 1.0


class MLAAttention(nn.Module):
    """MLA attention. Two execution modes (EngineConfig.mla_mode):

    * ``absorbed`` (default) — the latent form: the paged cache holds
      ONE shared row per token, [c_kv (kv_lora_rank) | k_pe (rope)]
      (576 dims on real configs), attended as MQA. W_UK is folded into
      the query (q_lat = q_nope @ W_UK) and W_UV applied to the latent
      attention output; mathematically identical to decompressed
      (q·(W_UK c) == (W_UK^T q)·c) at ~H*(dn+dv)/(lora+rope) less KV —
      ~71x for DeepSeek-V3. The v-cache is a zero-copy VIEW of the
      latent's first kv_lora dims. This layout is the contract for the
      round-2 gfx950 absorbed-decode kernel (576-dim MQA over the
      latent cache, reference layers/attention.py:653-925).
    * ``decompressed`` — per-head K/V cache, the numerics cross-check
      (tests/test_deepseek_cpu.py asserts both modes emit identical
      tokens).
    """

    def __init__(self, cfg, layer_idx: int, dtype=None, absorbed=True):
        super().__init__()
        from gllm_amd.parallel import get_tp_size
        tp = get_tp_size()
        hidden = cfg.hidden_size
        self.layer_idx = layer_idx
        self.absorbed = absorbed
        self.total_heads = cfg.num_attention_heads
        assert self.total_heads % tp == 0
        self.num_heads = self.total_heads // tp
        self.qk_nope = cfg.qk_nope_head_dim
        self.qk_rope = cfg.qk_rope_head_dim
        self.v_dim = cfg.v_head_dim
        self.qk_dim = self.qk_nope + self.qk_rope
        self.kv_lora_rank = cfg.kv_lora_rank
        self.q_lora_rank = getattr(cfg, "q_lora_rank", None)

        if self.q_lora_rank:
            self.q_a_proj = ReplicatedLinear(hidden, self.q_lora_rank,
                                             params_dtype=dtype)
            self.q_a_layernorm = RMSNorm(self.q_lora_rank,
                                         getattr(cfg, "rms_norm_eps", 1e-6))
            self.q_b_proj = ColumnParallelLinear(
                self.q_lora_rank, self.total_heads * self.qk_dim,
                params_dtype=dtype)
        else:
            self.q_proj = ColumnParallelLinear(
                hidden, self.total_heads * self.qk_dim, params_dtype=dtype)
        self.kv_a_proj_with_mqa = ReplicatedLinear(
            hidden, self.kv_lora_rank + self.qk_rope, params_dtype=dtype)
        self.kv_a_layernorm = RMSNorm(self.kv_lora_rank,
                                      getattr(cfg, "rms_norm_eps", 1e-6))
        self.kv_b_proj = ColumnParallelLinear(
            self.kv_lora_rank,
            self.total_heads * (self.qk_nope + self.v_dim),
            params_dtype=dtype)
        self.o_proj = RowParallelLinear(self.total_heads * self.v_dim,
                                        hidden, params_dtype=dtype)

        rope_scaling = getattr(cfg, "rope_scaling", None)
        max_pos = getattr(cfg, "max_position_embeddings", 32768)
        self.scale = self.qk_dim ** -0.5
        if rope_scaling and rope_scaling.get("type",
                                             rope_scaling.get("rope_type")
                                             ) == "yarn":
            factor = rope_scaling["factor"]
            ms_all = rope_scaling.get("mscale_all_dim", 0.0)
            if ms_all:
                m = _yarn_mscale(factor, ms_all)
                self.scale = self.scale * m * m
        self.rotary_emb = get_rope(self.qk_rope, self.qk_rope, max_pos,
                                   getattr(cfg, "rope_theta", 10000.0),
                                   is_neox=True, rope_scaling=rope_scaling)

    def _dsa_select(self, positions, hidden, q_resid, fctx):
        """DSA hook (DeepSeek-V3.2): per-query top-k token positions, or
        None for dense MLA. Overridden in models/deepseek_v32.py."""
        return None

    def _uk_uv(self):
        """Per-head absorption views of kv_b_proj's weight:
        W_UK [H, dn, lora], W_UV [H, dv, lora]."""
        w = self.kv_b_proj.weight.view(
            self.num_heads, self.qk_nope + self.v_dim, self.kv_lora_rank)
        return w[:, :self.qk_nope, :], w[:, self.qk_nope:, :]

    def forward(self, positions, hidden, fctx: ForwardContext):
        T = hidden.shape[0]
        H = self.num_heads
        if self.q_lora_rank:
            q_resid = self.q_a_layernorm(self.q_a_proj(hidden))
            q = self.q_b_proj(q_resid)
        else:
            q_resid = None
            q = self.q_proj(hidden)
        q = q.view(T, H, self.qk_dim)
        q_nope, q_pe = q.split([self.qk_nope, self.qk_rope], dim=-1)

        kv_a = self.kv_a_proj_with_mqa(hidden)
        c_kv, k_pe = kv_a.split([self.kv_lora_rank, self.qk_rope], dim=-1)
        c_kv = self.kv_a_layernorm(c_kv.contiguous())

        # rope on q_pe (per head) and the shared k_pe (one "head")
        q_pe = q_pe.reshape(T, H * self.qk_rope).contiguous()
        k_pe = k_pe.contiguous()
        q_pe, k_pe = self.rotary_emb(positions, q_pe, k_pe)
        q_pe = q_pe.view(T, H, self.qk_rope)

        if self.absorbed:
            return self._forward_absorbed(positions, hidden, q_resid,
                                          q_nope, q_pe, c_kv, k_pe, fctx)
        kv = self.kv_b_proj(c_kv)
        kv = kv.view(T, H, self.qk_nope + self.v_dim)
        k_nope, v = kv.split([self.qk_nope, self.v_dim], dim=-1)
        k = torch.cat(
            [k_nope, k_pe.unsqueeze(1).expand(T, H, self.qk_rope)], dim=-1)
        qf = torch.cat([q_nope, q_pe], dim=-1).contiguous()
        k = k.contiguous()
        v = v.contiguous()

        if fctx.is_profile_run:
            return v.reshape(T, -1)
        k_cache = fctx.k_caches[self.layer_idx]
        v_cache = fctx.v_caches[self.layer_idx]
        ops.reshape_and_cache(k, v, k_cache, v_cache, fctx.slot_mapping)
        topk_pos = self._dsa_select(positions, hidden, q_resid, fctx)
        out = ops.mla_paged_attention(
            qf, k_cache, v_cache, fctx.block_table, fctx.seq_lens,
            fctx.query_start_loc, self.scale, topk_positions=topk_pos,
            seq_lens_cpu=fctx.host_seq_lens(),
            query_start_loc_cpu=fctx.host_qsl())
        return self.o_proj(out.reshape(T, -1))

    def _forward_absorbed(self, positions, hidden, q_resid, q_nope, q_pe,
                          c_kv, k_pe, fctx):
        T, H = q_nope.shape[0], self.num_heads
        if fctx.is_profile_run:
            # peak-shaped dummy matching the o_proj input
            return q_nope.new_zeros(T, H * self.v_dim)
        w_uk, w_uv = self._uk_uv()
        # q_lat[t,h,l] = sum_d q_nope[t,h,d] * W_UK[h,d,l]
        q_lat = torch.einsum("thd,hdl->thl", q_nope.float(),
                             w_uk.float()).to(q_nope.dtype)
        qf = torch.cat([q_lat, q_pe], dim=-1).contiguous()  # [T,H,l+r]
        k_lat = torch.cat([c_kv, k_pe], dim=-1).unsqueeze(1)  # [T,1,l+r]
        k_cache = fctx.k_caches[self.layer_idx]   # [P,page,1,lora+rope]
        v_cache = fctx.v_caches[self.layer_idx]   # VIEW: [..., :lora]
        # write the latent row once (the v view aliases its first lora
        # dims, so caching k fills both)
        ops.cache_latent(k_lat.contiguous(), k_cache, fctx.slot_mapping)
        topk_pos = self._dsa_select(positions, hidden, q_resid, fctx)
        out_lat = ops.mla_paged_attention(
            qf, k_cache, v_cache, fctx.block_table, fctx.seq_lens,
            fctx.query_start_loc, self.scale, topk_positions=topk_pos,
            seq_lens_cpu=fctx.host_seq_lens(),
            query_start_loc_cpu=fctx.host_qsl())
        # out[t,h,v] = sum_l out_lat[t,h,l] * W_UV[h,v,l]
        out = torch.einsum("thl,hvl->thv", out_lat.float(),
                           w_uv.float()).to(out_lat.dtype)
        return self.o_proj(out.reshape(T, -1))


class DeepseekMoE(nn.Module):
    def __init__(self, cfg, engine_config, dtype=None):
        super().__init__()
        self.num_experts = cfg.n_routed_experts
        self.top_k = cfg.num_experts_per_tok
        self.n_group = getattr(cfg, "n_group", 1)
        self.topk_group = getattr(cfg, "topk_group", 1)
        self.routed_scaling = getattr(cfg, "routed_scaling_factor", 1.0)
        self.norm_topk = getattr(cfg, "norm_topk_prob", False)
        self.scoring = getattr(cfg, "scoring_func", "softmax")
        self.topk_method = getattr(cfg, "topk_method", "greedy")
        self.gate = ReplicatedLinear(cfg.hidden_size, self.num_experts,
                                     params_dtype=dtype)
        if self.topk_method == "noaux_tc":
            self.gate_bias = nn.Parameter(
                torch.zeros(self.num_experts, dtype=torch.float32),
                requires_grad=False)
        else:
            self.gate_bias = None
        self.experts = FusedMoE(self.num_experts, self.top_k,
                                cfg.hidden_size, cfg.moe_intermediate_size,
                                renormalize=False,
                                use_ep=engine_config.use_ep,
                                params_dtype=dtype)
        n_shared = getattr(cfg, "n_shared_experts", 0) or 0
        if n_shared:
            self.shared_experts = DenseMLP(
                cfg.hidden_size, cfg.moe_intermediate_size * n_shared,
                dtype=dtype)
        else:
            self.shared_experts = None

    def _route(self, logits: torch.Tensor):
        if self.topk_method in ("group_limited_greedy", "noaux_tc") and \
                self.n_group > 1:
            w, ids = ops.grouped_topk(
                logits, self.top_k, self.n_group, self.topk_group,
                renormalize=self.norm_topk,
                scoring="sigmoid" if self.topk_method == "noaux_tc"
                else self.scoring,
                e_bias=self.gate_bias)
        else:
            w, ids = ops.topk_softmax(logits, self.top_k,
                                      renormalize=self.norm_topk)
        return w * self.routed_scaling, ids

    def forward(self, x):
        logits = self.gate(x)
        weights, ids = self._route(logits)
        out = self.experts.forward_routed(x, weights, ids)
        if self.shared_experts is not None:
            out = out + self.shared_experts(x)
        return out


class DeepseekDecoderLayer(nn.Module):
    def __init__(self, cfg, engine_config, layer_idx, global_idx,
                 dtype=None, attn_cls=None):
        super().__init__()
        eps = getattr(cfg, "rms_norm_eps", 1e-6)
        self.self_attn = (attn_cls or MLAAttention)(
            cfg, layer_idx, dtype=dtype,
            absorbed=engine_config.mla_mode == "absorbed")
        first_dense = getattr(cfg, "first_k_dense_replace", 0)
        step = getattr(cfg, "moe_layer_freq", 1)
        is_moe = (getattr(cfg, "n_routed_experts", None)
                  and global_idx >= first_dense
                  and global_idx % step == 0)
        if is_moe:
            self.mlp = DeepseekMoE(cfg, engine_config, dtype=dtype)
        else:
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


class DeepseekV2ForCausalLM(nn.Module):
    attn_cls = None  # default MLAAttention; V3.2 swaps in the DSA variant

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
            DeepseekDecoderLayer(cfg, engine_config, local,
                                 self.layer_start + local, dtype=dtype,
                                 attn_cls=self.attn_cls)
            for local in range(self.layer_end - self.layer_start)])
        if self.is_last_stage:
            self.norm = RMSNorm(cfg.hidden_size,
                                getattr(cfg, "rms_norm_eps", 1e-6))
            self.lm_head = ParallelLMHead(cfg.vocab_size, cfg.hidden_size,
                                          params_dtype=dtype)

    @property
    def num_local_layers(self):
        return self.layer_end - self.layer_start

    @property
    def kv_geometry(self):
        a = self.layers[0].self_attn
        if a.absorbed:
            return 1, a.kv_lora_rank + a.qk_rope, a.kv_lora_rank
        return a.num_heads, a.qk_dim, a.v_dim

    @property
    def kv_share_latent(self):
        """Absorbed MLA: v-cache = view of k-cache[..., :kv_lora_rank]
        (runtime/model_runner.py honors this at allocation)."""
        a = self.layers[0].self_attn
        return a.kv_lora_rank if a.absorbed else None

    def forward(self, input_ids, positions, fctx, hidden_states=None,
                residual=None):
        if self.is_first_stage:
            hidden_states = self.embed_tokens(input_ids)
            if fctx.mm_rows is not None:
                # multimodal merge (Kimi-K2.5 rides the DeepSeek-V3
                # backbone; models/kimi_k25.py)
                hidden_states = hidden_states.index_copy(
                    0, fctx.mm_rows,
                    fctx.mm_embeds.to(hidden_states.dtype))
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
        stacked = [("gate_up_proj", "gate_proj", 0),
                   ("gate_up_proj", "up_proj", 1)]
        expert_map = [("gate_proj", "w13_weight", 0),
                      ("up_proj", "w13_weight", 1),
                      ("down_proj", "w2_weight", None)]
        for name, w in weights:
            if name.startswith("model."):
                name = name[len("model."):]
            if name.startswith("layers."):
                parts = name.split(".")
                g = int(parts[1])
                if not (self.layer_start <= g < self.layer_end):
                    continue
                parts[1] = str(g - self.layer_start)
                name = ".".join(parts)
            elif name.startswith("embed_tokens"):
                if self.is_first_stage:
                    p = params["embed_tokens.weight"]
                    p.weight_loader(p, w)
                continue
            elif name.startswith("lm_head"):
                if self.is_last_stage:
                    p = params["lm_head.weight"]
                    p.weight_loader(p, w)
                continue
            elif name.startswith("norm."):
                if not self.is_last_stage:
                    continue
            if ".experts." in name:
                parts = name.split(".")
                eidx = parts.index("experts")
                expert_id = int(parts[eidx + 1])
                wname = parts[eidx + 2]
                suffix = parts[eidx + 3] if len(parts) > eidx + 3 \
                    else "weight"
                prefix = ".".join(parts[:eidx + 1])
                for ckpt, fused, shard in expert_map:
                    if wname == ckpt:
                        if suffix == "weight_scale_inv":  # fp8 scales
                            fused = fused + "_scale_inv"
                        elif suffix == "weight_packed":  # ct int4
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
                continue
            if "e_score_correction_bias" in name:
                tgt = name.replace("gate.e_score_correction_bias",
                                   "gate_bias")
                if tgt in params:
                    params[tgt].data.copy_(w.float())
                continue
            if "shared_experts" in name or ".mlp." in name:
                hit = False
                for fused, ckpt, shard in stacked:
                    if ckpt in name:
                        tgt = name.replace(ckpt, fused)
                        if tgt in params:
                            p = params[tgt]
                            p.weight_loader(p, w, shard)
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


class DeepseekV3ForCausalLM(DeepseekV2ForCausalLM):
    pass
