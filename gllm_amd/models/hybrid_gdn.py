"""Hybrid Gated-DeltaNet + full-attention decoder (Qwen3.5 / Qwen3-Next
family). Parity target: reference models/qwen3_5.py + qwen3_5_moe.py
(the MoE variant is config-driven here: ``num_experts`` swaps the dense
MLP for a routed MoE block on sparse layers).

Layer interleave from ``layer_types`` (or ``full_attention_interval``):
linear_attention layers run the GDN recurrence over per-seq SSM state
slots (core/ssm.py); full_attention layers use the paged KV cache with
DENSE kv-layer indices (a 24-layer stack with 6 softmax layers allocates
6 KV layers, reference qwen3_5.py header notes).

Compute path: torch GDN ops (ops/gdn_ref.py — chunk-parallel WY-form
prefill + sequential decode; also the oracle for the round-2 HIP
kernels). Prefix caching is ON: recurrent state snapshots at page
boundaries restore on hits (core/ssm.py). Checkpoint loading covers
the fused in_proj_qkvz/ba layout.
"""

from typing import Iterable, List, Tuple

import torch
import torch.nn as nn

from gllm_amd.layers.attention import Attention
from gllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from gllm_amd.layers.layernorm import GemmaRMSNorm, RMSNorm
from gllm_amd.layers.linear import (ColumnParallelLinear,
                                    MergedColumnParallelLinear,
                                    QKVParallelLinear, RowParallelLinear)
from gllm_amd.layers.rotary import get_rope
from gllm_amd.models.llama_family import DenseMLP
from gllm_amd.ops import gdn_ref
from gllm_amd.runtime.forward_context import ForwardContext


def get_layer_types(cfg) -> List[str]:
    lt = getattr(cfg, "layer_types", None) or \
        getattr(cfg, "layers_block_type", None)
    if lt is not None:
        return ["linear_attention" if t in ("linear_attention",
                                            "linear_attn") else
                "full_attention" for t in lt]
    interval = getattr(cfg, "full_attention_interval", 4)
    n = cfg.num_hidden_layers
    return ["full_attention" if (i + 1) % interval == 0
            else "linear_attention" for i in range(n)]


class GatedDeltaNet(nn.Module):
    """Reference: Qwen3_5GatedDeltaNet (qwen3_5.py:177-506); weight
    names follow the checkpoint (in_proj_qkvz, in_proj_ba, conv1d,
    A_log, dt_bias, norm, out_proj)."""

    def __init__(self, cfg, ssm_layer_id: int, dtype=None):
        super().__init__()
        from gllm_amd.parallel import get_tp_size
        tp = get_tp_size()
        self.ssm_layer_id = ssm_layer_id
        self.num_v_heads = cfg.linear_num_value_heads
        self.num_k_heads = cfg.linear_num_key_heads
        self.head_k_dim = cfg.linear_key_head_dim
        self.head_v_dim = cfg.linear_value_head_dim
        self.key_dim = self.head_k_dim * self.num_k_heads
        self.value_dim = self.head_v_dim * self.num_v_heads
        self.conv_kernel = cfg.linear_conv_kernel_dim
        self.conv_dim = self.key_dim * 2 + self.value_dim
        assert self.num_v_heads % tp == 0 and self.num_k_heads % tp == 0
        self.tp_v = self.num_v_heads // tp
        self.tp_k_heads = self.num_k_heads // tp

        self.in_proj_qkvz = MergedColumnParallelLinear(
            cfg.hidden_size,
            [self.key_dim, self.key_dim, self.value_dim, self.value_dim],
            params_dtype=dtype)
        self.in_proj_ba = MergedColumnParallelLinear(
            cfg.hidden_size, [self.num_v_heads, self.num_v_heads],
            params_dtype=dtype)
        self.conv1d_weight = nn.Parameter(
            torch.empty(self.conv_dim // tp, self.conv_kernel, dtype=dtype),
            requires_grad=False)
        self.conv1d_weight.weight_loader = self._load_conv
        self.dt_bias = nn.Parameter(torch.ones(self.tp_v),
                                    requires_grad=False)
        self.dt_bias.weight_loader = self._load_headed
        self.A_log = nn.Parameter(torch.zeros(self.tp_v,
                                              dtype=torch.float32),
                                  requires_grad=False)
        self.A_log.weight_loader = self._load_headed
        self.norm_weight = nn.Parameter(
            torch.ones(self.head_v_dim, dtype=dtype), requires_grad=False)
        self.out_proj = RowParallelLinear(self.value_dim, cfg.hidden_size,
                                          params_dtype=dtype)
        self.eps = getattr(cfg, "rms_norm_eps", 1e-6)
        self.scale = self.head_k_dim ** -0.5

    def _load_conv(self, param, loaded):
        # checkpoint shape [conv_dim, 1, kernel]; TP shard along channels
        from gllm_amd.parallel import get_tp_rank, get_tp_size
        w = loaded.reshape(loaded.shape[0], -1)
        n = w.shape[0] // get_tp_size()
        param.data.copy_(w.narrow(0, get_tp_rank() * n, n))

    @staticmethod
    def _load_headed(param, loaded):
        # per-value-head vectors (A_log, dt_bias): TP narrow
        from gllm_amd.parallel import get_tp_rank, get_tp_size
        n = loaded.shape[0] // get_tp_size()
        param.data.copy_(loaded.narrow(0, get_tp_rank() * n, n))

    def load_fused_qkvz(self, loaded):
        """Checkpoint ships in_proj_qkvz as ONE fused tensor; split into
        the (q, k, v, z) segments and route each through the merged
        column loader (reference weight_loader.py GDN fused-proj
        pre-pass)."""
        p = self.in_proj_qkvz.weight
        off = 0
        for sid, size in enumerate([self.key_dim, self.key_dim,
                                    self.value_dim, self.value_dim]):
            p.weight_loader(p, loaded.narrow(0, off, size), sid)
            off += size

    def load_fused_ba(self, loaded):
        p = self.in_proj_ba.weight
        off = 0
        for sid, size in enumerate([self.num_v_heads, self.num_v_heads]):
            p.weight_loader(p, loaded.narrow(0, off, size), sid)
            off += size

    def forward(self, hidden: torch.Tensor, fctx: ForwardContext):
        if fctx.is_profile_run or fctx.ssm_pool is None:
            return torch.zeros_like(hidden)
        T = hidden.shape[0]
        k_tp = self.tp_k_heads * self.head_k_dim
        v_tp = self.tp_v * self.head_v_dim
        qkvz = self.in_proj_qkvz(hidden)
        ba = self.in_proj_ba(hidden)
        q, k, v, z = qkvz.split([k_tp, k_tp, v_tp, v_tp], dim=-1)
        b, a = ba.split([self.tp_v, self.tp_v], dim=-1)
        mixed = torch.cat([q, k, v], dim=-1)         # [T, conv_dim/tp]

        pool = fctx.ssm_pool
        conv_states = pool.conv_state[self.ssm_layer_id]
        ssm_states = pool.ssm_state[self.ssm_layer_id]
        g_all, beta_all = gdn_ref.gdn_gating(self.A_log, a, b, self.dt_bias)

        out_core = torch.empty(T, self.tp_v, self.head_v_dim,
                               dtype=hidden.dtype, device=hidden.device)
        # under graph capture/replay only the device slot buffer exists
        B = len(fctx.ssm_slots) if fctx.ssm_slots is not None else T

        # ---- batched decode fast path (gfx950 kernels): every seq is a
        # single-token step with carried state -> one conv-update launch
        # + one fused recurrent launch for the whole batch (the per-seq
        # loop below costs ~8 launches PER SEQUENCE)
        from gllm_amd import ops as _ops
        graph_path = fctx.ssm_slots_dev is not None
        if (hidden.is_cuda and self.head_k_dim == 128
                and self.conv1d_weight.shape[1] == 4
                and _ops.has_kernels()
                and (graph_path or
                     (T == B and all(bool(h) for h in fctx.ssm_has_init)))):
            if graph_path:
                slots_t = fctx.ssm_slots_dev[:T]
            else:
                slots_t = torch.as_tensor(
                    [int(s) for s in fctx.ssm_slots], dtype=torch.long,
                    device=hidden.device)
            conv_out = _ops.gdn_conv_update(
                mixed.contiguous(), self.conv1d_weight, conv_states,
                slots_t)
            qd, kd, vd = conv_out.split([k_tp, k_tp, v_tp], dim=-1)
            G = self.tp_v // self.tp_k_heads
            qn = gdn_ref.l2norm(
                qd.view(T, self.tp_k_heads, self.head_k_dim).float()) \
                * self.scale
            kn = gdn_ref.l2norm(
                kd.view(T, self.tp_k_heads, self.head_k_dim).float())
            o = _ops.gdn_decode(
                qn.repeat_interleave(G, dim=1),
                kn.repeat_interleave(G, dim=1),
                vd.view(T, self.tp_v, self.head_v_dim).float(),
                g_all, beta_all, ssm_states, slots_t)
            gated = _ops.rmsnorm_gated(
                o.reshape(T * self.tp_v, self.head_v_dim),
                z.reshape(T * self.tp_v, self.head_v_dim).contiguous(),
                self.norm_weight, self.eps)
            return self.out_proj(gated.reshape(T, -1))

        qsl = fctx.host_qsl()
        # ---- GPU mixed/prefill path: conv stays per segment (cheap),
        # but ALL prefill segments run ONE padded batched WY pass
        # (chunk=256; padding rows carry beta=0/g=0 so they are inert)
        # instead of a per-seq python loop of ~20 launches per 64-token
        # chunk — the r2 TTFT fix for hybrid models.
        prefill_batched = hidden.is_cuda and B > 0
        conv_outs = [None] * B
        prefill_idx = []
        for i in range(B):
            s, e = qsl[i], qsl[i + 1]
            slot = int(fctx.ssm_slots[i])
            has_init = bool(fctx.ssm_has_init[i])
            if not has_init:
                conv_states[slot].zero_()
                ssm_states[slot].zero_()
            if e - s == 1 and has_init:
                conv_out = gdn_ref.causal_conv1d_update(
                    mixed[s], self.conv1d_weight, conv_states[slot]
                ).unsqueeze(0)
            else:
                conv_out = gdn_ref.causal_conv1d_prefill(
                    mixed[s:e], self.conv1d_weight, conv_states[slot],
                    has_init)
                if prefill_batched:
                    conv_outs[i] = conv_out
                    prefill_idx.append(i)
                    continue
            qd, kd, vd = conv_out.split([k_tp, k_tp, v_tp], dim=-1)
            rule = gdn_ref.gated_delta_rule if e - s == 1 \
                else gdn_ref.gated_delta_rule_chunked
            o = rule(
                qd.view(e - s, self.tp_k_heads, self.head_k_dim),
                kd.view(e - s, self.tp_k_heads, self.head_k_dim),
                vd.view(e - s, self.tp_v, self.head_v_dim),
                g_all[s:e], beta_all[s:e], self.scale, ssm_states[slot])
            out_core[s:e] = o

        if prefill_idx:
            Bp = len(prefill_idx)
            Tmax = max(qsl[i + 1] - qsl[i] for i in prefill_idx)
            dev = hidden.device
            qb = torch.zeros(Bp, Tmax, self.tp_k_heads, self.head_k_dim,
                             dtype=torch.float32, device=dev)
            kb = torch.zeros_like(qb)
            vb = torch.zeros(Bp, Tmax, self.tp_v, self.head_v_dim,
                             dtype=torch.float32, device=dev)
            gb = torch.zeros(Bp, Tmax, self.tp_v, dtype=torch.float32,
                             device=dev)
            bb = torch.zeros_like(gb)
            slots_l = [int(fctx.ssm_slots[i]) for i in prefill_idx]
            for j, i in enumerate(prefill_idx):
                s, e = qsl[i], qsl[i + 1]
                n = e - s
                qd, kd, vd = conv_outs[i].split([k_tp, k_tp, v_tp], -1)
                qb[j, :n] = qd.view(n, self.tp_k_heads, self.head_k_dim)
                kb[j, :n] = kd.view(n, self.tp_k_heads, self.head_k_dim)
                vb[j, :n] = vd.view(n, self.tp_v, self.head_v_dim)
                gb[j, :n] = g_all[s:e]
                bb[j, :n] = beta_all[s:e]
            states_b = ssm_states[slots_l].contiguous()
            ob = _ops.gdn_chunk_prefill(
                qb, kb, vb, gb, bb, states_b, self.scale)
            ssm_states[slots_l] = states_b
            for j, i in enumerate(prefill_idx):
                s, e = qsl[i], qsl[i + 1]
                out_core[s:e] = ob[j, :e - s].to(out_core.dtype)
        gated = gdn_ref.rmsnorm_gated(
            out_core.reshape(T * self.tp_v, self.head_v_dim),
            z.reshape(T * self.tp_v, self.head_v_dim),
            self.norm_weight, self.eps)
        return self.out_proj(gated.reshape(T, -1))


class HybridFullAttention(nn.Module):
    """Full-attention block with optional sigmoid output gate
    (attn_output_gate, qwen3_5.py header) and partial rotary."""

    def __init__(self, cfg, kv_layer_idx: int, dtype=None):
        super().__init__()
        hidden = cfg.hidden_size
        self.total_heads = cfg.num_attention_heads
        self.total_kv = getattr(cfg, "num_key_value_heads",
                                self.total_heads)
        self.head_dim = getattr(cfg, "head_dim",
                                hidden // self.total_heads)
        self.gate = bool(getattr(cfg, "attn_output_gate", False))
        q_mult = 2 if self.gate else 1
        # q(+gate) | k | v fused, sharded by head (gate doubles q heads)
        self.qkv_proj = QKVParallelLinear(
            hidden, self.head_dim, self.total_heads * q_mult,
            self.total_kv,
            bias=bool(getattr(cfg, "attention_bias", False)),
            params_dtype=dtype)
        self.q_mult = q_mult
        self.o_proj = RowParallelLinear(self.total_heads * self.head_dim,
                                        hidden, params_dtype=dtype)
        eps = getattr(cfg, "rms_norm_eps", 1e-6)
        # reference qwen3_5.py:564-565 uses the (1+w) Gemma convention
        # for EVERY non-gated norm in this family (q/k, block, final)
        self.q_norm = GemmaRMSNorm(self.head_dim, eps)
        self.k_norm = GemmaRMSNorm(self.head_dim, eps)
        rot = int(self.head_dim *
                  getattr(cfg, "partial_rotary_factor", 1.0))
        self.rotary_emb = get_rope(
            self.head_dim, rot,
            getattr(cfg, "max_position_embeddings", 32768),
            getattr(cfg, "rope_theta", 10000.0), is_neox=True,
            rope_scaling=getattr(cfg, "rope_scaling", None))
        self.num_heads = self.qkv_proj.num_heads // q_mult
        self.attn = Attention(kv_layer_idx, self.num_heads,
                              self.qkv_proj.num_kv_heads, self.head_dim,
                              self.head_dim ** -0.5)

    def forward(self, positions, hidden, fctx):
        qg, k, v = self.qkv_proj(hidden)
        T = hidden.shape[0]
        if self.gate:
            qg = qg.view(T, self.num_heads, 2 * self.head_dim)
            q, gate = qg.split([self.head_dim, self.head_dim], dim=-1)
            q = q.reshape(T, -1)
            gate = gate.reshape(T, -1)
        else:
            q = qg
            gate = None
        q = self.q_norm(q.contiguous().view(T, -1, self.head_dim)
                        ).view(T, -1)
        k = self.k_norm(k.contiguous().view(T, -1, self.head_dim)
                        ).view(T, -1)
        q, k = self.rotary_emb(positions, q, k)
        o = self.attn(q, k, v, fctx)
        if gate is not None:
            o = o * torch.sigmoid(gate.float()).to(o.dtype)
        return self.o_proj(o)


class HybridDecoderLayer(nn.Module):
    def __init__(self, cfg, layer_type: str, ssm_or_kv_idx: int,
                 dtype=None, mlp: nn.Module = None):
        super().__init__()
        eps = getattr(cfg, "rms_norm_eps", 1e-6)
        self.layer_type = layer_type
        if layer_type == "linear_attention":
            self.linear_attn = GatedDeltaNet(cfg, ssm_or_kv_idx,
                                             dtype=dtype)
        else:
            self.self_attn = HybridFullAttention(cfg, ssm_or_kv_idx,
                                                 dtype=dtype)
        self.mlp = mlp if mlp is not None else DenseMLP(
            cfg.hidden_size, cfg.intermediate_size, dtype=dtype)
        self.input_layernorm = GemmaRMSNorm(cfg.hidden_size, eps)
        self.post_attention_layernorm = GemmaRMSNorm(cfg.hidden_size, eps)

    def forward(self, positions, hidden, residual, fctx):
        if residual is None:
            residual = hidden
            hidden = self.input_layernorm(hidden)
        else:
            hidden, residual = self.input_layernorm(hidden, residual)
        if self.layer_type == "linear_attention":
            hidden = self.linear_attn(hidden, fctx)
        else:
            hidden = self.self_attn(positions, hidden, fctx)
        hidden, residual = self.post_attention_layernorm(hidden, residual)
        hidden = self.mlp(hidden)
        return hidden, residual


class Qwen3_5ForCausalLM(nn.Module):
    """Hybrid GDN decoder; the MoE variant (Qwen3-Next / Qwen3.5-MoE,
    reference qwen3_5_moe.py) is config-driven — layers with
    ``num_experts > 1`` (minus ``mlp_only_layers`` /
    ``decoder_sparse_step`` exceptions) get a routed MoE MLP."""

    @staticmethod
    def _is_sparse_layer(cfg, global_idx: int) -> bool:
        if getattr(cfg, "num_experts", 0) in (0, 1, None):
            return False
        step = getattr(cfg, "decoder_sparse_step", 1) or 1
        mlp_only = getattr(cfg, "mlp_only_layers", []) or []
        return global_idx not in mlp_only and (global_idx + 1) % step == 0

    def _make_mlp(self, cfg, engine_config, global_idx, dtype):
        if not self._is_sparse_layer(cfg, global_idx):
            return None  # HybridDecoderLayer builds a DenseMLP
        from gllm_amd.models.moe_family import MoEBlock
        return MoEBlock(
            cfg, engine_config, dtype=dtype,
            num_experts=cfg.num_experts,
            top_k=cfg.num_experts_per_tok,
            moe_intermediate=cfg.moe_intermediate_size,
            shared_intermediate=getattr(
                cfg, "shared_expert_intermediate_size", 0) or 0,
            norm_topk_prob=getattr(cfg, "norm_topk_prob", True))

    def __init__(self, cfg, engine_config):
        super().__init__()
        self.cfg = cfg
        dtype = engine_config.torch_dtype()
        from gllm_amd.parallel import get_pp_rank, get_tp_size, \
            is_first_pp_rank, is_last_pp_rank
        num_layers = cfg.num_hidden_layers
        self.layer_start, self.layer_end = engine_config.pp_layer_range(
            get_pp_rank(), num_layers)
        self.is_first_stage = is_first_pp_rank()
        self.is_last_stage = is_last_pp_rank()
        types = get_layer_types(cfg)
        if self.is_first_stage:
            self.embed_tokens = VocabParallelEmbedding(
                cfg.vocab_size, cfg.hidden_size, params_dtype=dtype)
        # dense indices within this stage
        layers = []
        kv_idx = 0
        ssm_idx = 0
        self._local_types = []
        for g in range(self.layer_start, self.layer_end):
            t = types[g]
            self._local_types.append(t)
            mlp = self._make_mlp(cfg, engine_config, g, dtype)
            if t == "linear_attention":
                layers.append(HybridDecoderLayer(cfg, t, ssm_idx,
                                                 dtype=dtype, mlp=mlp))
                ssm_idx += 1
            else:
                layers.append(HybridDecoderLayer(cfg, t, kv_idx,
                                                 dtype=dtype, mlp=mlp))
                kv_idx += 1
        self.layers = nn.ModuleList(layers)
        self.num_kv_layers = kv_idx
        tp = get_tp_size()
        from gllm_amd.core.ssm import SSMSpec
        self.ssm_spec = SSMSpec(
            num_ssm_layers=ssm_idx,
            conv_dim=(cfg.linear_key_head_dim * cfg.linear_num_key_heads
                      * 2 + cfg.linear_value_head_dim
                      * cfg.linear_num_value_heads) // tp,
            conv_kernel=cfg.linear_conv_kernel_dim,
            num_v_heads=cfg.linear_num_value_heads // tp,
            head_k_dim=cfg.linear_key_head_dim,
            head_v_dim=cfg.linear_value_head_dim)
        if self.is_last_stage:
            self.norm = GemmaRMSNorm(cfg.hidden_size,
                                     getattr(cfg, "rms_norm_eps", 1e-6))
            self.lm_head = ParallelLMHead(cfg.vocab_size, cfg.hidden_size,
                                          params_dtype=dtype)
            if getattr(cfg, "tie_word_embeddings", False) and \
                    self.is_first_stage:
                self.lm_head.tie_to(self.embed_tokens)

    @property
    def num_local_layers(self):
        return self.layer_end - self.layer_start

    @property
    def kv_geometry(self):
        for layer in self.layers:
            if layer.layer_type == "full_attention":
                a = layer.self_attn
                return a.qkv_proj.num_kv_heads, a.head_dim
        return 1, 64

    def forward(self, input_ids, positions, fctx, hidden_states=None,
                residual=None):
        if self.is_first_stage:
            hidden_states = self.embed_tokens(input_ids)
            if fctx.mm_rows is not None:
                # Qwen3.5-VL: vision embeddings replace image-pad rows
                hidden_states = hidden_states.index_copy(
                    0, fctx.mm_rows,
                    fctx.mm_embeds.to(hidden_states.dtype))
            residual = None
        ds = getattr(fctx, "mm_deepstack", None)
        for li, layer in enumerate(self.layers):
            hidden_states, residual = layer(positions, hidden_states,
                                            residual, fctx)
            if ds is not None:
                # deepstack level li ADDED at the image rows after
                # decoder layer li (reference qwen3_5.py:795-812 —
                # same contract as Qwen3-VL, GDN layers included)
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

    def compute_logits(self, hidden_states, fctx):
        rows = hidden_states
        if fctx.logits_indices is not None:
            rows = hidden_states.index_select(0, fctx.logits_indices)
        return self.lm_head(rows)

    def load_weights(self, weights: Iterable[Tuple[str, torch.Tensor]]):
        """Qwen3.5 / Qwen3-Next checkpoint mapping: fused-GDN pre-pass
        (in_proj_qkvz/ba), conv1d/A_log/dt_bias TP shards, gated
        q_proj (q|gate per head) into the fused qkv, dense + MoE MLPs.
        """
        params = dict(self.named_parameters())
        tied = getattr(self.cfg, "tie_word_embeddings", False)
        stacked = [("gate_up_proj", "gate_proj", 0),
                   ("gate_up_proj", "up_proj", 1)]
        qkv_shards = {"q_proj": "q", "k_proj": "k", "v_proj": "v"}
        expert_map = [("gate_proj", "w13_weight", 0),
                      ("up_proj", "w13_weight", 1),
                      ("down_proj", "w2_weight", None)]
        for name, w in weights:
            if name.startswith("model."):
                name = name[len("model."):]
            if name.startswith("layers."):
                parts = name.split(".")
                g_idx = int(parts[1])
                if not (self.layer_start <= g_idx < self.layer_end):
                    continue
                local = g_idx - self.layer_start
                parts[1] = str(local)
                name = ".".join(parts)
                layer = self.layers[local]
                rest = ".".join(parts[2:])
                if rest.startswith("linear_attn."):
                    la = layer.linear_attn
                    sub = rest[len("linear_attn."):]
                    if sub == "in_proj_qkvz.weight":
                        la.load_fused_qkvz(w)
                    elif sub == "in_proj_ba.weight":
                        la.load_fused_ba(w)
                    elif sub == "conv1d.weight":
                        la._load_conv(la.conv1d_weight, w)
                    elif sub in ("A_log", "dt_bias"):
                        p = getattr(la, sub)
                        p.weight_loader(p, w)
                    elif sub == "norm.weight":
                        la.norm_weight.data.copy_(
                            w.to(la.norm_weight.dtype))
                    elif sub == "out_proj.weight":
                        p = la.out_proj.weight
                        p.weight_loader(p, w)
                    continue
                if rest.startswith("self_attn."):
                    sub = rest[len("self_attn."):]
                    base = sub.split(".")[0]
                    if base in qkv_shards:
                        p = layer.self_attn.qkv_proj.weight
                        p.weight_loader(p, w, qkv_shards[base])
                        continue
                    # o_proj / q_norm / k_norm fall through by name
                if ".experts." in name:
                    eidx = name.split(".").index("experts")
                    nparts = name.split(".")
                    expert_id = int(nparts[eidx + 1])
                    wname = nparts[eidx + 2]
                    prefix = ".".join(nparts[:eidx + 1])
                    for ckpt, fused, shard in expert_map:
                        if wname == ckpt:
                            p = params[f"{prefix}.{fused}"]
                            if shard is None:
                                p.weight_loader(p, w, expert_id)
                            else:
                                p.weight_loader(p, w, expert_id, shard)
                            break
                    continue
            elif name.startswith("embed_tokens"):
                if self.is_first_stage:
                    p = params["embed_tokens.weight"]
                    p.weight_loader(p, w)
                if tied and self.is_last_stage and not self.is_first_stage:
                    p = params["lm_head.weight"]
                    p.weight_loader(p, w)
                continue
            elif name.startswith("lm_head"):
                if self.is_last_stage and not tied:
                    p = params["lm_head.weight"]
                    p.weight_loader(p, w)
                continue
            elif name.startswith("norm."):
                if not self.is_last_stage:
                    continue
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
                    p.data.copy_(w.to(p.dtype))
