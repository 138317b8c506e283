"""ModelRunner: owns the model, KV cache, forward orchestration, sampling.

Parity: reference model_runner.py init/profile/KV-sizing/step_once
(:491-552, :1482-1523, :1809-1875). hipGraph decode capture lands in
runtime/graph_runner.py and is driven from here.
"""

from typing import List, Optional

import torch

from gllm_amd.config import EngineConfig
from gllm_amd.core.kv_cache import (KVCacheSpec, MemoryManager,
                                    PrefixMemoryManager)
from gllm_amd.core.scheduler import ScheduledBatch
from gllm_amd.layers.sampler import Sampler, build_sampling_metadata
from gllm_amd.logger import logger
from gllm_amd.models.loader import load_hf_config, load_model
from gllm_amd.runtime.batch_builder import BatchBuilder
from gllm_amd.runtime.forward_context import ForwardContext


class ModelRunner:
    def __init__(self, config: EngineConfig):
        self.config = config
        self.device = config.device
        self.model = None
        self.hf_config = None
        self.k_caches: List[torch.Tensor] = []
        self.v_caches: List[torch.Tensor] = []
        self.memory_manager: Optional[MemoryManager] = None
        self.sampler = Sampler()
        self.builder: Optional[BatchBuilder] = None
        self.graph_runner = None
        self.kv_dtype = None
        self.idx_caches = None
        self.index_head_dim = None

    # ------------------------------------------------------------------
    def init(self, num_pages_override: Optional[int] = None):
        cfg = self.config
        if cfg.device.startswith("cuda"):
            torch.cuda.set_device(cfg.device if ":" in cfg.device
                                  else "cuda:0")
        self.model, self.hf_config = load_model(cfg, cfg.device)
        self.kv_dtype = cfg.torch_dtype()
        if getattr(self.model, "ssm_spec", None) is not None:
            # prefix caching stays ON: page-boundary recurrent-state
            # snapshots restore on hits (core/ssm.py + the hit_filter
            # wired below). Graphs capture the batched GDN decode path
            # (device slot buffer + dummy scratch slot); batches with a
            # pending state restore fall back to eager (graph_runner
            # can_replay). Requires the gfx950 GDN kernels.
            from gllm_amd import ops
            spec = self.model.ssm_spec
            if not (cfg.device.startswith("cuda") and ops.has_kernels()
                    and spec.head_k_dim == 128 and spec.conv_kernel == 4):
                cfg.use_graph = False
        self.uses_mrope = bool(getattr(self.model, "uses_mrope", False))
        # mrope models capture with [3, B] position buffers (decode
        # positions are a per-seq scalar across all three sections);
        # prefix caching stays ON — image runs get content-hash cache
        # keys (multimodal/prepare.py + core/kv_cache.py _key_ids)
        self.index_head_dim = getattr(self.model, "index_head_dim", None)
        if self.index_head_dim:
            # DSA selector round-1 runs the eager per-seq torch path;
            # the graph-safe tile-static scorer is round 2 (ROADMAP.md)
            cfg.use_graph = False
        if cfg.dp_size > 1:
            # DP rounds gate forwards on dp_meta_barrier counts and the
            # MoE DP gather runs collectives inside the forward — both
            # incompatible with capture (graph-captured DP decode with
            # uniform gather is a round-2 item)
            cfg.use_graph = False
        else:
            from gllm_amd import ops
            from gllm_amd.layers.moe.layer import FusedMoE
            moe_mods = [m for m in self.model.modules()
                        if isinstance(m, FusedMoE)]
            if moe_mods and not (cfg.device.startswith("cuda")
                                 and ops.has_kernels()
                                 and all((m.fp8_block is None
                                          or m.fp8_block == (128, 128))
                                         and (m.int4_cfg is None
                                              or m.int4_cfg[1] == 128)
                                         for m in moe_mods)):
                # bf16, fp8(128-block) and int4(group-128) MoE on GPU
                # run the device-resident grouped MFMA GEMM pipelines
                # (ops.fused_moe / fused_moe_fp8 / fused_moe_int4) and
                # are capture-safe; other quant shapes keep the host
                # segment loop (illegal sync under capture)
                cfg.use_graph = False
        num_pages = num_pages_override or self._size_kv_cache()
        self._allocate_kv(num_pages)
        self.num_kv_pages_total = num_pages
        mgr_cls = PrefixMemoryManager if cfg.enable_prefix_caching \
            else MemoryManager
        # the LAST page is reserved as the hipGraph dummy/scratch page
        # (reference input_data.py:611-671 dummy-page padding trick)
        self.memory_manager = mgr_cls(num_pages - 1, cfg.page_size)
        self.builder = BatchBuilder(cfg.page_size, cfg.device)
        # overlap-mode sampled-token ring (FutureMap equivalent,
        # reference async_utils.py:56-61): next batch's decode inputs may
        # be negative placeholders resolved against this GPU buffer
        self.ring_slots = 8
        self.token_ring = torch.zeros(
            (self.ring_slots, cfg.maxd), dtype=torch.long,
            device=cfg.device)
        from gllm_amd.core.penalty import PenaltyPool
        self.penalty_pool = PenaltyPool(
            cfg.maxd + 64, getattr(self.hf_config, "vocab_size", 32000),
            cfg.device)
        self.memory_manager.free_hooks.append(self.penalty_pool.free)
        self.ssm_pool = None
        self.ssm_snapshot_enabled = True
        if getattr(self.model, "ssm_spec", None) is not None:
            from gllm_amd.core.ssm import SSMPool
            self.ssm_pool = SSMPool(self.model.ssm_spec, cfg.maxd + 64,
                                    cfg.device, dtype=self.kv_dtype)
            self.memory_manager.free_hooks.append(self.ssm_pool.free)
            if isinstance(self.memory_manager, PrefixMemoryManager):
                # hybrid prefix caching: hits only at boundaries whose
                # recurrent state was snapshotted; snapshots taken when
                # registration lands exactly on a page boundary
                pool = self.ssm_pool
                page = cfg.page_size

                def hit_filter(chains):
                    for k in range(len(chains), 0, -1):
                        if pool.has_snapshot(chains[k - 1]):
                            return k
                    return 0

                def on_register(seq, chains, n_full):
                    if not self.ssm_snapshot_enabled:
                        return
                    if seq.ssm_slot < 0 or not seq.ssm_state_ready:
                        return
                    if seq.computed_token_num == n_full * page:
                        pool.snapshot(chains[n_full - 1], seq.ssm_slot)

                self.memory_manager.hit_filter = hit_filter
                self.memory_manager.on_register = on_register
        from gllm_amd.parallel import get_pp_size, get_tp_size
        if (cfg.use_graph and cfg.device.startswith("cuda")
                and get_pp_size() == 1 and get_tp_size() == 1):
            from gllm_amd.runtime.graph_runner import GraphRunner
            self.graph_runner = GraphRunner(self)
            self.graph_runner.capture_all()
        return self

    # ------------------------------------------------------------------
    def kv_spec(self) -> KVCacheSpec:
        hf = self.hf_config
        # hybrid models allocate KV only for their full-attention layers
        num_local_layers = getattr(self.model, "num_kv_layers", None)
        if num_local_layers is None:
            num_local_layers = self.model.num_local_layers
        # prefer the model's own attention geometry (configs name kv
        # heads differently, e.g. ChatGLM's multi_query_group_num)
        attn = getattr(self.model, "kv_geometry", None)
        if attn is not None:
            if len(attn) == 3:
                kv_per_rank, head_dim, v_dim = attn
            else:
                kv_per_rank, head_dim = attn
                v_dim = head_dim
            shared = getattr(self.model, "kv_share_latent", None)
            return KVCacheSpec(num_local_layers, kv_per_rank, head_dim,
                               self.config.page_size,
                               dtype_bytes=self.kv_dtype.itemsize,
                               v_head_dim=v_dim,
                               v_shared=shared is not None)
        if True:
            total_kv = getattr(hf, "num_key_value_heads",
                               hf.num_attention_heads)
            from gllm_amd.parallel import get_tp_size
            tp = get_tp_size()
            kv_per_rank = max(1, total_kv // tp)
            head_dim = getattr(hf, "head_dim", None) or \
                hf.hidden_size // hf.num_attention_heads
        return KVCacheSpec(num_local_layers, kv_per_rank, head_dim,
                           self.config.page_size,
                           dtype_bytes=self.kv_dtype.itemsize)

    def _size_kv_cache(self) -> int:
        cfg = self.config
        spec = self.kv_spec()
        if not cfg.device.startswith("cuda"):
            # CPU test path: small fixed pool
            return 512
        self._profile_run()
        free, total = torch.cuda.mem_get_info()
        usable = total * cfg.gpu_memory_util - (total - free)
        num_pages = max(64, int(usable // spec.bytes_per_page))
        # min across ranks so every rank sizes identically
        from gllm_amd.parallel import get_world_size
        if get_world_size() > 1:
            import torch.distributed as dist
            t = torch.tensor([num_pages], dtype=torch.int64,
                             device=self.device)
            dist.all_reduce(t, op=dist.ReduceOp.MIN)
            num_pages = int(t.item())
        logger.info("KV cache: %d pages x %d tokens (%.1f GiB, %d layers)",
                    num_pages, cfg.page_size,
                    num_pages * spec.bytes_per_page / 2**30, spec.num_layers)
        return num_pages

    def _profile_run(self):
        """Peak-activation dummy forward to reserve torch workspace before
        sizing the KV cache (reference model_runner.py:1482-1523)."""
        cfg = self.config
        T = min(cfg.maxp, cfg.profile_batch)
        input_ids = torch.zeros(T, dtype=torch.long, device=self.device)
        positions = torch.arange(T, dtype=torch.long, device=self.device)
        fctx = ForwardContext(
            num_tokens=T, positions=positions,
            slot_mapping=torch.zeros(T, dtype=torch.long,
                                     device=self.device),
            block_table=torch.zeros((1, 1), dtype=torch.int32,
                                    device=self.device),
            seq_lens=torch.full((1,), T, dtype=torch.int32,
                                device=self.device),
            query_start_loc=torch.tensor([0, T], dtype=torch.int32,
                                         device=self.device),
            max_query_len=T, max_seq_len=T, k_caches=[], v_caches=[],
            is_profile_run=True)
        fctx.logits_indices = torch.tensor([T - 1], device=self.device)
        with torch.no_grad():
            hidden, residual = self._stage_forward(input_ids, positions, fctx)
            if self.model.is_last_stage:
                self.model.compute_logits(hidden, fctx)
        torch.cuda.synchronize()

    @torch.no_grad()
    def step_dummy(self) -> None:
        """1-token lockstep dummy forward for DP attention (reference
        worker.py:750-889): an idle replica must still enter every MoE
        collective of the round. ``is_profile_run`` makes attention a
        no-op (no KV touched); MoE layers run and join the DP gather."""
        T = 1
        dev = self.device
        fctx = ForwardContext(
            num_tokens=T,
            positions=torch.zeros(T, dtype=torch.long, device=dev),
            slot_mapping=torch.zeros(T, dtype=torch.long, device=dev),
            block_table=torch.zeros((1, 1), dtype=torch.int32, device=dev),
            seq_lens=torch.ones(1, dtype=torch.int32, device=dev),
            query_start_loc=torch.tensor([0, T], dtype=torch.int32,
                                         device=dev),
            max_query_len=T, max_seq_len=T, k_caches=[], v_caches=[],
            is_profile_run=True)
        if self.model.is_first_stage:
            input_ids = torch.zeros(T, dtype=torch.long, device=dev)
            self._stage_forward(input_ids, fctx.positions, fctx)
        else:
            # DP under PP: idle mid/last stages run their local layers on
            # zero hidden states (content irrelevant — only the MoE
            # collectives must be entered)
            z = torch.zeros(T, self.hf_config.hidden_size,
                            dtype=self.config.torch_dtype(), device=dev)
            self._stage_forward(None, fctx.positions, fctx,
                                hidden_states=z, residual=z.clone())

    def _allocate_kv(self, num_pages: int):
        spec = self.kv_spec()
        kshape = (num_pages, spec.page_size, spec.num_kv_heads,
                  spec.head_dim)
        vshape = (num_pages, spec.page_size, spec.num_kv_heads,
                  spec.v_head_dim)
        self.k_caches = [torch.zeros(kshape, dtype=self.kv_dtype,
                                     device=self.device)
                         for _ in range(spec.num_layers)]
        if spec.v_shared:
            # absorbed MLA: v = zero-copy view of the latent's first
            # v_head_dim dims
            self.v_caches = [k[..., :spec.v_head_dim]
                             for k in self.k_caches]
        else:
            self.v_caches = [torch.zeros(vshape, dtype=self.kv_dtype,
                                         device=self.device)
                             for _ in range(spec.num_layers)]
        # DSA (DeepSeek-V3.2): paged index-K cache parallel to the KV
        # pool, one per local layer (reference memory_manager.py:334-362)
        self.idx_caches = None
        if self.index_head_dim:
            ishape = (num_pages, spec.page_size, self.index_head_dim)
            self.idx_caches = [torch.zeros(ishape, dtype=self.kv_dtype,
                                           device=self.device)
                               for _ in range(spec.num_layers)]

    # ------------------------------------------------------------------
    def _stage_forward(self, input_ids, positions, fctx,
                       hidden_states=None, residual=None):
        return self.model(input_ids, positions, fctx,
                          hidden_states=hidden_states, residual=residual)

    @torch.no_grad()
    def step_first_stage(self, batch: ScheduledBatch):
        """Stage-0 forward. PP=1: returns sampled tokens. PP>1: returns
        (hidden, residual, fctx) for the PP send."""
        if self.graph_runner is not None and \
                self.graph_runner.can_replay(batch):
            return self.graph_runner.replay(batch)
        tokens, fctx = self.builder.build(
            batch, self.k_caches, self.v_caches,
            need_logits=self.model.is_last_stage,
            use_mrope=self.uses_mrope)
        self._attach_ssm(batch, fctx)
        self._attach_mm(batch, fctx)
        fctx.idx_caches = self.idx_caches
        if fctx.has_placeholders:
            tokens = self.resolve_tokens(tokens)
        hidden, residual = self._stage_forward(tokens, fctx.positions, fctx)
        if self.model.is_last_stage:
            return self._sample(batch, hidden, fctx)
        return hidden, residual, fctx

    @torch.no_grad()
    def step_mid_stage(self, batch: ScheduledBatch, hidden, residual):
        """PP stage > 0: forward received hidden states."""
        _, fctx = self.builder.build(
            batch, self.k_caches, self.v_caches,
            need_logits=self.model.is_last_stage,
            use_mrope=self.uses_mrope)
        self._attach_ssm(batch, fctx)
        fctx.idx_caches = self.idx_caches
        hidden, residual = self._stage_forward(
            None, fctx.positions, fctx, hidden_states=hidden,
            residual=residual)
        if self.model.is_last_stage:
            return self._sample(batch, hidden, fctx)
        return hidden, residual, fctx

    def _attach_ssm(self, batch, fctx) -> None:
        if self.ssm_pool is None:
            return
        fctx.ssm_pool = self.ssm_pool
        slots, has_init = [], []
        page = self.config.page_size
        for it in batch.items:
            seq = it.seq
            slot = self.ssm_pool.ensure(seq)
            if it.start > 0 and not seq.ssm_state_ready:
                # prefix-cache hit on a fresh slot: restore the boundary
                # snapshot (the hit_filter guaranteed it exists)
                chain = seq.page_hashes[it.start // page - 1]
                ok = self.ssm_pool.restore(chain, slot)
                assert ok, "prefix hit without an SSM snapshot"
            seq.ssm_state_ready = True
            slots.append(slot)
            has_init.append(it.start > 0)
        fctx.ssm_slots = slots
        fctx.ssm_has_init = has_init

    def _attach_mm(self, batch, fctx) -> None:
        """Collect vision-embedding rows for this batch's prefill chunks
        (chunk-aware: an image span may straddle chunk boundaries)."""
        rows, embeds = [], []
        qsl_off = 0
        for it in batch.items:
            seq = it.seq
            if seq.mm_embeds is not None and it.start < seq.prompt_len:
                emb_off = 0
                for (span_s, span_n) in seq.mm_spans:
                    lo = max(span_s, it.start)
                    hi = min(span_s + span_n, it.start + it.num_tokens)
                    if lo < hi:
                        rows.extend(range(qsl_off + lo - it.start,
                                          qsl_off + hi - it.start))
                        embeds.append(
                            seq.mm_embeds[emb_off + lo - span_s:
                                          emb_off + hi - span_s])
                    emb_off += span_n
            qsl_off += it.num_tokens
        if rows:
            fctx.mm_rows = torch.tensor(rows, dtype=torch.long,
                                        device=self.device)
            emb = torch.cat(embeds).to(self.device)
            H = self.hf_config.hidden_size
            if emb.shape[1] > H:
                # Qwen3-VL deepstack: [hidden | D*hidden] multiscale
                fctx.mm_embeds = emb[:, :H]
                fctx.mm_deepstack = emb[:, H:]
            else:
                fctx.mm_embeds = emb

    def resolve_tokens(self, tokens: torch.Tensor) -> torch.Tensor:
        """Replace negative placeholder ids with sampled tokens from the
        ring (overlap mode)."""
        ring_flat = self.token_ring.view(-1)
        idx = (-tokens - 1).clamp_min(0)
        return torch.where(tokens < 0, ring_flat[idx], tokens)

    def _prompt_logprobs(self, batch: ScheduledBatch, hidden, fctx):
        """Prompt logprobs (reference model_runner.py:1700-1807): for
        seqs requesting them, the prefill-chunk rows' logits score the
        NEXT prompt token; computed chunk by chunk so chunked prefill
        accumulates the full prompt. PP=1 scope (the values live where
        sampling runs)."""
        want = [i for i, it in enumerate(batch.items)
                if it.seq.sampling.prompt_logprobs
                and it.start < it.seq.prompt_len - 1]
        if not want:
            return
        import types
        qsl = fctx.host_qsl()
        shim = types.SimpleNamespace(logits_indices=None)
        for i in want:
            item = batch.items[i]
            seq = item.seq
            k = seq.sampling.prompt_logprobs
            qs, qe = qsl[i], qsl[i + 1]
            n_rows = qe - qs
            # row r predicts prompt position item.start + r + 1
            last = min(n_rows, seq.prompt_len - 1 - item.start)
            if last <= 0:
                continue
            # preemption recompute restarts the prefill: drop entries
            # this chunk is about to recompute
            del seq.prompt_logprobs_out[item.start:]
            logits = self.model.compute_logits(hidden[qs:qs + last], shim)
            logp = torch.log_softmax(logits.float(), dim=-1)
            targets = torch.tensor(
                seq.token_ids[item.start + 1:item.start + 1 + last],
                device=logp.device)
            chosen = logp.gather(1, targets.unsqueeze(1)).squeeze(1)
            topv, topi = logp.topk(min(k, logp.shape[-1]), dim=-1)
            for r in range(last):
                seq.prompt_logprobs_out.append(
                    (float(chosen[r]),
                     dict(zip(topi[r].tolist(), topv[r].tolist()))))

    def _sample(self, batch: ScheduledBatch, hidden, fctx):
        self._prompt_logprobs(batch, hidden, fctx)
        logits = self.model.compute_logits(hidden, fctx)
        meta = build_sampling_metadata(batch.items, logits.device,
                                       penalty_pool=self.penalty_pool)
        out = self.sampler(logits, meta)
        if meta.any_penalty and meta.penalty_slots is not None:
            # GPU-side mask update for rows that really sampled (no host
            # sync: tokens may still be in flight under overlap)
            rows = [i for i in meta.sample_rows
                    if int(meta.penalty_slots[i]) >= 0]                 if not logits.is_cuda else meta.sample_rows
            if rows:
                idx = torch.tensor(rows, dtype=torch.long,
                                   device=logits.device)
                slots = meta.penalty_slots.index_select(0, idx)
                ok = slots >= 0
                self.penalty_pool.append(slots[ok],
                                         out.next_tokens.index_select(
                                             0, idx)[ok])
        return out
