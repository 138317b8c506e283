"""hipGraph-captured decode buckets.

Parity: reference CUDA-graph capture (model_runner.py:1525-1615) with
the dummy-page padding trick (input_data.py:611-671), re-done on HIP:
torch.cuda.CUDAGraph IS hipGraph on ROCm. Pure-decode batches replay a
captured graph for the smallest bucket >= B; everything else runs eager.

Capture covers embed -> layers -> final norm (hidden out). Logits GEMM
and sampling stay eager (cheap, metadata-dependent). All graphs share
one memory pool; inputs live in persistent device buffers fed from
persistent pinned staging.
"""

from typing import Dict, List, Optional

import numpy as np
import torch

from gllm_amd.logger import logger
from gllm_amd.runtime.forward_context import ForwardContext

DECODE_BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256, 384, 512]


class GraphRunner:
    def __init__(self, runner):
        self.runner = runner
        cfg = runner.config
        self.device = cfg.device
        self.buckets = [b for b in DECODE_BUCKETS if b <= cfg.max_graph_bs]
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.hidden_out: Dict[int, torch.Tensor] = {}
        self.captured = False
        self.max_bs = self.buckets[-1]
        max_len = cfg.model_max_length or 32768
        self.pages_cap = -(-max_len // cfg.page_size)
        # the reserved scratch page (never allocated to a real seq)
        self.dummy_page = runner.num_kv_pages_total - 1

        B = self.max_bs
        dev = self.device
        self.uses_mrope = bool(getattr(runner, "uses_mrope", False))
        self.in_ids = torch.zeros(B, dtype=torch.long, device=dev)
        pshape = (3, B) if self.uses_mrope else (B,)
        self.positions = torch.zeros(pshape, dtype=torch.long, device=dev)
        self.slots = torch.zeros(B, dtype=torch.long, device=dev)
        self.block_table = torch.zeros((B, self.pages_cap),
                                       dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(B, dtype=torch.int32, device=dev)
        self.qsl = torch.arange(B + 1, dtype=torch.int32, device=dev)
        # pinned staging (one flat int64 area reused per copy)
        self.pin_ids = torch.zeros(B, dtype=torch.long).pin_memory()
        self.pin_pos = torch.zeros(pshape, dtype=torch.long).pin_memory()
        self.pin_slots = torch.zeros(B, dtype=torch.long).pin_memory()
        self.pin_seq_lens = torch.zeros(B, dtype=torch.int32).pin_memory()
        self.pin_bt = torch.zeros((B, self.pages_cap),
                                  dtype=torch.int32).pin_memory()
        # hybrid (GDN/SSM) models: persistent device slot buffer; all
        # captured rows are initialized decode steps, padding rows point
        # at the pool's dummy scratch slot
        self.ssm_slots_dev = None
        self.pin_ssm = None
        if runner.ssm_pool is not None:
            dummy = runner.ssm_pool.dummy_slot
            self.ssm_slots_dev = torch.full((B,), dummy, dtype=torch.long,
                                            device=dev)
            self.pin_ssm = torch.full((B,), dummy,
                                      dtype=torch.long).pin_memory()

    def _fctx_for(self, bs: int) -> ForwardContext:
        pos = self.positions[:, :bs] if self.uses_mrope \
            else self.positions[:bs]
        fctx = ForwardContext(
            num_tokens=bs,
            positions=pos,
            slot_mapping=self.slots[:bs],
            block_table=self.block_table[:bs],
            seq_lens=self.seq_lens[:bs],
            query_start_loc=self.qsl[:bs + 1],
            max_query_len=1, max_seq_len=1,
            k_caches=self.runner.k_caches,
            v_caches=self.runner.v_caches)
        if self.ssm_slots_dev is not None:
            fctx.ssm_pool = self.runner.ssm_pool
            fctx.ssm_slots_dev = self.ssm_slots_dev[:bs]
        return fctx

    @torch.no_grad()
    def capture_all(self):
        runner = self.runner
        # dummy metadata: every row points at the scratch page
        self.block_table.fill_(0)
        self.block_table[:, 0] = self.dummy_page
        self.seq_lens.fill_(1)
        base = self.dummy_page * runner.config.page_size
        ps = runner.config.page_size
        self.slots.copy_(base + torch.arange(self.max_bs) % ps)
        pool = torch.cuda.graphs.graph_pool_handle()
        stream = torch.cuda.Stream()
        torch.cuda.synchronize()
        # capture largest first so the shared pool is sized once
        for bs in reversed(self.buckets):
            fctx = self._fctx_for(bs)
            with torch.cuda.stream(stream):
                pos = self.positions[:, :bs] if self.uses_mrope \
                    else self.positions[:bs]
                for _ in range(2):  # warmup outside capture
                    ids = runner.resolve_tokens(self.in_ids[:bs])
                    runner.model(ids, pos, fctx)
            torch.cuda.current_stream().wait_stream(stream)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=pool, stream=stream):
                ids = runner.resolve_tokens(self.in_ids[:bs])
                hidden, _ = runner.model(ids, pos, fctx)
            self.graphs[bs] = g
            self.hidden_out[bs] = hidden
        torch.cuda.synchronize()
        self.captured = True
        logger.info("hipGraph capture done: %d decode buckets (max bs %d)",
                    len(self.graphs), self.max_bs)

    # ------------------------------------------------------------------
    def can_replay(self, batch) -> bool:
        if not self.captured:
            return False
        items = batch.items
        if len(items) > self.max_bs:
            return False
        if not all(it.num_tokens == 1 and it.start >= it.seq.prompt_len
                   for it in items):
            return False
        if self.ssm_slots_dev is not None:
            # a pending recurrent-state restore (prefix hit on a fresh
            # slot) needs the eager path's host-side copy
            if not all(it.seq.ssm_slot >= 0 and it.seq.ssm_state_ready
                       for it in items):
                return False
        return True

    def _bucket(self, n: int) -> int:
        for b in self.buckets:
            if b >= n:
                return b
        raise AssertionError

    @torch.no_grad()
    def replay(self, batch):
        items = batch.items
        B = len(items)
        bs = self._bucket(B)
        ps = self.runner.config.page_size
        ids = self.pin_ids.numpy()
        pos = self.pin_pos.numpy()
        slots = self.pin_slots.numpy()
        seq_lens = self.pin_seq_lens.numpy()
        bt = self.pin_bt.numpy()
        max_pages_used = 0
        for i, it in enumerate(items):
            seq = it.seq
            ids[i] = seq.token_ids[it.start]
            if self.uses_mrope:
                # decode rows: all three sections advance together from
                # the prompt's mrope delta (batch_builder.py math)
                pos[:, i] = (seq.mrope_delta + (it.start - seq.prompt_len)
                             if seq.mrope_positions is not None
                             else it.start)
            else:
                pos[i] = it.start
            page_tab = seq.page_table
            n_pages = len(page_tab)
            bt[i, :n_pages] = page_tab
            max_pages_used = max(max_pages_used, n_pages)
            slots[i] = page_tab[it.start // ps] * ps + it.start % ps
            seq_lens[i] = it.start + 1
        dummy_base = self.dummy_page * ps
        for i in range(B, bs):
            ids[i] = 0
            if self.uses_mrope:
                pos[:, i] = 0
            else:
                pos[i] = 0
            bt[i, 0] = self.dummy_page
            slots[i] = dummy_base + i % ps
            seq_lens[i] = 1
        if self.pin_ssm is not None:
            sl = self.pin_ssm.numpy()
            dummy = self.runner.ssm_pool.dummy_slot
            for i, it in enumerate(items):
                sl[i] = it.seq.ssm_slot
            for i in range(B, bs):
                sl[i] = dummy
            self.ssm_slots_dev[:bs].copy_(self.pin_ssm[:bs],
                                          non_blocking=True)
        # H2D into the captured buffers
        self.in_ids[:bs].copy_(self.pin_ids[:bs], non_blocking=True)
        if self.uses_mrope:
            self.positions[:, :bs].copy_(self.pin_pos[:, :bs],
                                         non_blocking=True)
        else:
            self.positions[:bs].copy_(self.pin_pos[:bs],
                                      non_blocking=True)
        self.slots[:bs].copy_(self.pin_slots[:bs], non_blocking=True)
        self.seq_lens[:bs].copy_(self.pin_seq_lens[:bs], non_blocking=True)
        npg = max(1, max_pages_used)
        self.block_table[:bs, :npg].copy_(self.pin_bt[:bs, :npg],
                                          non_blocking=True)
        self.graphs[bs].replay()
        hidden = self.hidden_out[bs][:B]
        fctx = self._fctx_for(bs)
        fctx.logits_indices = None  # hidden already one row per seq
        return self.runner._sample(batch, hidden, fctx)
