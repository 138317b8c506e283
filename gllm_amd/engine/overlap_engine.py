"""Launch-first / collect-later engine (PP = 1).

Parity: reference OverlapWorker + OverlapModelRunner + FutureMap
(overlap_worker.py, model_runner.py:1961-2311, async_utils.py). The CPU
side schedules and launches batch b+1 while batch b still runs on the
GPU; b+1's decode inputs carry negative placeholders resolved on-GPU
against the sampled-token ring, and b's outputs are finalized only after
an async D2H lands (event-gated). Up to DEPTH batches stay in flight.
"""

import collections
from typing import List, Optional

import torch

from gllm_amd.config import EngineConfig
from gllm_amd.engine.pp_engine import PPEngine
from gllm_amd.sequence import Sequence


class OverlapEngine(PPEngine):
    DEPTH = 2

    def __init__(self, config: EngineConfig,
                 num_pages_override: Optional[int] = None):
        super().__init__(config, num_pages_override=num_pages_override)
        assert self.pp_size == 1, "overlap engine is the PP=1 fast path"
        # deferred finalize registers pages AFTER later batches may have
        # advanced the recurrent state past the boundary — snapshots
        # would be stale, so hybrid models skip them under overlap
        self.runner.ssm_snapshot_enabled = False
        self.maxd = config.maxd
        self.ring_slots = self.runner.ring_slots
        self._slot = 0
        # pending: (batch, records, event, pinned_tokens, n_rows)
        self.pending = collections.deque()
        self.is_cuda = config.device.startswith("cuda")
        if self.is_cuda:
            self._pinned = [torch.zeros(config.maxd, dtype=torch.long)
                            .pin_memory() for _ in range(self.ring_slots)]
        self._finished_since: List[Sequence] = []
        # serving hook: called after each finalize with the scheduler's
        # last_emissions list
        self.on_finalized = None

    # ------------------------------------------------------------------
    def _launch_overlap(self, batch) -> None:
        slot = self._slot
        self._slot = (self._slot + 1) % self.ring_slots
        out = self.runner.step_first_stage(batch)   # SamplerOutput, async
        B = len(batch.items)
        ring = self.runner.token_ring
        ring[slot, :B].copy_(out.next_tokens)
        phs = [-(slot * self.maxd + i) - 1 for i in range(B)]
        records = self.scheduler.process_output_deferred(batch, phs)
        lp_host = None
        if self.is_cuda:
            pinned = self._pinned[slot]
            pinned[:B].copy_(out.next_tokens, non_blocking=True)
            if out.logprobs is not None:
                # logprobs ride the same event-gated async D2H as the
                # tokens (pinned staging per ring slot) — no sync in
                # the launch path (r1 traded the overlap away here)
                lp_host = self._lp_stage(slot, out)
            ev = torch.cuda.Event()
            ev.record()
        else:
            pinned = out.next_tokens
            ev = None
            if out.logprobs is not None:
                lp_host = (out.logprobs, out.topk_logprobs,
                           out.topk_token_ids)
        self.pending.append((batch, records, ev, pinned, B, lp_host))

    def _lp_stage(self, slot, out):
        """Async-copy the logprob tensors into per-slot pinned staging
        (grown when a request raises top-k)."""
        if not hasattr(self, "_lp_pinned"):
            self._lp_pinned = {}
        B, K = out.topk_logprobs.shape
        cur = self._lp_pinned.get(slot)
        if cur is None or cur[1].shape[1] < K:
            cur = (torch.zeros(self.maxd).pin_memory(),
                   torch.zeros(self.maxd, K).pin_memory(),
                   torch.zeros(self.maxd, K, dtype=torch.long)
                   .pin_memory())
            self._lp_pinned[slot] = cur
        lp, tv, ti = cur
        lp[:B].copy_(out.logprobs.float(), non_blocking=True)
        tv[:B, :K].copy_(out.topk_logprobs.float(), non_blocking=True)
        ti[:B, :K].copy_(out.topk_token_ids, non_blocking=True)
        return (lp[:B], tv[:B, :K], ti[:B, :K])

    def _collect_one(self) -> int:
        batch, records, ev, pinned, B, lp_host = self.pending.popleft()
        if ev is not None:
            ev.synchronize()
        if lp_host is not None:
            from types import SimpleNamespace
            self._stash_logprobs(batch, SimpleNamespace(
                logprobs=lp_host[0], topk_logprobs=lp_host[1],
                topk_token_ids=lp_host[2]))
        tokens = pinned[:B].tolist()
        finished = self.scheduler.finalize_output(batch, tokens, records)
        self._finished_since.extend(finished)
        if self.on_finalized is not None and \
                getattr(self.scheduler, "last_emissions", None):
            self.on_finalized(self.scheduler.last_emissions)
        return len(records)

    # ------------------------------------------------------------------
    def step_tick(self) -> int:
        """Returns sampled tokens finalized during this tick."""
        if self.dp_size > 1:
            return self._step_tick_dp()
        n_final = 0
        while len(self.pending) >= self.DEPTH:
            n_final += self._collect_one()
        batch = self.scheduler.schedule_once()
        if batch is None:
            while self.pending:
                n_final += self._collect_one()
            return n_final
        # repetition penalty reads real token histories: drain first
        if any(it.seq.sampling.repetition_penalty != 1.0
               for it in batch.items):
            while self.pending:
                n_final += self._collect_one()
        self._launch_overlap(batch)
        return n_final

    def _step_tick_dp(self) -> int:
        """DP-attention overlap tick (reference overlap_worker.py
        :258-309): one dp_meta_barrier per tick; a replica with nothing
        scheduled launches a 1-token dummy so the MoE collectives stay
        matched. Collection stays local (pp=1: no cross-rank traffic in
        finalize)."""
        P = self.P
        n_final = 0
        while len(self.pending) >= self.DEPTH:
            n_final += self._collect_one()
        batch = self.scheduler.schedule_once() \
            if self.scheduler.has_work() else None
        nt = batch.num_tokens if batch is not None else 0
        counts, flags = P.dp_meta_barrier(
            nt, self.scheduler.has_work() or bool(self.pending))
        self._dp_global_work = sum(flags) + sum(counts)
        if max(counts) == 0:
            while self.pending:
                n_final += self._collect_one()
            return n_final
        if batch is not None and any(
                it.seq.sampling.repetition_penalty != 1.0
                for it in batch.items):
            while self.pending:
                n_final += self._collect_one()
        P.set_dp_forward_counts(counts)
        try:
            if batch is None:
                self.runner.step_dummy()
            else:
                self._launch_overlap(batch)
        finally:
            P.set_dp_forward_counts(None)
        return n_final

    def pop_finished(self) -> List[Sequence]:
        out = self._finished_since
        self._finished_since = []
        return out

    def run_until_done(self, max_steps: Optional[int] = None):
        done = []
        steps = 0
        while self.scheduler.has_work() or self.pending or \
                (self.dp_size > 1 and self._dp_global_work > 0):
            self.step_tick()
            done.extend(self.pop_finished())
            steps += 1
            if max_steps is not None and steps >= max_steps:
                break
        return done

    def drain(self) -> None:
        while self.pending:
            self._collect_one()
