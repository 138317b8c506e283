"""Continuous-batching scheduler.

Capability parity with the reference scheduler (gllm/scheduler.py):
  * three policies: ``chunked_prefill``, ``token_throttling`` (the SC'25
    paper's WT prefill budget), ``split_pd`` (prefill-priority);
  * SGLang-style adaptive decode-reserve admission control
    (``new_token_ratio`` rises on preemption, decays per tick);
  * largest-first preemption with full recompute;
  * balanced decode budget with deterministic rotating jitter so all TP
    column drivers stay in lockstep (reference scheduler.py:60-69);
  * up to ``pp_size`` batches in flight (PP micro-batch pipelining).

Design difference: a chunk in flight is represented by a ``ScheduledSeq``
record (seq, start, n) plus a per-seq ``scheduled`` cursor — not a
deepcopy of the sequence (reference scheduler.py:497-500) — so one long
prompt can occupy several pipeline stages at once without object cloning.
Determinism contract: every column driver runs this scheduler on identical
inputs and MUST produce identical batches; nothing here may consult a
random source or rank-dependent state.
"""

import dataclasses
import time
from collections import deque
from typing import Deque, Dict, List, Optional

from gllm_amd.core.kv_cache import MemoryManager, PrefixMemoryManager
from gllm_amd.logger import logger
from gllm_amd.sequence import Sequence


@dataclasses.dataclass
class ScheduledSeq:
    seq: Sequence
    start: int            # first token position computed this chunk
    num_tokens: int       # chunk length (1 for decode)

    @property
    def is_decode(self) -> bool:
        return self.start >= self.seq.prompt_len

    @property
    def ends_prompt(self) -> bool:
        """True if this chunk covers through the final prompt token, i.e.
        its logits row yields the first sampled token (or it is a decode)."""
        return self.start + self.num_tokens >= self.seq.prompt_len


@dataclasses.dataclass
class ScheduledBatch:
    items: List[ScheduledSeq]
    batch_id: int = 0

    @property
    def num_tokens(self) -> int:
        return sum(i.num_tokens for i in self.items)

    def __len__(self):
        return len(self.items)

    def __iter__(self):
        return iter(self.items)


class Scheduler:
    def __init__(self, config, memory_manager: MemoryManager):
        self.config = config
        self.mm = memory_manager
        self.pp_size = config.pp_size
        self.page_size = config.page_size
        self.maxp = config.maxp
        self.maxd = config.maxd
        self.minp = config.minp
        self.iterp = config.iterp
        self.schedule_method = config.schedule_method
        assert self.schedule_method in (
            "chunked_prefill", "token_throttling", "split_pd")

        # queues
        self.seqs_to_prefill: Deque[Sequence] = deque()
        self.seqs_to_decode: Deque[Sequence] = deque()
        # in-flight scheduled-but-not-committed batches (<= pp_size)
        self.batch_running: Deque[ScheduledBatch] = deque()
        # prefilling seqs whose prompt is fully *scheduled* but whose last
        # chunk has not committed yet (they re-enter seqs_to_decode there)
        self._scheduled_cursor: Dict[int, int] = {}

        # adaptive admission control
        self.new_token_ratio = config.init_new_token_ratio
        self.min_new_token_ratio = config.min_new_token_ratio
        self.new_token_ratio_step = 0.05
        self.new_token_ratio_decay = 0.002
        self.min_reserve_pages = max(1, self.pp_size)

        self.num_preempt_seqs = 0
        self.num_wait_tokens = 0
        self._decode_budget_jitter = 0
        self._batch_counter = 0
        self.abort_ids = set()
        self.log = True
        self._log_time = 0.0
        # world_size for the WT gate (single-GPU runs skip the ramp)
        self.world_size = config.world_size

    # ------------------------------------------------------------------
    # intake / abort
    # ------------------------------------------------------------------
    def add_seqs(self, seqs: List[Sequence]) -> None:
        for seq in seqs:
            self.seqs_to_prefill.append(seq)

    def abort_seqs(self, seq_ids) -> None:
        ids = set(seq_ids)
        # drop queued (not in-flight) seqs immediately
        for q in (self.seqs_to_prefill, self.seqs_to_decode):
            keep, dropped = [], []
            for s in q:
                if s.seq_id in ids and not self._in_flight(s):
                    dropped.append(s)
                else:
                    keep.append(s)
            q.clear()
            q.extend(keep)
            for s in dropped:
                s.finish_reason = Sequence.FINISH_ABORT
                self.mm.free_seq(s)
                ids.discard(s.seq_id)
        # Only ids this scheduler actually has in flight go to the
        # deferred set — an unknown id already finished here or belongs
        # to another DP replica; registering it would abort a FUTURE
        # request if the frontend recycles the seq id.
        known = set(self._scheduled_cursor) | \
            {s.seq_id for s in self.seqs_to_prefill} | \
            {s.seq_id for s in self.seqs_to_decode}
        self.abort_ids.update(ids & known)

    def _in_flight(self, seq: Sequence) -> bool:
        return seq.seq_id in self._scheduled_cursor and \
            self._scheduled_cursor[seq.seq_id] > seq.computed_token_num

    # ------------------------------------------------------------------
    # stats
    # ------------------------------------------------------------------
    def has_work(self) -> bool:
        return bool(self.seqs_to_prefill or self.seqs_to_decode
                    or self.batch_running)

    def get_num_decode_seqs(self) -> int:
        n = len(self.seqs_to_decode)
        for b in self.batch_running:
            n += sum(1 for it in b if it.start >= it.seq.prompt_len)
        return n

    # ------------------------------------------------------------------
    # admission reserve (SGLang-style)
    # ------------------------------------------------------------------
    def _seq_reserve_pages(self, seq: Sequence) -> int:
        if not seq.computed_prompt and not self._in_flight(seq):
            return 0
        remaining = seq.output_len - seq.num_output_tokens
        # clamp by the context cap: a request asking for 1e9 tokens must not
        # reserve more KV than the model could ever hold
        cap = (self.config.model_max_length or 32768) - len(seq.token_ids)
        remaining = min(remaining, max(0, cap))
        if remaining <= 0:
            return 0
        projected = self.new_token_ratio * remaining
        tail_slack = (-len(seq.token_ids)) % self.page_size
        need = projected - tail_slack
        return max(0, int(-(-need // self.page_size)))

    def _decode_reserve_pages(self) -> int:
        total = self.min_reserve_pages
        for seq in self.seqs_to_decode:
            total += self._seq_reserve_pages(seq)
        for b in self.batch_running:
            for it in b:
                if it.start >= it.seq.prompt_len:
                    total += self._seq_reserve_pages(it.seq)
        return total

    # ------------------------------------------------------------------
    # decode budget balancing (PP lockstep)
    # ------------------------------------------------------------------
    def get_balanced_decode_token_budget(self, num_decode_seqs: int) -> int:
        if self.pp_size == 1:
            return num_decode_seqs
        base, rem = divmod(num_decode_seqs, self.pp_size)
        bonus = 1 if (self._decode_budget_jitter % self.pp_size) < rem else 0
        self._decode_budget_jitter += 1
        return base + bonus

    # ------------------------------------------------------------------
    # preemption
    # ------------------------------------------------------------------
    def check_preempt(self, num_decode_scheduling: int) -> None:
        """Ensure the decode batch about to be scheduled can grow by one
        token per seq; preempt the longest queued decode seqs if not."""
        while True:
            need = 0
            for i, seq in enumerate(self.seqs_to_decode):
                if i >= num_decode_scheduling:
                    break
                pos = len(seq.token_ids)
                if pos % self.page_size == 0 or \
                        pos // self.page_size >= len(seq.page_table):
                    need += 1
            if need <= self.mm.get_num_free_pages():
                return
            victim_i = None
            victim_len = -1
            for i in range(len(self.seqs_to_decode) - 1, -1, -1):
                s = self.seqs_to_decode[i]
                if self._in_flight(s):
                    continue
                if len(s.token_ids) > victim_len:
                    victim_len = len(s.token_ids)
                    victim_i = i
            if victim_i is None:
                return  # nothing safely preemptible
            victim = self.seqs_to_decode[victim_i]
            del self.seqs_to_decode[victim_i]
            self.mm.free_seq(victim)
            victim.preempt()
            self._scheduled_cursor.pop(victim.seq_id, None)
            self.seqs_to_prefill.appendleft(victim)
            self.num_preempt_seqs += 1
            self.new_token_ratio = min(
                1.0, self.new_token_ratio + self.new_token_ratio_step)
            if self.num_preempt_seqs % 10 == 1:
                logger.warning(
                    "preempted seq %d (len %d); KV pressure high "
                    "(%d preemptions total)", victim.seq_id, victim_len,
                    self.num_preempt_seqs)

    # ------------------------------------------------------------------
    # batch construction
    # ------------------------------------------------------------------
    def _sched_cursor(self, seq: Sequence) -> int:
        return self._scheduled_cursor.get(seq.seq_id, seq.computed_token_num)

    def _schedule_prefill(self, token_budget: int, max_seqs: int,
                          reserve_pages: int):
        batch: List[ScheduledSeq] = []
        parked: List[Sequence] = []
        total_tokens = 0
        while (self.seqs_to_prefill and token_budget > 0
               and len(batch) < max_seqs):
            seq = self.seqs_to_prefill.popleft()
            if seq.seq_id in self.abort_ids:
                seq.finish_reason = Sequence.FINISH_ABORT
                self.mm.free_seq(seq)
                self.abort_ids.discard(seq.seq_id)
                continue
            # prefix-cache lookup on first touch
            if seq.computed_token_num == 0 and not seq.page_table:
                self.mm.lookup_prefix(seq)
            start = self._sched_cursor(seq)
            n = min(seq.prompt_len - start, token_budget)
            if n <= 0:
                parked.append(seq)
                continue
            # page feasibility against the reserve
            pages_have = len(seq.page_table)
            pages_total = -(-(start + n) // self.page_size)
            pages_to_alloc = max(0, pages_total - pages_have)
            headroom = self.mm.get_num_free_pages() - reserve_pages
            if pages_to_alloc > headroom:
                # shrink the chunk to the headroom, else park
                fit_tokens = (pages_have * self.page_size - start) + \
                    max(0, headroom) * self.page_size
                n = min(n, fit_tokens)
                if n <= 0:
                    parked.append(seq)
                    break
                pages_total = -(-(start + n) // self.page_size)
                pages_to_alloc = max(0, pages_total - pages_have)
            if pages_to_alloc:
                seq.page_table.extend(self.mm._allocate_fresh(pages_to_alloc))
            batch.append(ScheduledSeq(seq, start, n))
            total_tokens += n
            token_budget -= n
            self._scheduled_cursor[seq.seq_id] = start + n
            if start + n < seq.prompt_len:
                # prompt not fully scheduled: stays at the queue head so its
                # next chunk can enter the next micro-batch
                parked.append(seq)
        for seq in reversed(parked):
            self.seqs_to_prefill.appendleft(seq)
        return batch, total_tokens

    def _schedule_decode(self, budget: int) -> List[ScheduledSeq]:
        self.check_preempt(min(budget, len(self.seqs_to_decode)))
        batch: List[ScheduledSeq] = []
        for _ in range(budget):
            if not self.seqs_to_decode:
                break
            seq = self.seqs_to_decode.popleft()
            if seq.seq_id in self.abort_ids:
                seq.finish_reason = Sequence.FINISH_ABORT
                self.mm.free_seq(seq)
                self.abort_ids.discard(seq.seq_id)
                continue
            start = seq.computed_token_num
            pos = len(seq.token_ids)
            if pos > len(seq.page_table) * self.page_size:
                seq.page_table.extend(self.mm._allocate_fresh(1))
            batch.append(ScheduledSeq(seq, start, 1))
            self._scheduled_cursor[seq.seq_id] = start + 1
        return batch

    # ------------------------------------------------------------------
    # the tick
    # ------------------------------------------------------------------
    def schedule_once(self) -> Optional[ScheduledBatch]:
        if len(self.batch_running) >= self.pp_size:
            return None
        if self.schedule_method in ("chunked_prefill", "split_pd"):
            items = self._tick_chunked_prefill()
        else:
            items = self._tick_token_throttling()
        self.new_token_ratio = max(
            self.min_new_token_ratio,
            self.new_token_ratio - self.new_token_ratio_decay)
        if not items:
            return None
        self._batch_counter += 1
        batch = ScheduledBatch(items, self._batch_counter)
        self.batch_running.append(batch)
        self._maybe_log(items)
        return batch

    def _tick_chunked_prefill(self) -> List[ScheduledSeq]:
        budget = self.maxp
        reserve = self._decode_reserve_pages()
        n_decode_total = self.get_num_decode_seqs()
        decode_budget = min(
            self.get_balanced_decode_token_budget(n_decode_total), budget)
        prefer_prefill = (self.schedule_method == "split_pd"
                          and self.seqs_to_prefill
                          and self.mm.get_num_free_pages() >= reserve)
        if prefer_prefill:
            decode_budget = 0
        decode_batch = self._schedule_decode(decode_budget)
        budget -= len(decode_batch)
        budget = min(budget, self.page_size *
                     max(self.mm.get_num_free_pages() - reserve, 0))
        prefill_batch, _ = self._schedule_prefill(
            budget, max_seqs=self.maxd - len(decode_batch),
            reserve_pages=reserve)
        if prefer_prefill and not prefill_batch and not decode_batch:
            # deadlock guard (reference scheduler.py:569-574)
            fb = min(self.get_balanced_decode_token_budget(n_decode_total),
                     self.maxp)
            decode_batch = self._schedule_decode(fb)
        return decode_batch + prefill_batch

    def _tick_token_throttling(self) -> List[ScheduledSeq]:
        reserve = self._decode_reserve_pages()
        prefill_budget = self.page_size * max(
            self.mm.get_num_free_pages() - reserve, 0)
        if self.world_size > 1 and prefill_budget:
            self.num_wait_tokens = sum(
                max(0, s.prompt_len - self._sched_cursor(s))
                for s in self.seqs_to_prefill)
            free_ratio = self.mm.get_memory_free()
            reserve_ratio = min(0.99, reserve / max(1, self.mm.num_pages))
            prefill_ratio = max(
                0.0, (free_ratio - reserve_ratio) / (1 - reserve_ratio))
            prefill_budget = min(round(prefill_ratio * self.maxp),
                                 prefill_budget)
            if len(self.seqs_to_prefill) > 1:
                prefill_budget = min(
                    max(self.num_wait_tokens // self.iterp, self.minp),
                    prefill_budget)
        else:
            prefill_budget = min(self.maxp, prefill_budget)
        prefill_batch, _ = self._schedule_prefill(
            prefill_budget, max_seqs=self.maxd, reserve_pages=reserve)
        n_decode_total = self.get_num_decode_seqs()
        decode_budget = min(
            self.get_balanced_decode_token_budget(n_decode_total),
            self.maxd - len(prefill_batch))
        decode_batch = self._schedule_decode(decode_budget)
        return decode_batch + prefill_batch

    # ------------------------------------------------------------------
    # output processing
    # ------------------------------------------------------------------
    def process_output(self, batch: ScheduledBatch,
                       next_tokens: List[int]) -> List[Sequence]:
        """Commit a finished forward pass. ``next_tokens[i]`` is the sampled
        token for item i (meaningful only when the item ends its prompt or
        is a decode step). Returns seqs finished this tick."""
        assert self.batch_running and self.batch_running[0] is batch, \
            "batches must commit in FIFO order"
        self.batch_running.popleft()
        finished: List[Sequence] = []
        for item, tok in zip(batch.items, next_tokens):
            seq = item.seq
            seq.computed_token_num = item.start + item.num_tokens
            aborted = seq.seq_id in self.abort_ids
            if aborted:
                seq.finish_reason = Sequence.FINISH_ABORT
            if item.ends_prompt and not aborted:
                seq.append_token(int(tok))
                seq.check_finish()
            if seq.is_finished:
                if not self._in_flight(seq):
                    self.mm.free_seq(seq)
                    self._scheduled_cursor.pop(seq.seq_id, None)
                    self.abort_ids.discard(seq.seq_id)
                finished.append(seq)
            elif item.ends_prompt:
                if isinstance(self.mm, PrefixMemoryManager):
                    self.mm.register_computed_pages(seq)
                self.seqs_to_decode.append(seq)
            elif isinstance(self.mm, PrefixMemoryManager):
                # mid-prompt chunks publish their full pages too:
                # concurrent identical prompts share, and hybrid models
                # snapshot recurrent state at page-aligned chunk ends
                self.mm.register_computed_pages(seq)
            # chunks that don't end the prompt stay in seqs_to_prefill
            # (they were parked there at schedule time)
        return finished

    # ------------------------------------------------------------------
    # overlap mode: deferred finalize (reference OverlapScheduler,
    # scheduler.py:699-783). The sampled token is appended as a NEGATIVE
    # placeholder resolved on-GPU next step; finish checks run at
    # finalize once the real token lands on host.
    # ------------------------------------------------------------------
    def process_output_deferred(self, batch: ScheduledBatch,
                                placeholders: List[int]):
        """Commit cursors and append placeholder tokens. Returns records
        [(item_idx, seq, token_pos)] for finalize_output."""
        assert self.batch_running and self.batch_running[0] is batch
        self.batch_running.popleft()
        records = []
        for i, item in enumerate(batch.items):
            seq = item.seq
            seq.computed_token_num = item.start + item.num_tokens
            if not item.ends_prompt:
                continue
            if seq.is_finished:
                # finished retroactively while this chunk was in flight
                if not self._in_flight(seq):
                    self.mm.free_seq(seq)
                    self._scheduled_cursor.pop(seq.seq_id, None)
                continue
            seq.append_token(placeholders[i])
            records.append((i, seq, len(seq.token_ids) - 1))
            # keep decoding only if the output budget allows another step
            if seq.num_output_tokens < seq.sampling.max_tokens and                     seq.seq_id not in self.abort_ids:
                self.seqs_to_decode.append(seq)
        return records

    def finalize_output(self, batch: ScheduledBatch, tokens: List[int],
                        records) -> List[Sequence]:
        """Replace placeholders with real tokens; run finish checks.
        Also fills ``self.last_emissions`` with (seq_id, token, finish)
        tuples for the serving worker's output stream."""
        finished: List[Sequence] = []
        self.last_emissions = []
        for (i, seq, pos) in records:
            already_finished = seq.is_finished
            if already_finished:
                # extra speculative token of a retro-finished seq
                if pos < len(seq.token_ids) and seq.token_ids[pos] < 0:
                    del seq.token_ids[pos]
                continue
            if seq.token_ids[pos] < 0:
                seq.token_ids[pos] = int(tokens[i])
            if seq.seq_id in self.abort_ids:
                seq.finish_reason = Sequence.FINISH_ABORT
            # position-aware: a later placeholder may already sit beyond
            # pos when the pipeline runs >1 batch deep
            seq.check_finish(pos)
            if seq.is_finished:
                # drop speculative tokens appended after the finish
                del seq.token_ids[pos + 1:]
            elif isinstance(self.mm, PrefixMemoryManager):
                self.mm.register_computed_pages(seq)
            if seq.finish_reason == Sequence.FINISH_ABORT:
                self.last_emissions.append((seq.seq_id, -1,
                                            seq.finish_reason))
            else:
                self.last_emissions.append((seq.seq_id, seq.token_ids[pos],
                                            seq.finish_reason))
            if seq.is_finished:
                finished.append(seq)
                # drop from the decode queue if parked there
                try:
                    self.seqs_to_decode.remove(seq)
                except ValueError:
                    pass
                if not self._in_flight(seq):
                    self.mm.free_seq(seq)
                    self._scheduled_cursor.pop(seq.seq_id, None)
                    self.abort_ids.discard(seq.seq_id)
        return finished

    # ------------------------------------------------------------------
    def _maybe_log(self, items: List[ScheduledSeq]) -> None:
        if not self.log:
            return
        now = time.time()
        if now - self._log_time < 1.0:
            return
        self._log_time = now
        n_prefill_toks = sum(i.num_tokens for i in items
                             if i.start < i.seq.prompt_len)
        n_decode = sum(1 for i in items if i.start >= i.seq.prompt_len)
        msg = ("#wait: %4d #run: %4d #prefill: %4d #decode: %4d "
               "memory_util: %5.2f%%" % (
                   len(self.seqs_to_prefill), self.get_num_decode_seqs(),
                   n_prefill_toks, n_decode, self.mm.get_memory_util()))
        if isinstance(self.mm, PrefixMemoryManager):
            msg += " cache_hit_rate: %5.2f%%" % self.mm.get_cache_hit_rate()
        logger.info(msg)
