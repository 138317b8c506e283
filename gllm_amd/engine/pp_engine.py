"""Pipeline-parallel engine: one process per GPU over RCCL/xGMI.

MI355X-native design (differs from the reference's driver/follower
split, worker.py:354-391 + dist_schedule.py): EVERY rank runs an
identical deterministic replica of the scheduler on the same request
stream, so batch geometry is known everywhere and no schedule broadcast
/ delta-payload mirror is needed. Only two things move between ranks:

  * PP legs: hidden_states + residual [T, hidden] bf16, point-to-point
    send/recv — exactly one xGMI link per stage pair;
  * sampled tokens: broadcast from the last stage on a DEDICATED
    process group, so token broadcasts never interleave with the p2p
    stream on one communicator (ordering safety).

Pipelining: stage 0 keeps up to pp_size micro-batches in flight
(scheduler.batch_running bound); each rank processes batches in FIFO
order, so the pipeline fills across ranks.
"""

import time
from collections import deque
from typing import List, Optional

import torch
import torch.distributed as dist

from gllm_amd.config import EngineConfig
from gllm_amd.core.scheduler import ScheduledBatch, Scheduler
from gllm_amd.logger import logger
from gllm_amd.runtime.model_runner import ModelRunner
from gllm_amd.sequence import Sequence


class PPEngine:
    def __init__(self, config: EngineConfig,
                 num_pages_override: Optional[int] = None):
        from gllm_amd import parallel as P
        self.P = P
        P.init_distributed(config)
        self.config = config
        self.runner = ModelRunner(config).init(
            num_pages_override=num_pages_override)
        self.scheduler = Scheduler(config, self.runner.memory_manager)
        self.pp_rank = P.get_pp_rank()
        self.pp_size = P.get_pp_size()
        self.dp_rank = P.get_dp_rank()
        self.dp_size = P.get_dp_size()
        self._dp_global_work = 1 if self.dp_size > 1 else 0
        self.is_first = P.is_first_pp_rank()
        self.is_last = P.is_last_pp_rank()
        self.device = config.device
        self.hidden = self.runner.hf_config.hidden_size
        self.dtype = config.torch_dtype()
        # PP sends ride a dedicated HIP stream so the next micro-batch's
        # compute never queues behind a send that is waiting for the
        # receiver (reference dist_utils.py:8-22 uses isend; on RCCL a
        # side stream gives the same overlap with plain send)
        self._comm_stream = torch.cuda.Stream() \
            if config.device.startswith("cuda") else None
        from collections import deque
        self._send_bufs = deque(maxlen=max(2, self.pp_size))
        # dedicated communicator for token broadcasts — one per DP
        # replica (a replica's pipeline spans all pp stages x tp ranks
        # of its dp index; tokens differ per replica, so the broadcast
        # must not cross replicas)
        if P.get_world_size() > 1:
            stage = config.stage_size
            tokg = None
            for dp in range(self.dp_size):
                ranks = [pp * stage + dp * config.tp_size + t
                         for pp in range(self.pp_size)
                         for t in range(config.tp_size)]
                g = dist.new_group(ranks)
                if P.get_rank() in ranks:
                    tokg = g
            self.token_group = tokg
            self.last_stage_rank = (self.pp_size - 1) * stage + \
                self.dp_rank * config.tp_size
            # all-ranks control channel (serving intake broadcast from
            # global rank 0) — kept off the p2p default group
            self.ctrl_group = dist.new_group(
                list(range(P.get_world_size())))
        else:
            self.token_group = None
            self.ctrl_group = None
        # in-flight (batch, pending-state) on this rank
        self.inflight = deque()

    # ------------------------------------------------------------------
    def add_requests(self, seqs: List[Sequence]) -> None:
        self.scheduler.add_seqs(seqs)

    def _launch(self, batch: ScheduledBatch) -> None:
        """Run this rank's part of the forward for ``batch``."""
        if self.pp_size == 1:
            out = self.runner.step_first_stage(batch)
            self.inflight.append((batch, out))
            return
        if self.is_first:
            hidden, residual, _ = self.runner.step_first_stage(batch)
            self._send_pp(hidden, residual)
            self.inflight.append((batch, None))
        else:
            T = batch.num_tokens
            hidden, residual = self._recv_pp(T)
            out = self.runner.step_mid_stage(batch, hidden, residual)
            if self.is_last:
                self.inflight.append((batch, out))
            else:
                h2, r2, _ = out
                self._send_pp(h2, r2)
                self.inflight.append((batch, None))

    def _send_pp(self, hidden: torch.Tensor, residual: torch.Tensor):
        """ONE fused [2T, H] message per hop (the reference sends hidden
        and residual as two p2p ops), launched on the comm stream."""
        msg = torch.cat([hidden, residual], dim=0).contiguous()
        dst = self.P.get_next_pp_rank()
        if self._comm_stream is not None:
            evt = torch.cuda.Event()
            evt.record()  # msg materialized on the compute stream
            self._comm_stream.wait_event(evt)
            with torch.cuda.stream(self._comm_stream):
                dist.send(msg, dst=dst)
            # keep the buffer alive until the send drains (bounded by
            # <= pp_size micro-batches in flight)
            self._send_bufs.append(msg)
        else:
            dist.send(msg, dst=dst)

    def _recv_pp(self, T: int):
        msg = torch.empty((2 * T, self.hidden), dtype=self.dtype,
                          device=self.device)
        dist.recv(msg, src=self.P.get_prev_pp_rank())
        return msg[:T], msg[T:]

    @staticmethod
    def _stash_logprobs(batch, out) -> None:
        if out.logprobs is None:
            return
        lp = out.logprobs.tolist()
        tv = out.topk_logprobs.tolist()
        ti = out.topk_token_ids.tolist()
        for i, item in enumerate(batch.items):
            if not item.ends_prompt:
                continue
            if item.seq.sampling.logprobs:
                k = item.seq.sampling.logprobs
                item.seq.out_logprobs.append(
                    (lp[i], dict(zip(ti[i][:k], tv[i][:k]))))

    def _complete_oldest(self) -> List[Sequence]:
        batch, out = self.inflight.popleft()
        B = len(batch.items)
        if self.pp_size == 1:
            self._stash_logprobs(batch, out)
            tokens = out.next_tokens.tolist()
        else:
            if self.is_last:
                tok = out.next_tokens.to(self.device)
            else:
                tok = torch.empty(B, dtype=torch.long, device=self.device)
            dist.broadcast(tok, src=self.last_stage_rank,
                           group=self.token_group)
            tokens = tok.tolist()
            # per-token logprobs ride the same channel when requested
            # (reference worker.py:682-700 sends (next_tokens, logprobs,
            # prompt_logprobs) across stages); every rank derives the
            # same max-k from its replicated scheduler state
            max_lp = max((it.seq.sampling.logprobs or 0)
                         for it in batch.items)
            if max_lp > 0:
                from gllm_amd.layers.sampler import SamplerOutput
                if self.is_last:
                    lp = out.logprobs.float().to(self.device)
                    topv = out.topk_logprobs.float().to(self.device)
                    topi = out.topk_token_ids.to(self.device)
                else:
                    lp = torch.empty(B, dtype=torch.float32,
                                     device=self.device)
                    topv = torch.empty(B, max_lp, dtype=torch.float32,
                                       device=self.device)
                    topi = torch.empty(B, max_lp, dtype=torch.long,
                                       device=self.device)
                for t in (lp, topv, topi):
                    dist.broadcast(t, src=self.last_stage_rank,
                                   group=self.token_group)
                self._stash_logprobs(batch, SamplerOutput(tok, lp, topv,
                                                          topi))
        finished = self.scheduler.process_output(batch, tokens)
        if self.pp_size > 1:
            # prompt logprobs accumulate on the last stage (where
            # sampling runs); ship them to the other stages at finish
            for s in finished:
                if s.sampling.prompt_logprobs:
                    obj = [s.prompt_logprobs_out if self.is_last else None]
                    dist.broadcast_object_list(
                        obj, src=self.last_stage_rank,
                        group=self.token_group)
                    if not self.is_last:
                        s.prompt_logprobs_out = obj[0]
        return finished

    # ---------------------------------------------------- DP attention
    def dp_forward(self, batch: Optional[ScheduledBatch]) -> bool:
        """One lockstep DP-attention round (reference worker.py:750-889
        _schedule_forward_dp{,_pp} re-shaped for replicated schedulers):
        every replica enters a metadata barrier with (tokens scheduled
        this round, has pending work); if any replica scheduled tokens,
        ALL replicas forward — an idle one runs a 1-token dummy so the
        MoE DP-gather and EP all-reduce collectives stay matched. Under
        PP each stage runs the barrier on its OWN per-stage DP group
        with identical (replicated-scheduler) inputs, so every stage of
        a replica takes the same branch; dummies run each stage's local
        layers with no p2p. Returns True if a forward ran this round."""
        P = self.P
        nt = batch.num_tokens if batch is not None else 0
        counts, flags = P.dp_meta_barrier(nt, self.scheduler.has_work())
        self._dp_global_work = sum(flags) + sum(counts)
        if max(counts) == 0:
            return False
        P.set_dp_forward_counts(counts)
        try:
            if batch is None:
                self.runner.step_dummy()
            else:
                self._launch(batch)
        finally:
            P.set_dp_forward_counts(None)
        return True

    def _run_until_done_dp(self, max_steps: Optional[int] = None):
        done, steps = [], 0
        while True:
            b = self.scheduler.schedule_once() \
                if self.scheduler.has_work() else None
            ran = self.dp_forward(b)
            if self.inflight:
                done.extend(self._complete_oldest())
            if not ran and self._dp_global_work == 0:
                break
            steps += 1
            if max_steps is not None and steps >= max_steps:
                break
        return done

    # ------------------------------------------------------------------
    def run_until_done(self, max_steps: Optional[int] = None):
        """Drive the loop until all requests finish. Returns finished seqs."""
        if self.dp_size > 1:
            return self._run_until_done_dp(max_steps)
        done = []
        steps = 0
        while self.scheduler.has_work():
            launched = False
            while len(self.inflight) < max(1, self.pp_size):
                b = self.scheduler.schedule_once()
                if b is None:
                    break
                self._launch(b)
                launched = True
            if self.inflight:
                done.extend(self._complete_oldest())
            elif not launched:
                break
            steps += 1
            if max_steps is not None and steps >= max_steps:
                break
        return done

    def step_tick(self) -> int:
        """One pipeline tick: launch as many batches as fit, complete the
        oldest. Returns decode tokens committed this tick."""
        while len(self.inflight) < max(1, self.pp_size):
            b = self.scheduler.schedule_once()
            if b is None:
                break
            self._launch(b)
        if not self.inflight:
            return 0
        batch = self.inflight[0][0]
        n_sampled = sum(1 for it in batch.items if it.ends_prompt)
        self._complete_oldest()
        return n_sampled

    def drain(self) -> None:
        while self.inflight:
            self._complete_oldest()

    def barrier_sync(self) -> None:
        if self.token_group is not None:
            dist.barrier()
        if self.device.startswith("cuda"):
            torch.cuda.synchronize()
