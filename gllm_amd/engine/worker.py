"""Per-GPU serving worker: replicated-scheduler event loop.

Design (MI355X-native, replaces the reference's column-driver /
follower-mirror split, worker.py + dist_schedule.py):

Every rank runs an IDENTICAL deterministic scheduler over the SAME
ordered request stream. The only cross-rank coordination is:
  1. intake sync — global rank 0 decides how many queued control
     messages to admit before each scheduling round and broadcasts that
     count (one tiny collective); every rank then consumes exactly that
     many messages from its own in-order zmq queue, so all schedulers
     stay in lockstep;
  2. PP hidden-state legs (RCCL send/recv over one xGMI link each);
  3. sampled-token broadcast from the last stage.
Nothing else moves: no schedule payloads, no follower KV mirrors.

Global rank 0 is the output rank: after the token broadcast it owns the
full output stream and pushes (seq_id, token, finish) to the frontend.
"""

import os
import time
from typing import List

import torch
import torch.distributed as dist

from gllm_amd.config import EngineConfig
from gllm_amd.engine.ipc import WorkerComm
from gllm_amd.engine.overlap_engine import OverlapEngine
from gllm_amd.engine.pp_engine import PPEngine
from gllm_amd.logger import logger
from gllm_amd.sequence import SamplingParams, Sequence


class ServingMixin:
    """Intake sync, control commands, stats and output emission shared by
    the sync (PP) and overlap serving workers."""

    def _init_serving(self, config: EngineConfig, req_queue, out_queue):
        self._last_stats = 0.0
        from gllm_amd.parallel import (get_dp_rank, get_dp_size, get_rank,
                                       get_tp_rank, get_world_size)
        self.rank = get_rank()
        self.world = get_world_size()
        self.dp_rank = get_dp_rank()
        self.dp_size = get_dp_size()
        # DP: every replica emits its own outputs (pp-0/tp-0 rank of
        # the replica). Non-DP: global rank 0.
        if self.dp_size > 1:
            from gllm_amd.parallel import get_pp_rank
            self.is_output_rank = get_tp_rank() == 0 and get_pp_rank() == 0
        else:
            self.is_output_rank = self.rank == 0
        self._req_counter = 0  # deterministic DP round-robin routing
        self._seqs_by_id = {}  # live seqs (for prompt-logprob emission)
        self._emit_counts = {}  # seq_id -> tokens emitted (logprob idx)
        self.comm = WorkerComm(req_queue, out_queue, self.is_output_rank)
        self._intake_buf = torch.zeros(1, dtype=torch.int64)
        if self.world > 1 and config.device.startswith("cuda"):
            self._intake_buf = self._intake_buf.to(config.device)
        self.profiler = None
        self.shutdown = False

    # ------------------------------------------------------------------
    def _apply_messages(self, msgs: List[tuple]) -> None:
        for kind, idx, payload in msgs:
            if kind == "req":
                # every rank sees every request in the same order;
                # DP replicas take theirs round-robin (reference
                # llm_engine.py:490-519 routes at the frontend — here
                # routing is a deterministic function of the stream)
                target = self._req_counter % self.dp_size
                self._req_counter += 1
                if target != self.dp_rank:
                    continue
                seq = Sequence(payload["seq_id"], payload["token_ids"],
                               SamplingParams(**payload["sampling"]),
                               eos_token_id=payload.get("eos_token_id"))
                if payload.get("mm") is not None:
                    from gllm_amd.multimodal.prepare import prepare_mm_seq
                    prepare_mm_seq(self.runner.model, seq, payload["mm"])
                self._seqs_by_id[seq.seq_id] = seq
                self.scheduler.add_seqs([seq])
            elif kind == "abort":
                self.scheduler.abort_seqs(payload)
            elif kind == "cmd":
                self._handle_cmd(payload)

    def _handle_cmd(self, name: str) -> None:
        if name == "shutdown":
            self.shutdown = True
        elif name == "start_profile":
            d = os.environ.get("GLLM_TORCH_PROFILER_DIR", "/tmp/gllm_prof")
            os.makedirs(d, exist_ok=True)
            self.profiler = torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CPU,
                            torch.profiler.ProfilerActivity.CUDA],
                with_stack=False)
            self.profiler.__enter__()
            logger.info("profiler started -> %s", d)
        elif name == "stop_profile" and self.profiler is not None:
            self.profiler.__exit__(None, None, None)
            d = os.environ.get("GLLM_TORCH_PROFILER_DIR", "/tmp/gllm_prof")
            trace = os.path.join(d, f"rank{self.rank}.json")
            try:
                self.profiler.export_chrome_trace(trace)
            except Exception as e:  # pragma: no cover
                logger.warning("trace export failed: %s", e)
            self.profiler = None
            logger.info("profiler stopped; trace at %s", trace)

    # ------------------------------------------------------------------
    def _sync_intake(self, block: bool) -> None:
        """Rank 0 counts queued messages (optionally waiting for at least
        one); every rank admits exactly that count, in order."""
        if self.world == 1:
            if block and not self.comm.poll(100):
                return
            self._apply_messages(self.comm.drain())
            return
        if self.rank == 0:
            if block:
                self.comm.poll(100)
            msgs = self.comm.drain()
            self._intake_buf.fill_(len(msgs))
            dist.broadcast(self._intake_buf, src=0, group=self.ctrl_group)
            self._apply_messages(msgs)
        else:
            dist.broadcast(self._intake_buf, src=0, group=self.ctrl_group)
            n = int(self._intake_buf.item())
            if n:
                self._apply_messages(self.comm.recv_blocking(n))

    def _token_logprob(self, seq: Sequence):
        """The just-emitted token's (logprob, {top-k}) when requested
        (reference sampler logprobs served per stream chunk)."""
        if not seq.sampling.logprobs or not seq.out_logprobs:
            return None
        idx = self._emit_counts.get(seq.seq_id, 0)
        self._emit_counts[seq.seq_id] = idx + 1
        if idx < len(seq.out_logprobs):
            return seq.out_logprobs[idx]
        return None

    def _emit_prompt_logprobs(self, seq: Sequence) -> None:
        """Ship prompt logprobs BEFORE the finish token so the frontend
        attaches them to the final stream chunk."""
        if seq.sampling.prompt_logprobs and seq.prompt_logprobs_out:
            self.comm.send_output(
                ("plp", [(seq.seq_id, seq.prompt_logprobs_out)], {}))

    # ------------------------------------------------------------------
    def _maybe_send_stats(self) -> None:
        # one stats stream (rank 0 reports its own replica under DP)
        if not self.is_output_rank or self.rank != 0:
            return
        now = time.time()
        if now - self._last_stats < 1.0:
            return
        self._last_stats = now
        sched = self.scheduler
        mm = self.runner.memory_manager
        stats = {
            "num_waiting": len(sched.seqs_to_prefill),
            "num_running": sched.get_num_decode_seqs(),
            "num_preempted_total": sched.num_preempt_seqs,
            "kv_memory_util_pct": round(mm.get_memory_util(), 2),
            "kv_pages_free": mm.get_num_free_pages(),
            "prefix_cache_hit_rate_pct": round(mm.get_cache_hit_rate(), 2),
        }
        self.comm.send_output(("stats", stats, {}))

    def close(self):
        self.comm.close()


class ServingWorker(ServingMixin, PPEngine):
    """Synchronous serving worker (any PP size)."""

    def __init__(self, config: EngineConfig, req_queue, out_queue):
        PPEngine.__init__(self, config)
        self._init_serving(config, req_queue, out_queue)

    def _complete_oldest(self) -> List[Sequence]:
        batch, _ = self.inflight[0]
        finished = super()._complete_oldest()
        # tokens just appended: emit each sampled token
        outs = []
        if self.is_output_rank:
            for item in batch.items:
                if not item.ends_prompt:
                    continue
                seq = item.seq
                if seq.finish_reason == Sequence.FINISH_ABORT:
                    outs.append((seq.seq_id, -1, seq.finish_reason, None))
                else:
                    outs.append((seq.seq_id, seq.token_ids[-1],
                                 seq.finish_reason,
                                 self._token_logprob(seq)))
                if seq.finish_reason:
                    self._emit_prompt_logprobs(seq)
            if outs:
                self.comm.send_output(("out", outs, {}))
        for s in finished:
            self._seqs_by_id.pop(s.seq_id, None)
            self._emit_counts.pop(s.seq_id, None)
        return finished

    def _run_loop_dp(self) -> None:
        """DP-attention serving loop: lockstep rounds over the replica
        grid (idle replicas run dummies so MoE collectives match)."""
        logger.info("worker %d ready (dp=%d of %d, lockstep)",
                    self.rank, self.dp_rank, self.dp_size)
        while not self.shutdown:
            # rank 0 blocks on intake only when the LAST barrier said the
            # whole grid is idle (its own replica included)
            block = (not self.scheduler.has_work() and not self.inflight
                     and self._dp_global_work == 0)
            self._sync_intake(block=block)
            if self.shutdown:
                break
            b = self.scheduler.schedule_once() \
                if self.scheduler.has_work() else None
            self.dp_forward(b)
            if self.inflight:
                self._complete_oldest()
            self._maybe_send_stats()
        self.drain()
        self.comm.close()
        logger.info("worker %d shut down", self.rank)

    def run_loop(self) -> None:
        if self.dp_size > 1:
            return self._run_loop_dp()
        logger.info("worker %d ready (pp=%d, sync)", self.rank, self.pp_size)
        while not self.shutdown:
            has_work = self.scheduler.has_work()
            self._sync_intake(block=not has_work)
            if self.shutdown:
                break
            if not self.scheduler.has_work():
                # keep the 1 Hz stats line fresh while idle (the last
                # busy-loop snapshot may still show reaped seqs)
                self._maybe_send_stats()
                continue
            while len(self.inflight) < max(1, self.pp_size):
                b = self.scheduler.schedule_once()
                if b is None:
                    break
                self._launch(b)
            if self.inflight:
                self._complete_oldest()
            self._maybe_send_stats()
        self.drain()
        self.comm.close()
        logger.info("worker %d shut down", self.rank)


class OverlapServingWorker(ServingMixin, OverlapEngine):
    """PP=1 serving worker on the launch-first/collect-later engine."""

    def __init__(self, config: EngineConfig, req_queue, out_queue):
        OverlapEngine.__init__(self, config)
        self._init_serving(config, req_queue, out_queue)
        if self.is_output_rank:
            self.on_finalized = self._emit

    def _emit(self, emissions) -> None:
        out = []
        for seq_id, tok, fin in emissions:
            seq = self._seqs_by_id.get(seq_id)
            lp = self._token_logprob(seq) if seq is not None and \
                tok >= 0 else None
            out.append((seq_id, tok, fin, lp))
            if fin:
                self._seqs_by_id.pop(seq_id, None)
                self._emit_counts.pop(seq_id, None)
                if seq is not None:
                    self._emit_prompt_logprobs(seq)
        self.comm.send_output(("out", out, {}))

    def run_loop(self) -> None:
        logger.info("worker %d ready (overlap%s)", self.rank,
                    f", dp={self.dp_rank} of {self.dp_size}"
                    if self.dp_size > 1 else "")
        while not self.shutdown:
            has_work = self.scheduler.has_work() or bool(self.pending)
            self._sync_intake(block=not has_work
                              and self._dp_global_work == 0)
            if self.shutdown:
                break
            if not (self.scheduler.has_work() or self.pending
                    or self._dp_global_work > 0):
                self._maybe_send_stats()
                continue
            self.step_tick()
            self._maybe_send_stats()
        self.drain()
        self.comm.close()
        logger.info("worker %d shut down", self.rank)


def run_worker(rank: int, config: EngineConfig, req_queue, out_queue,
               ready_queue=None) -> None:
    os.environ["RANK"] = str(rank)
    # master/slave launch: this node hosts config.worker_ranks; the
    # device index is the rank's position within them
    local = rank - min(config.worker_ranks) if config.worker_ranks \
        else rank
    os.environ.setdefault("LOCAL_RANK", str(local))
    os.environ["MASTER_ADDR"] = config.master_addr
    os.environ["MASTER_PORT"] = str(config.master_port)
    if config.device.startswith("cuda"):
        config.device = f"cuda:{local % max(1, torch.cuda.device_count())}"
    try:
        # overlap covers PP=1 worlds (incl. DP replicas and TP shards);
        # multi-stage pipelines use the sync PP worker
        use_overlap = (config.pp_size == 1 and config.enable_overlap
                       and config.device.startswith("cuda"))
        cls = OverlapServingWorker if use_overlap else ServingWorker
        worker = cls(config, req_queue, out_queue)
        if ready_queue is not None:
            ready_queue.put(("ready", rank))
        worker.run_loop()
    except Exception as e:  # pragma: no cover
        logger.exception("worker %d died: %s", rank, e)
        if ready_queue is not None:
            ready_queue.put(("dead", rank))
        raise
