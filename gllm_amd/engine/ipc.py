"""Control plane: frontend <-> workers.

Parity with the reference's comm.py role (CPU-side control off the GPU
fabric). The reference uses zmq; this image ships no pyzmq, so the
transport is multiprocessing SimpleQueue pairs (same in-order delivery
guarantee, single-node). The surface is transport-agnostic: a TCP
backend can slot in for multi-node (launch-mode master/slave) later.

Because every worker runs an identical replicated scheduler
(engine/worker.py), the frontend broadcasts ONE ordered request stream
to every worker, and the single output rank pushes sampled tokens back.

Message format:
  ("req", msg_idx, {seq_id, token_ids, sampling, eos_token_id})
  ("abort", msg_idx, [seq_ids])
  ("cmd", msg_idx, name)                 # profile start/stop, shutdown
  outputs: ("out", [(seq_id, token_id, finish_reason)], stats)
"""

import queue as pyqueue
from typing import Any, List, Optional

import torch.multiprocessing as mp


class FrontendComm:
    def __init__(self, world_size: int,
                 local_ranks: Optional[List[int]] = None):
        ctx = mp.get_context("spawn")
        # master launch mode: only this node's ranks get local queues;
        # the remote_sender hook (engine/multinode.py MasterRelay) ships
        # the same ordered stream to slave nodes over TCP
        self.local_ranks = list(local_ranks) if local_ranks is not None \
            else list(range(world_size))
        self.req_queues = {r: ctx.Queue() for r in self.local_ranks}
        self.out_queue = ctx.Queue()
        self.remote_sender = None
        self._msg_idx = 0

    def worker_endpoints(self, rank: int):
        """Picklable handles passed to the spawned worker."""
        return self.req_queues[rank], self.out_queue

    def send_to_all(self, kind: str, payload: Any) -> int:
        idx = self._msg_idx
        self._msg_idx += 1
        msg = (kind, idx, payload)
        for q in self.req_queues.values():
            q.put(msg)
        if self.remote_sender is not None:
            self.remote_sender(msg)
        return idx

    def recv_output(self, timeout_ms: Optional[int] = None):
        try:
            if timeout_ms is None:
                return self.out_queue.get()
            return self.out_queue.get(timeout=timeout_ms / 1000.0)
        except pyqueue.Empty:
            return None

    def close(self):
        pass


class WorkerComm:
    def __init__(self, req_queue, out_queue, is_output_rank: bool):
        self.req_queue = req_queue
        self.out_queue = out_queue if is_output_rank else None
        self._peeked: List[tuple] = []

    def drain(self, max_msgs: Optional[int] = None) -> List[tuple]:
        """Non-blocking: receive whatever is queued (in order)."""
        out = list(self._peeked)
        self._peeked = []
        while max_msgs is None or len(out) < max_msgs:
            try:
                out.append(self.req_queue.get_nowait())
            except pyqueue.Empty:
                break
        return out

    def recv_blocking(self, n: int, timeout_s: float = 60.0) -> List[tuple]:
        """Receive exactly n messages (the intake count agreed by rank 0)."""
        out = list(self._peeked[:n])
        self._peeked = self._peeked[n:]
        while len(out) < n:
            out.append(self.req_queue.get(timeout=timeout_s))
        return out

    def poll(self, timeout_ms: int) -> bool:
        if self._peeked:
            return True
        try:
            self._peeked.append(
                self.req_queue.get(timeout=timeout_ms / 1000.0))
            return True
        except pyqueue.Empty:
            return False

    def send_output(self, payload) -> None:
        assert self.out_queue is not None
        self.out_queue.put(payload)

    def close(self):
        pass
