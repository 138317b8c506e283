"""Multi-node (master/slave) launch: the control plane over TCP.

Parity: reference launch modes (api_server.py --launch-mode
normal/master/slave, comm.py zmq socket topology + port bootstrap over
send/recv_object_list). The DATA plane — RCCL collectives, PP
send/recv, the intake-count broadcast — is torch.distributed with a TCP
rendezvous at master_addr:master_port and already spans nodes; what
must cross nodes explicitly is the CONTROL plane:

  * the ordered request stream (req / abort / cmd) master -> slaves,
  * sampled-token outputs from slave-local output ranks -> master.

The master listens on ``relay_port`` (default master_port + 1). Each
slave node connects, sends a hello naming the ranks it hosts, then
receives the exact message stream the master's local workers get over
mp queues and fans it into its own local workers (TCP per-connection
FIFO preserves the replicated schedulers' ordering guarantee). Output
tuples flow back over the same socket into the master's out queue.
"""

import pickle
import socket
import struct
import threading
import time
from typing import Dict, List, Optional

from gllm_amd.logger import logger

_LEN = struct.Struct("!I")


def send_msg(sock: socket.socket, obj) -> None:
    data = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    sock.sendall(_LEN.pack(len(data)) + data)


def recv_msg(sock: socket.socket):
    hdr = _recv_exact(sock, _LEN.size)
    if hdr is None:
        return None
    (n,) = _LEN.unpack(hdr)
    data = _recv_exact(sock, n)
    if data is None:
        return None
    return pickle.loads(data)


def _recv_exact(sock: socket.socket, n: int) -> Optional[bytes]:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            return None
        buf += chunk
    return buf


class MasterRelay:
    """Runs on the frontend node: accepts one connection per slave node,
    broadcasts the request stream, funnels outputs back."""

    def __init__(self, bind_host: str, relay_port: int,
                 remote_ranks: List[int], out_queue):
        self.remote_ranks = set(remote_ranks)
        self.out_queue = out_queue
        self._conns: List[socket.socket] = []
        self._covered: set = set()
        self._ready_conns = 0
        self._lock = threading.Lock()
        self._closed = False
        self._srv = socket.create_server((bind_host, relay_port),
                                         reuse_port=False)
        self._srv.settimeout(1.0)

    # ---------------------------------------------------------- accept
    def wait_for_slaves(self, timeout_s: float = 600.0) -> None:
        """Block until every remote rank is claimed by a connected slave
        and all slaves reported their workers ready."""
        deadline = time.time() + timeout_s
        while self._covered != self.remote_ranks:
            if time.time() > deadline:
                raise TimeoutError(
                    f"slaves for ranks {sorted(self.remote_ranks - self._covered)} "
                    "did not connect")
            try:
                conn, addr = self._srv.accept()
            except socket.timeout:
                continue
            hello = recv_msg(conn)
            if not isinstance(hello, dict) or hello.get("type") != "hello":
                logger.warning("relay: bad hello from %s", addr)
                conn.close()
                continue
            ranks = hello["ranks"]
            logger.info("relay: slave %s hosts ranks %s", addr, ranks)
            self._covered.update(ranks)
            with self._lock:
                self._conns.append(conn)
            threading.Thread(target=self._reader, args=(conn,),
                             daemon=True).start()
        while self._ready_conns < len(self._conns):
            if time.time() > deadline:
                raise TimeoutError("slave workers did not become ready")
            time.sleep(0.1)

    def _reader(self, conn: socket.socket) -> None:
        while not self._closed:
            try:
                msg = recv_msg(conn)
            except OSError:
                msg = None
            if msg is None:
                if not self._closed:
                    logger.error("relay: slave connection lost")
                    self.out_queue.put(("worker_dead", None, {}))
                return
            if isinstance(msg, dict):
                if msg.get("type") == "ready":
                    self._ready_conns += 1
                elif msg.get("type") == "dead":
                    self.out_queue.put(("worker_dead", msg.get("rank"), {}))
                continue
            self.out_queue.put(msg)  # ("out"/"stats", payload, extra)

    # ------------------------------------------------------- broadcast
    def broadcast(self, msg) -> None:
        with self._lock:
            for conn in list(self._conns):
                try:
                    send_msg(conn, msg)
                except OSError:
                    logger.error("relay: send to slave failed")
                    self._conns.remove(conn)

    def close(self) -> None:
        self._closed = True
        with self._lock:
            for c in self._conns:
                try:
                    c.close()
                except OSError:
                    pass
            self._conns.clear()
        self._srv.close()


# ---------------------------------------------------------------- slave
def _connect_with_retry(addr: str, port: int,
                        timeout_s: float = 600.0) -> socket.socket:
    deadline = time.time() + timeout_s
    while True:
        try:
            return socket.create_connection((addr, port), timeout=10)
        except OSError:
            if time.time() > deadline:
                raise
            time.sleep(1.0)


def run_slave_node(config) -> None:
    """Entry point for ``--launch-mode slave``: spawn this node's worker
    ranks, bridge their control plane to the master's relay, block until
    shutdown."""
    import torch.multiprocessing as mp

    from gllm_amd.engine.worker import run_worker

    ranks = config.worker_ranks
    assert ranks, "slave launch requires worker_ranks"
    relay_port = config.relay_port or (config.master_port + 1)
    sock = _connect_with_retry(config.master_addr, relay_port)
    send_msg(sock, {"type": "hello", "ranks": list(ranks)})

    ctx = mp.get_context("spawn")
    req_queues: Dict[int, object] = {r: ctx.Queue() for r in ranks}
    out_queue = ctx.Queue()
    ready_q = ctx.Queue()
    procs = []
    for r in ranks:
        p = ctx.Process(target=run_worker,
                        args=(r, config, req_queues[r], out_queue, ready_q),
                        daemon=True)
        p.start()
        procs.append(p)
    for _ in ranks:
        status, rank = ready_q.get()
        if status != "ready":
            send_msg(sock, {"type": "dead", "rank": rank})
            raise RuntimeError(f"slave worker {rank} failed to start")
    send_msg(sock, {"type": "ready"})
    logger.info("slave node ready: ranks %s", list(ranks))

    stop = threading.Event()

    def pump_outputs():
        while not stop.is_set():
            try:
                msg = out_queue.get(timeout=0.2)
            except Exception:
                continue
            try:
                send_msg(sock, msg)
            except OSError:
                return

    t = threading.Thread(target=pump_outputs, daemon=True)
    t.start()

    # inbound request stream -> every local worker queue, in order
    while True:
        try:
            msg = recv_msg(sock)
        except OSError:
            msg = None
        if msg is None:
            logger.warning("slave: master connection closed; shutting down")
            msg = ("cmd", -1, "shutdown")
        for q in req_queues.values():
            q.put(msg)
        if msg[0] == "cmd" and msg[2] == "shutdown":
            break
    for p in procs:
        p.join(timeout=60)
    stop.set()
    t.join(timeout=5)
    try:
        sock.close()
    except OSError:
        pass
    logger.info("slave node shut down")
