"""TTL'd service discovery for disaggregated deployments.

Parity target: reference disagg/discovery.py (zmq discovery server +
client, :172/:261): encoder servers register (service, model, address)
with a heartbeat; LM frontends look up a live encoder for their model.
Entries expire after ``ttl_s`` without a heartbeat.
"""

import socket
import threading
import time
from typing import Dict, List, Tuple

from gllm_amd.engine.multinode import recv_msg, send_msg
from gllm_amd.logger import logger


class DiscoveryServer:
    def __init__(self, host: str = "0.0.0.0", port: int = 29800,
                 ttl_s: float = 10.0):
        self.ttl_s = ttl_s
        # (service, model) -> {addr: last_heartbeat}
        self._entries: Dict[Tuple[str, str], Dict[str, float]] = {}
        self._lock = threading.Lock()
        self._srv = socket.create_server((host, port))
        self._srv.settimeout(0.5)
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True)

    def start(self) -> "DiscoveryServer":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        self._thread.join(timeout=5)
        self._srv.close()

    def _loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    def _serve_conn(self, conn: socket.socket) -> None:
        try:
            while not self._stop.is_set():
                msg = recv_msg(conn)
                if msg is None:
                    return
                send_msg(conn, self._handle(msg))
        except OSError:
            pass
        finally:
            conn.close()

    def _handle(self, msg: dict) -> dict:
        op = msg.get("op")
        key = (msg.get("service", ""), msg.get("model", ""))
        now = time.time()
        with self._lock:
            if op in ("register", "heartbeat"):
                self._entries.setdefault(key, {})[msg["addr"]] = now
                return {"ok": True}
            if op == "lookup":
                live = [a for a, t in self._entries.get(key, {}).items()
                        if now - t < self.ttl_s]
                return {"ok": True, "addrs": live}
            if op == "deregister":
                self._entries.get(key, {}).pop(msg.get("addr"), None)
                return {"ok": True}
        return {"ok": False, "error": f"unknown op {op}"}


class DiscoveryClient:
    def __init__(self, addr: str):
        host, port = addr.rsplit(":", 1)
        self._sock = socket.create_connection((host, int(port)), timeout=10)
        self._lock = threading.Lock()

    def _call(self, msg: dict) -> dict:
        with self._lock:
            send_msg(self._sock, msg)
            out = recv_msg(self._sock)
        if out is None:
            raise ConnectionError("discovery server closed")
        return out

    def register(self, service: str, model: str, addr: str) -> None:
        self._call({"op": "register", "service": service, "model": model,
                    "addr": addr})

    def heartbeat(self, service: str, model: str, addr: str) -> None:
        self._call({"op": "heartbeat", "service": service, "model": model,
                    "addr": addr})

    def lookup(self, service: str, model: str) -> List[str]:
        return self._call({"op": "lookup", "service": service,
                           "model": model}).get("addrs", [])

    def deregister(self, service: str, model: str, addr: str) -> None:
        try:
            self._call({"op": "deregister", "service": service,
                        "model": model, "addr": addr})
        except OSError:  # pragma: no cover
            logger.warning("deregister failed (server gone)")

    def close(self) -> None:
        self._sock.close()


def start_heartbeat(client: DiscoveryClient, service: str, model: str,
                    addr: str, period_s: float = 3.0) -> threading.Event:
    """Background heartbeat; returns the stop event."""
    stop = threading.Event()

    def loop():
        while not stop.wait(period_s):
            try:
                client.heartbeat(service, model, addr)
            except OSError:
                return

    threading.Thread(target=loop, daemon=True).start()
    return stop
