"""Vision-encoder server: a standalone process hosting ONLY the ViT.

Parity target: reference disagg/encoder_runtime.py + encoder_engine.py
(vision-only process, content-hash dedup cache) and
entrypoints/encoder_server.py. The LM frontend sends EncoderJobs; this
server runs the tower and returns merged embeddings. A hash-first
probe protocol avoids resending pixels for cached content:

  client:  EncoderJob(pixel_values=None)   # probe by content hash
  server:  EncoderResult(embeds=..., cached=True)   on hit
           EncoderResult(embeds=None)               on miss
  client:  EncoderJob(pixel_values=...)    # full job
  server:  EncoderResult(embeds=...)

Round-2 data plane: hipIpc slot-pool WRITEs over xGMI (ROADMAP.md).
"""

import collections
import socket
import threading
from typing import Optional

import torch

from gllm_amd.disagg.protocol import EncoderJob, EncoderResult
from gllm_amd.engine.multinode import recv_msg, send_msg
from gllm_amd.logger import logger


class EmbeddingCache:
    """Content-hash LRU (reference model_runner.py:161-221)."""

    def __init__(self, max_items: int = 256):
        self.max_items = max_items
        self._d: "collections.OrderedDict[str, torch.Tensor]" = \
            collections.OrderedDict()
        self.hits = 0
        self.misses = 0

    def get(self, key: str) -> Optional[torch.Tensor]:
        if key in self._d:
            self._d.move_to_end(key)
            self.hits += 1
            return self._d[key]
        self.misses += 1
        return None

    def put(self, key: str, value: torch.Tensor) -> None:
        self._d[key] = value
        self._d.move_to_end(key)
        while len(self._d) > self.max_items:
            self._d.popitem(last=False)


class EncoderServer:
    def __init__(self, config, host: str = "0.0.0.0", port: int = 29820,
                 cache_items: int = 256):
        import os

        from gllm_amd.models.loader import load_model
        from gllm_amd.parallel import init_distributed
        init_distributed(pp_size=1, dp_size=1, tp_size=1)
        # full-model construction keeps dummy-init weight names (and so
        # the crc32 weights) identical to the LM side; only .visual runs
        self.model, _ = load_model(config, config.device)
        assert getattr(self.model, "visual", None) is not None, \
            "model has no vision tower"
        self.cache = EmbeddingCache(cache_items)
        self.port = port
        # fault injection (reference GLLM_ENC_FAIL_FIRST_N,
        # disagg/config.py:13-18): fail the first N real encode jobs so
        # the client's redispatch path can be exercised
        self._fail_first_n = int(os.environ.get("GLLM_ENC_FAIL_FIRST_N",
                                                "0"))
        self._jobs_seen = 0
        self._srv = socket.create_server((host, port))
        self._srv.settimeout(0.5)
        self._stop = threading.Event()

    def serve_forever(self) -> None:
        logger.info("encoder server listening on %d", self.port)
        while not self._stop.is_set():
            try:
                conn, _ = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    def stop(self) -> None:
        self._stop.set()
        self._srv.close()

    def _serve_conn(self, conn: socket.socket) -> None:
        pool = None  # per-connection GPU-direct mapping
        try:
            while not self._stop.is_set():
                job = recv_msg(conn)
                if job is None:
                    return
                if job == "stats":
                    send_msg(conn, {"hits": self.cache.hits,
                                    "misses": self.cache.misses})
                    continue
                if job == "shutdown":
                    send_msg(conn, {"ok": True})
                    self.stop()
                    return
                from gllm_amd.disagg.protocol import PoolRegistration
                if isinstance(job, PoolRegistration):
                    try:
                        from gllm_amd.disagg.gpu_plane import RemotePool
                        assert torch.cuda.is_available()
                        pool = RemotePool(job)
                        send_msg(conn, {"pool": True})
                        logger.info("GPU-direct plane mapped "
                                    "(%d slots x %d elems)",
                                    job.n_slots, job.slot_elems)
                    except Exception as e:
                        logger.warning("pool mapping failed (%s); "
                                       "TCP payload fallback", e)
                        send_msg(conn, {"pool": False})
                    continue
                send_msg(conn, self._run(job, pool))
        except OSError:
            pass
        finally:
            if pool is not None:
                pool.close()
            conn.close()

    def _run(self, job: EncoderJob, pool=None) -> EncoderResult:
        emb = self.cache.get(job.content_hash)
        if emb is not None:
            if pool is not None and job.slot is not None:
                t, d = emb.shape
                pool.write(job.slot, emb.cuda())
                return EncoderResult(job.job_id, None, cached=True,
                                     via_pool=True, n_tokens=t,
                                     embed_dim=d)
            return EncoderResult(job.job_id, emb, cached=True)
        if job.pixel_values is None:
            return EncoderResult(job.job_id, None)  # probe miss
        self._jobs_seen += 1
        if self._jobs_seen <= self._fail_first_n:
            return EncoderResult(job.job_id, None,
                                 error="injected failure "
                                       f"({self._jobs_seen}/"
                                       f"{self._fail_first_n})")
        try:
            with torch.no_grad():
                emb = self.model.encode_images(job.pixel_values,
                                               job.grids).cpu()
        except Exception as e:  # pragma: no cover
            logger.exception("encoder job %d failed", job.job_id)
            return EncoderResult(job.job_id, None, error=str(e))
        self.cache.put(job.content_hash, emb)
        if pool is not None and job.slot is not None:
            # GPU-direct: WRITE into the client's slot; TCP carries only
            # the readiness notification
            t, d = emb.shape
            pool.write(job.slot, emb.cuda())
            return EncoderResult(job.job_id, None, via_pool=True,
                                 n_tokens=t, embed_dim=d)
        return EncoderResult(job.job_id, emb)


class EncoderClient:
    """LM-frontend side: probe-by-hash then send pixels on miss.

    Failure handling (reference DisaggCoordinator watchdog +
    GLLM_DISAGG_MAX_REDISPATCH): a failed or dropped job is re-sent up
    to ``max_redispatch`` times (reconnecting on a dead socket) before
    the request is surfaced as an error — the API layer turns that into
    a 400/abort, never a hung sequence."""

    def __init__(self, addr: str, max_redispatch: int = None):
        import os
        self._addr = addr
        self._sock = None
        self._pool = None
        self._use_pool = os.environ.get("GLLM_DISAGG_GPU_DIRECT",
                                        "1") != "0"
        self._connect()
        self._lock = threading.Lock()
        self._job_id = 0
        self.max_redispatch = max_redispatch if max_redispatch is not None \
            else int(os.environ.get("GLLM_DISAGG_MAX_REDISPATCH", "2"))

    def _connect(self):
        host, port = self._addr.rsplit(":", 1)
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
        self._sock = socket.create_connection((host, int(port)),
                                              timeout=60)
        self._register_pool()

    def _register_pool(self):
        """One-time hipIpc handshake per connection; downgrades to the
        TCP payload path when the encoder can't map us (cross-node)."""
        if not self._use_pool:
            return
        from gllm_amd.disagg.gpu_plane import try_make_pool
        from gllm_amd.disagg.protocol import PoolRegistration
        if self._pool is None:
            self._pool = try_make_pool()
        if self._pool is None:
            self._use_pool = False
            return
        send_msg(self._sock, PoolRegistration(
            bytes(self._pool.handle), self._pool.n_slots,
            self._pool.slot_elems))
        ack = recv_msg(self._sock)
        if not (isinstance(ack, dict) and ack.get("pool")):
            logger.info("encoder declined GPU-direct plane; TCP payload")
            self._use_pool = False

    def _attempt(self, key, grids, pixel_values):
        self._job_id += 1
        jid = self._job_id
        slot = self._pool.acquire() if self._use_pool else None
        send_msg(self._sock, EncoderJob(jid, key, list(grids), slot=slot))
        res: EncoderResult = recv_msg(self._sock)
        if res is None:
            raise ConnectionError("encoder connection closed")
        if res.embeds is None and res.error is None and not res.via_pool:
            send_msg(self._sock,
                     EncoderJob(jid, key, list(grids), pixel_values,
                                slot=slot))
            res = recv_msg(self._sock)
            if res is None:
                raise ConnectionError("encoder connection closed")
        if res.error:
            raise RuntimeError(f"encoder job failed: {res.error}")
        if res.via_pool:
            # the notification gates the xGMI write; clone out of the
            # slot so it can be reused
            return self._pool.view(slot, res.n_tokens,
                                   res.embed_dim).clone()
        return res.embeds

    def encode(self, pixel_values: torch.Tensor, grids) -> torch.Tensor:
        from gllm_amd.disagg.protocol import content_hash
        key = content_hash(pixel_values, grids)
        last = None
        with self._lock:
            for attempt in range(self.max_redispatch + 1):
                try:
                    return self._attempt(key, grids, pixel_values)
                except ConnectionError as e:
                    last = e
                    logger.warning("encoder redispatch %d/%d: %s",
                                   attempt + 1, self.max_redispatch, e)
                    self._connect()
                except RuntimeError as e:
                    last = e
                    logger.warning("encoder redispatch %d/%d: %s",
                                   attempt + 1, self.max_redispatch, e)
        raise RuntimeError(
            f"encoder job failed after {self.max_redispatch + 1} "
            f"attempts: {last}")

    def stats(self) -> dict:
        with self._lock:
            send_msg(self._sock, "stats")
            return recv_msg(self._sock)

    def close(self) -> None:
        self._sock.close()


def run_encoder_server(config, host: str = "0.0.0.0", port: int = 29820,
                       discovery_addr: Optional[str] = None) -> None:
    """Entry point for ``python -m gllm_amd.entrypoints.encoder_server``."""
    srv = EncoderServer(config, host, port)
    if discovery_addr:
        from gllm_amd.disagg.discovery import (DiscoveryClient,
                                               start_heartbeat)
        dc = DiscoveryClient(discovery_addr)
        my_addr = f"{socket.gethostname()}:{port}"
        dc.register("encoder", config.model, my_addr)
        start_heartbeat(dc, "encoder", config.model, my_addr)
    srv.serve_forever()
