"""Frontend async engine: request intake, worker spawn, output streams.

Parity: reference llm_engine.py (worker spawn, allocate_seq, schedule
plumbing) + async_llm_engine.py (AsyncStream, abort-on-disconnect), on
the replicated-scheduler worker design (engine/worker.py)."""

import asyncio
import dataclasses
import threading
import time
from typing import AsyncIterator, Dict, List, Optional

import torch.multiprocessing as mp

from gllm_amd.config import EngineConfig
from gllm_amd.engine.detokenizer import (IncrementalDetokenizer,
                                         check_stop_strings)
from gllm_amd.engine.ipc import FrontendComm
from gllm_amd.logger import logger
from gllm_amd.models.loader import load_hf_config
from gllm_amd.sequence import SamplingParams
from gllm_amd.utils.id_allocator import IDAllocator


@dataclasses.dataclass
class StreamChunk:
    seq_id: int
    token_id: int
    text: str
    finish_reason: Optional[str]
    n_output_tokens: int
    # attached to the FINAL chunk when sampling.prompt_logprobs was set
    prompt_logprobs: Optional[list] = None
    # (chosen_logprob, {token_id: logprob}) when sampling.logprobs
    logprob: Optional[tuple] = None


class RequestState:
    def __init__(self, seq_id: int, prompt_len: int, sampling: SamplingParams,
                 tokenizer, loop):
        self.seq_id = seq_id
        self.prompt_len = prompt_len
        self.sampling = sampling
        self.detok = IncrementalDetokenizer(
            tokenizer, skip_special_tokens=getattr(
                sampling, "skip_special_tokens", True)) \
            if tokenizer else None
        self.queue: asyncio.Queue = asyncio.Queue()
        self.loop = loop
        self.n_tokens = 0
        self.finished = False
        self.created = time.time()
        self.first_token_time: Optional[float] = None
        self.prompt_logprobs: Optional[list] = None


class AsyncLLMEngine:
    """Owns the frontend side: tokenizer, worker processes, streams."""

    def __init__(self, config: EngineConfig, base_port: int = 0):
        # base_port kept for CLI compat; the queue transport ignores it
        self.config = config
        self.hf_config = load_hf_config(config.model)
        self.tokenizer = self._load_tokenizer(config.model)
        eos = getattr(self.hf_config, "eos_token_id", None)
        if self.tokenizer is not None and self.tokenizer.eos_token_id is not None:
            eos = self.tokenizer.eos_token_id
        self.eos_token_id = eos[0] if isinstance(eos, list) else eos
        local_ranks = config.worker_ranks \
            if config.launch_mode == "master" else None
        self.comm = FrontendComm(config.world_size, local_ranks=local_ranks)
        self._relay = None
        # multimodal serving (Qwen2-VL family): native image processor
        vcfg = getattr(self.hf_config, "vision_config", None)
        self.image_token_id = getattr(self.hf_config, "image_token_id",
                                      None)
        if vcfg is not None and self.image_token_id is not None:
            from gllm_amd.multimodal.processor import ImageProcessor
            self.mm_processor = ImageProcessor.from_config(vcfg)
        else:
            self.mm_processor = None
        # encoder disaggregation: vision runs in a remote encoder
        # process; workers receive ready embeddings (disagg/)
        self.encoder_client = None
        addr = config.mm_encoder_addr
        if addr is None and config.discovery_addr:
            from gllm_amd.disagg.discovery import DiscoveryClient
            dc = DiscoveryClient(config.discovery_addr)
            addrs = dc.lookup("encoder", config.model)
            dc.close()
            if addrs:
                addr = addrs[0]
                logger.info("discovered encoder at %s", addr)
        if addr:
            from gllm_amd.disagg.encoder_server import EncoderClient
            self.encoder_client = EncoderClient(addr)
        # literal strings the chat template/tokenizer map to the image
        # tokens (Qwen2-VL conventions)
        self.image_pad_str = "<|image_pad|>"
        self.vision_wrap = ("<|vision_start|>", "<|vision_end|>")
        self.seq_ids = IDAllocator(1 << 20)
        self.requests: Dict[int, RequestState] = {}
        self._intake_lock = threading.Lock()
        self._procs: List = []
        self._ready = False
        self._output_thread: Optional[threading.Thread] = None
        self._stopping = False
        self.loop: Optional[asyncio.AbstractEventLoop] = None
        self.latest_stats: Dict = {}
        self.request_counter = 0
        self.token_counter = 0

    @staticmethod
    def _load_tokenizer(model_path: str):
        from gllm_amd.utils.tokenizer import load_tokenizer
        return load_tokenizer(model_path)

    # ------------------------------------------------------------------
    def start(self) -> None:
        ctx = mp.get_context("spawn")
        ready_q = ctx.Queue()
        from gllm_amd.engine.worker import run_worker
        cfg = self.config
        local_ranks = self.comm.local_ranks
        if cfg.launch_mode == "master":
            # control-plane relay for slave nodes (data plane = RCCL
            # rendezvous at master_addr:master_port, cross-node already)
            from gllm_amd.engine.multinode import MasterRelay
            remote = sorted(set(range(cfg.world_size)) - set(local_ranks))
            self._relay = MasterRelay("0.0.0.0",
                                      cfg.relay_port or cfg.master_port + 1,
                                      remote, self.comm.out_queue)
            self.comm.remote_sender = self._relay.broadcast
        for r in local_ranks:
            rq, oq = self.comm.worker_endpoints(r)
            p = ctx.Process(target=run_worker,
                            args=(r, self.config, rq, oq, ready_q),
                            daemon=True)
            p.start()
            self._procs.append(p)
        if self._relay is not None:
            # slaves must be connected before workers can finish their
            # torch.distributed init; accept them while workers load
            self._relay.wait_for_slaves()
        for _ in local_ranks:
            status, rank = ready_q.get()  # blocks until model loaded
            if status != "ready":
                raise RuntimeError(f"worker {rank} failed to start")
        self._ready = True
        self._output_thread = threading.Thread(target=self._output_loop,
                                               daemon=True)
        self._output_thread.start()
        self._watchdog = threading.Thread(target=self._watch_workers,
                                          daemon=True)
        self._watchdog.start()
        logger.info("engine ready: %d worker(s)", self.config.world_size)

    def _watch_workers(self) -> None:
        """Worker death is fatal (reference worker.py:988-999 /
        llm_engine.py:349-352): fail every pending stream, then stop."""
        import time as _t
        while not self._stopping:
            for p in self._procs:
                if not p.is_alive() and p.exitcode not in (0, None):
                    logger.error("worker %s died (exit %s) — failing %d "
                                 "pending requests", p.pid, p.exitcode,
                                 len(self.requests))
                    for st in list(self.requests.values()):
                        if not st.finished:
                            self._handle_token(st, -1, "abort")
                    self._stopping = True
                    return
            _t.sleep(0.5)

    def stop(self) -> None:
        self._stopping = True
        try:
            self.comm.send_to_all("cmd", "shutdown")
        except Exception:
            pass
        for p in self._procs:
            p.join(timeout=30)
        if self._relay is not None:
            self._relay.close()

    # ------------------------------------------------------------------
    def _output_loop(self) -> None:
        while not self._stopping:
            msg = self.comm.recv_output(timeout_ms=200)
            if msg is None:
                continue
            kind, outs, _stats = msg
            if kind == "stats":
                self.latest_stats = outs
                continue
            if kind == "plp":
                # prompt logprobs arrive just before the finish token
                for seq_id, plp in outs:
                    st = self.requests.get(seq_id)
                    if st is not None:
                        st.prompt_logprobs = plp
                continue
            if kind == "worker_dead":
                logger.error("remote worker %s died — failing %d pending "
                             "requests", outs, len(self.requests))
                for st in list(self.requests.values()):
                    if not st.finished:
                        self._handle_token(st, -1, "abort")
                self._stopping = True
                return
            if kind != "out":
                continue
            self.token_counter += sum(1 for o in outs if o[1] >= 0)
            for o in outs:
                seq_id, token_id, finish = o[0], o[1], o[2]
                lp = o[3] if len(o) > 3 else None
                st = self.requests.get(seq_id)
                if st is None or st.finished:
                    continue
                self._handle_token(st, token_id, finish, lp)

    def _handle_token(self, st: RequestState, token_id: int,
                      finish: Optional[str], lp=None) -> None:
        if st.first_token_time is None:
            st.first_token_time = time.time()
        text = ""
        if token_id >= 0:
            st.n_tokens += 1
            if st.detok is not None:
                text = st.detok.append(token_id)
                hit, _trunc = check_stop_strings(st.detok.text,
                                                 st.sampling.stop)
                if hit and finish is None:
                    finish = "stop"
                    self.abort([st.seq_id])
        if finish is not None:
            st.finished = True
        chunk = StreamChunk(st.seq_id, token_id, text, finish, st.n_tokens,
                            prompt_logprobs=st.prompt_logprobs
                            if finish is not None else None,
                            logprob=lp)
        st.loop.call_soon_threadsafe(st.queue.put_nowait, chunk)
        if st.finished:
            st.loop.call_soon_threadsafe(st.queue.put_nowait, None)

    # ------------------------------------------------------------------
    def add_request(self, token_ids: List[int],
                    sampling: SamplingParams,
                    mm: Optional[dict] = None) -> RequestState:
        loop = asyncio.get_event_loop()
        with self._intake_lock:
            seq_id = self.seq_ids.allocate()
            st = RequestState(seq_id, len(token_ids), sampling,
                              self.tokenizer, loop)
            self.requests[seq_id] = st
            self.request_counter += 1
            self.comm.send_to_all("req", {
                "seq_id": seq_id,
                "token_ids": token_ids,
                "sampling": dataclasses.asdict(sampling),
                "eos_token_id": self.eos_token_id,
                "mm": mm,
            })
        return st

    def abort(self, seq_ids: List[int]) -> None:
        with self._intake_lock:
            self.comm.send_to_all("abort", list(seq_ids))

    def send_command(self, name: str) -> None:
        with self._intake_lock:
            self.comm.send_to_all("cmd", name)

    def release(self, st: RequestState) -> None:
        self.requests.pop(st.seq_id, None)
        self.seq_ids.free(st.seq_id)

    # ------------------------------------------------------------------
    async def generate_stream(self, token_ids: List[int],
                              sampling: SamplingParams,
                              mm: Optional[dict] = None
                              ) -> AsyncIterator[StreamChunk]:
        st = self.add_request(token_ids, sampling, mm=mm)
        try:
            while True:
                chunk = await st.queue.get()
                if chunk is None:
                    break
                yield chunk
        finally:
            if not st.finished:
                self.abort([st.seq_id])
            self.release(st)

    def encode(self, prompt: str) -> List[int]:
        assert self.tokenizer is not None, "no tokenizer available"
        return self.tokenizer.encode(prompt)

    # ------------------------------------------------- multimodal intake
    def extract_images(self, messages: list):
        """OpenAI multi-part content -> (text-only messages with one
        image-pad sentinel per image, raw image bytes list)."""
        from gllm_amd.multimodal.processor import decode_image_url
        images = []
        out = []
        vs, ve = self.vision_wrap
        for m in messages:
            content = m.get("content")
            if not isinstance(content, list):
                out.append(m)
                continue
            parts = []
            for part in content:
                ptype = part.get("type")
                if ptype == "text":
                    parts.append(part["text"])
                elif ptype == "image_url":
                    url = part["image_url"]
                    if isinstance(url, dict):
                        url = url["url"]
                    images.append(decode_image_url(url))
                    parts.append(f"{vs}{self.image_pad_str}{ve}")
                else:
                    raise ValueError(f"unsupported content part {ptype}")
            out.append({**m, "content": "".join(parts)})
        return out, images

    def process_images(self, token_ids: List[int], images: list):
        """Preprocess images and expand each single image-pad sentinel
        to its per-image token count. Returns (token_ids, mm dict)."""
        if not images:
            return token_ids, None
        assert self.mm_processor is not None, \
            "model has no vision tower; cannot accept images"
        import torch
        from gllm_amd.multimodal.processor import expand_image_tokens
        pixels, grids, counts = [], [], []
        for raw in images:
            p, grid = self.mm_processor(raw)
            pixels.append(p)
            grids.append(grid)
            counts.append(self.mm_processor.num_tokens(grid))
        token_ids = expand_image_tokens(token_ids, self.image_token_id,
                                        counts, self.image_token_id)
        pixel_values = torch.cat(pixels, dim=0)
        if self.encoder_client is not None:
            # disaggregated: the remote encoder runs the tower; workers
            # get ready embeddings and skip their local tower
            embeds = self.encoder_client.encode(pixel_values, grids)
            return token_ids, {"embeds": embeds, "grids": grids}
        return token_ids, {"pixel_values": pixel_values, "grids": grids}

    def apply_chat_template(self, messages, **kwargs) -> List[int]:
        assert self.tokenizer is not None
        # DeepSeek-V3.2 ships its own DSML encoder instead of a Jinja
        # template (tokenizers/deepseek_v32.py)
        from gllm_amd.tokenizers.deepseek_v32 import (
            apply_dsv32_chat_template, load_dsv32_encoder)
        enc = load_dsv32_encoder(self.config.model)
        if enc is not None:
            ids = apply_dsv32_chat_template(
                enc, messages, self.tokenizer,
                tools=kwargs.pop("tools", None), tokenize=True, **kwargs)
            if not ids:
                raise ValueError("empty prompt after tokenization")
            return ids
        # render to text then encode: apply_chat_template's tokenize=True
        # return type varies across transformers versions (list vs dict)
        agp = kwargs.pop("add_generation_prompt", True)
        cfm = kwargs.pop("continue_final_message", False)
        if cfm:
            agp = False  # mutually exclusive (OpenAI/transformers rule)
        text = self.tokenizer.apply_chat_template(
            messages, add_generation_prompt=agp,
            continue_final_message=cfm, tokenize=False, **kwargs)
        ids = self.tokenizer.encode(text)
        if not ids:
            raise ValueError("empty prompt after tokenization")
        return ids
