"""Offline single-process engine: the `LLM.generate()` surface
(reference llm_engine.py:610-697) for world_size == 1.

The multi-process serving engine (frontend + one worker per GPU over
zmq) lives in engine/worker.py + engine/server_engine.py; this class is
the in-process path used by tests, examples and the N=1 bench.
"""

import time
from typing import List, Optional, Sequence as Seq, Union

import torch

from gllm_amd.config import EngineConfig
from gllm_amd.core.kv_cache import MemoryManager
from gllm_amd.core.scheduler import Scheduler
from gllm_amd.logger import logger
from gllm_amd.runtime.model_runner import ModelRunner
from gllm_amd.sequence import SamplingParams, Sequence
from gllm_amd.utils.id_allocator import IDAllocator


class RequestOutput:
    def __init__(self, seq: Sequence, text: str = ""):
        self.seq_id = seq.seq_id
        self.prompt_token_ids = seq.token_ids[:seq.prompt_len]
        self.token_ids = seq.output_token_ids
        self.finish_reason = seq.finish_reason
        self.text = text
        self.logprobs = seq.out_logprobs or None
        self.prompt_logprobs = seq.prompt_logprobs_out or None


class LLM:
    def __init__(self, model: str = "", config: Optional[EngineConfig] = None,
                 num_pages_override: Optional[int] = None, **kwargs):
        if config is None:
            config = EngineConfig(model=model, **kwargs)
        self.config = config
        from gllm_amd.parallel import init_distributed
        init_distributed(config)
        self.runner = ModelRunner(config).init(
            num_pages_override=num_pages_override)
        self.scheduler = Scheduler(config, self.runner.memory_manager)
        self.seq_id_alloc = IDAllocator(1 << 20)
        self.tokenizer = self._load_tokenizer(config.model)
        self.eos_token_id = None
        if self.tokenizer is not None:
            self.eos_token_id = self.tokenizer.eos_token_id
        elif getattr(self.runner.hf_config, "eos_token_id", None) is not None:
            eos = self.runner.hf_config.eos_token_id
            self.eos_token_id = eos[0] if isinstance(eos, list) else eos

    @staticmethod
    def _load_tokenizer(model_path: str):
        from gllm_amd.utils.tokenizer import load_tokenizer
        try:
            return load_tokenizer(model_path)
        except Exception as e:  # pragma: no cover
            logger.warning("tokenizer load failed: %s", e)
            return None

    # ------------------------------------------------------------------
    def allocate_seq(self, prompt_token_ids: List[int],
                     sampling: Optional[SamplingParams] = None,
                     mm_input: Optional[dict] = None) -> Sequence:
        seq = Sequence(self.seq_id_alloc.allocate(), prompt_token_ids,
                       sampling=sampling, eos_token_id=self.eos_token_id,
                       arrival_time=time.time())
        if mm_input is not None:
            self._prepare_mm(seq, mm_input)
        return seq

    def _prepare_mm(self, seq: Sequence, mm_input: dict) -> None:
        from gllm_amd.multimodal.prepare import prepare_mm_seq
        prepare_mm_seq(self.runner.model, seq, mm_input)

    def add_requests(self, seqs: List[Sequence]) -> None:
        self.scheduler.add_seqs(seqs)

    def abort(self, seq_ids) -> None:
        self.scheduler.abort_seqs(seq_ids)

    # ------------------------------------------------------------------
    def step(self) -> List[Sequence]:
        """One scheduler tick + forward. Returns seqs finished this tick."""
        batch = self.scheduler.schedule_once()
        if batch is None:
            return []
        out = self.runner.step_first_stage(batch)
        from gllm_amd.engine.pp_engine import PPEngine
        PPEngine._stash_logprobs(batch, out)
        tokens = out.next_tokens.tolist()
        finished = self.scheduler.process_output(batch, tokens)
        for s in finished:
            self.seq_id_alloc.free(s.seq_id)
        return finished

    # ------------------------------------------------------------------
    def generate(self,
                 prompts: Optional[Seq[Union[str, List[int]]]] = None,
                 sampling_params: Optional[Union[SamplingParams,
                                                 List[SamplingParams]]] = None,
                 mm_inputs: Optional[List[Optional[dict]]] = None
                 ) -> List[RequestOutput]:
        assert prompts is not None
        if isinstance(sampling_params, SamplingParams) or \
                sampling_params is None:
            sampling_params = [sampling_params or SamplingParams()
                               ] * len(prompts)
        if mm_inputs is None:
            mm_inputs = [None] * len(prompts)
        seqs = []
        for p, sp, mm in zip(prompts, sampling_params, mm_inputs):
            if isinstance(p, str):
                assert self.tokenizer is not None, \
                    "string prompts need a tokenizer"
                ids = self.tokenizer.encode(p)
            else:
                ids = list(p)
            seqs.append(self.allocate_seq(ids, sp, mm_input=mm))
        self.add_requests(seqs)
        pending = {s.seq_id for s in seqs}
        while pending:
            finished = self.step()
            for s in finished:
                pending.discard(s.seq_id)
            if not finished and not self.scheduler.has_work():
                raise RuntimeError("engine stalled with pending requests")
        outs = []
        for s in seqs:
            text = ""
            if self.tokenizer is not None:
                text = self.tokenizer.decode(s.output_token_ids)
            outs.append(RequestOutput(s, text))
        return outs

    def chat(self, messages, sampling_params: Optional[SamplingParams] = None
             ) -> RequestOutput:
        assert self.tokenizer is not None
        text = self.tokenizer.apply_chat_template(
            messages, add_generation_prompt=True, tokenize=False)
        ids = self.tokenizer.encode(text)
        return self.generate([ids], [sampling_params or SamplingParams()])[0]
