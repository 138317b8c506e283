"""Per-request state: tokens, KV page table, chunked-prefill cursors.

Capability parity with the reference Sequence (gllm/sequence.py): token
accumulation, chunked-prefill cursors (computed_token_num /
to_compute_token_num), preempt/resume, incremental detokenization,
per-page prefix-hash cache. Re-designed as a plain dataclass-style object
with explicit SamplingParams.
"""

import dataclasses
from typing import Dict, List, Optional


@dataclasses.dataclass
class SamplingParams:
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1                    # -1 = disabled
    min_p: float = 0.0                 # drop p < min_p * max_p (vLLM)
    repetition_penalty: float = 1.0
    presence_penalty: float = 0.0      # additive, over output tokens
    frequency_penalty: float = 0.0     # additive * count, output tokens
    max_tokens: int = 512
    min_tokens: int = 0
    ignore_eos: bool = False
    stop: Optional[List[str]] = None
    stop_token_ids: Optional[List[int]] = None
    include_stop_str_in_output: bool = False
    logprobs: Optional[int] = None     # top-k logprobs to return per token
    prompt_logprobs: Optional[int] = None
    seed: Optional[int] = None
    # OpenAI logit_bias: token id -> additive bias (applied every step)
    logit_bias: Optional[Dict[int, float]] = None
    # restrict sampling to this token set (OpenAI allowed_token_ids)
    allowed_token_ids: Optional[List[int]] = None
    # banned token SEQUENCES (pre-tokenized bad_words): the last token
    # of a sequence is masked whenever the preceding tokens match the
    # tail of the generated context
    bad_words_token_ids: Optional[List[List[int]]] = None
    skip_special_tokens: bool = True   # detokenization flag

    @property
    def is_greedy(self) -> bool:
        return self.temperature == 0.0


class Sequence:
    FINISH_LENGTH = "length"
    FINISH_STOP = "stop"
    FINISH_ABORT = "abort"

    def __init__(self, seq_id: int, prompt_token_ids: List[int],
                 sampling: Optional[SamplingParams] = None,
                 eos_token_id: Optional[int] = None,
                 arrival_time: float = 0.0):
        self.seq_id = seq_id
        self.token_ids: List[int] = list(prompt_token_ids)
        self.prompt_len = len(prompt_token_ids)
        # raw_prompt_len differs from prompt_len once multimodal sentinel
        # expansion exists; identical for text.
        self.raw_prompt_len = self.prompt_len
        self.sampling = sampling or SamplingParams()
        self.eos_token_id = eos_token_id
        self.arrival_time = arrival_time

        # --- paged KV state ---
        self.page_table: List[int] = []
        # tokens whose KV is already in cache (advances by chunk)
        self.computed_token_num: int = 0
        # tokens scheduled to be computed this tick (set by the scheduler)
        self.to_compute_token_num: int = 0
        # pages at the head that came from the prefix cache
        self.num_cached_pages: int = 0
        # per-page chained hashes (prefix cache keys), computed lazily
        self.page_hashes: List[int] = []
        # multimodal: prompt-length token list with image-pad runs
        # replaced by content-hash pseudo-ids — the prefix cache keys
        # (core/kv_cache.py _key_ids); None = text-only
        self.cache_key_ids: Optional[List[int]] = None

        # --- output state ---
        self.finish_reason: Optional[str] = None
        # incremental detokenize cursor (index into token_ids)
        self.detok_offset: int = self.prompt_len
        self.detok_text: str = ""
        # first-token timestamp for TTFT accounting
        self.first_token_time: Optional[float] = None

        # repetition-penalty mask slot / SSM slot (set by managers when used)
        self.penalty_slot: int = -1
        self.ssm_slot: int = -1
        # slot state matches computed_token_num (False on a fresh slot
        # whose prefix-hit state still needs a snapshot restore)
        self.ssm_state_ready: bool = False
        # per-output-token logprobs (filled when sampling.logprobs is set):
        # list of (chosen_logprob, {token_id: logprob} top-k)
        self.out_logprobs: List[tuple] = []
        # prompt logprobs (sampling.prompt_logprobs): one entry per
        # prompt position >= 1, (logprob_of_actual_token, {tok: lp})
        self.prompt_logprobs_out: List[tuple] = []

        # --- multimodal (set by the engine at admission) ---
        # concatenated vision embeddings for all image spans [N, hidden]
        self.mm_embeds = None
        # (token_start, length) spans of image placeholder runs
        self.mm_spans: List[tuple] = []
        # MRoPE position table [3, prompt_len] + decode delta
        self.mrope_positions = None
        self.mrope_delta: int = 0

    # ---- basic accounting ----
    def __len__(self) -> int:
        return len(self.token_ids)

    @property
    def output_len(self) -> int:
        return self.sampling.max_tokens

    @property
    def num_output_tokens(self) -> int:
        return len(self.token_ids) - self.prompt_len

    @property
    def computed_prompt(self) -> bool:
        """True once the whole prompt's KV is computed (decode phase)."""
        return self.computed_token_num >= self.prompt_len

    @property
    def last_token(self) -> int:
        return self.token_ids[-1]

    @property
    def is_finished(self) -> bool:
        return self.finish_reason is not None

    # ---- decoding-loop transitions ----
    def append_token(self, token_id: int) -> None:
        self.token_ids.append(token_id)

    def advance_computed(self) -> None:
        """Commit this tick's chunk after the forward pass."""
        self.computed_token_num += self.to_compute_token_num
        self.to_compute_token_num = 0

    def check_finish(self, pos: Optional[int] = None) -> None:
        """Evaluate finish conditions for the token at ``pos`` (default:
        the last token). Overlap mode passes an explicit pos because a
        later placeholder may already be appended beyond it."""
        if self.finish_reason is not None:
            return
        if pos is None:
            pos = len(self.token_ids) - 1
        n_out = pos - self.prompt_len + 1
        if n_out >= self.sampling.max_tokens:
            self.finish_reason = self.FINISH_LENGTH
            return
        if n_out < max(1, self.sampling.min_tokens):
            return
        last = self.token_ids[pos]
        if (not self.sampling.ignore_eos and self.eos_token_id is not None
                and last == self.eos_token_id):
            self.finish_reason = self.FINISH_STOP
            return
        if self.sampling.stop_token_ids and last in self.sampling.stop_token_ids:
            self.finish_reason = self.FINISH_STOP

    def preempt(self) -> None:
        """Drop all computed KV; the seq will re-prefill from scratch
        (reference: sequence.py:156-169 — recompute, not swap)."""
        self.computed_token_num = 0
        self.to_compute_token_num = 0
        self.num_cached_pages = 0
        self.page_table = []
        self.page_hashes = []

    # ---- output tokens view ----
    @property
    def output_token_ids(self) -> List[int]:
        return self.token_ids[self.prompt_len:]

    def __repr__(self):
        return (f"Sequence(id={self.seq_id}, len={len(self.token_ids)}, "
                f"computed={self.computed_token_num}, "
                f"pages={len(self.page_table)}, finish={self.finish_reason})")
