"""Paged KV cache + prefix cache for MI355X (288 GB HBM3E per GPU).

Capability parity with the reference memory_manager
(gllm/memory_manager.py: Segment/MemoryManager/PrefixMemoryManager) with a
different design:

* Cache layout per layer: K and V each ``[num_pages, page_size, kv_heads,
  head_dim]`` — page-contiguous so the decode kernel streams a page with
  wide coalesced loads and the cache-scatter kernel writes one token row
  per wavefront.
* Prefix identity is EXACT, not hashed: each full page's content is
  interned as a chain id keyed on (parent_chain_id, token tuple), so there
  is no hash-collision canary (reference needed one at
  memory_manager.py:866-935).
* Freed-but-cached pages stay resident (lazy eviction): the free list
  doubles as the eviction candidate pool, FIFO order.
"""

import dataclasses
from typing import Dict, List, Optional, Tuple

from gllm_amd.sequence import Sequence
from gllm_amd.utils.id_allocator import IDAllocator


@dataclasses.dataclass
class KVCacheSpec:
    num_layers: int          # layers on THIS pipeline stage
    num_kv_heads: int        # per TP rank
    head_dim: int            # K head dim
    page_size: int
    dtype_bytes: int = 2     # bf16
    v_head_dim: int = 0      # 0 => same as head_dim (MLA: Dv != Dk)
    # absorbed MLA: the v-cache is a zero-copy view of the latent
    # k-cache's first v_head_dim dims — only k bytes are allocated
    v_shared: bool = False

    def __post_init__(self):
        if not self.v_head_dim:
            self.v_head_dim = self.head_dim

    @property
    def bytes_per_page(self) -> int:
        v = 0 if self.v_shared else self.v_head_dim
        return (self.num_layers * self.page_size * self.num_kv_heads
                * (self.head_dim + v) * self.dtype_bytes)


class MemoryManager:
    """Page allocator + per-sequence page-table maintenance."""

    def __init__(self, num_pages: int, page_size: int):
        self.num_pages = num_pages
        self.page_size = page_size
        self.allocator = IDAllocator(num_pages)
        # auxiliary per-seq resources released with the seq (penalty
        # mask slots, later SSM slots): list of callables(seq)
        self.free_hooks = []

    # ---- stats ----
    def get_num_free_pages(self) -> int:
        return self.allocator.num_free

    def get_memory_util(self) -> float:
        return 100.0 * self.allocator.num_used / max(1, self.num_pages)

    def get_memory_free(self) -> float:
        return self.allocator.num_free / max(1, self.num_pages)

    # ---- prefix-cache interface (no-ops in the base manager) ----
    def lookup_prefix(self, seq: Sequence) -> None:
        """Attach cached prefix pages to ``seq`` (base manager: none)."""
        return None

    def get_cache_hit_rate(self) -> float:
        return 0.0

    # ---- allocation ----
    def pages_needed(self, seq: Sequence) -> int:
        total_tokens = seq.computed_token_num + seq.to_compute_token_num
        pages_total = -(-total_tokens // self.page_size)
        return max(0, pages_total - len(seq.page_table))

    def can_allocate(self, seqs: List[Sequence]) -> bool:
        return sum(self.pages_needed(s) for s in seqs) <= self.allocator.num_free

    def pre_allocate_page(self, seqs: List[Sequence]) -> None:
        for seq in seqs:
            n = self.pages_needed(seq)
            if n:
                seq.page_table.extend(self._allocate_fresh(n))

    def _allocate_fresh(self, n: int) -> List[int]:
        return self.allocator.allocate_many(n)

    def free_seq(self, seq: Sequence) -> None:
        self.allocator.free_many(seq.page_table)
        seq.page_table = []
        for hook in self.free_hooks:
            hook(seq)

    # ---- slot mapping helper ----
    def slots_for(self, seq: Sequence) -> List[int]:
        """Flat cache slot index for each token computed this tick."""
        out = []
        for pos in range(seq.computed_token_num,
                         seq.computed_token_num + seq.to_compute_token_num):
            page = seq.page_table[pos // self.page_size]
            out.append(page * self.page_size + pos % self.page_size)
        return out


class PrefixMemoryManager(MemoryManager):
    """Prefix-cached page allocator.

    Page states:
      * in ``allocator`` free list, not in ``page_chain``  -> blank free page
      * in ``allocator`` free list, in ``page_chain``      -> cached, evictable
      * allocated, refcount >= 1                            -> live (maybe shared)
    """

    def __init__(self, num_pages: int, page_size: int):
        super().__init__(num_pages, page_size)
        # exact interning of page contents: (parent_chain, tokens) -> chain id
        self._intern: Dict[Tuple[int, Tuple[int, ...]], int] = {}
        self._next_chain = 1  # 0 = root (empty prefix)
        # chain id -> resident page holding that content
        self.chain_to_page: Dict[int, int] = {}
        # page id -> chain id it currently holds (or None)
        self.page_chain: List[Optional[int]] = [None] * num_pages
        self.page_ref: List[int] = [0] * num_pages
        # stats
        self.lookup_tokens = 0
        self.hit_tokens = 0
        # hybrid-SSM hooks (runtime/model_runner.py): hit_filter trims a
        # hit to the deepest boundary whose recurrent-state snapshot
        # exists; on_register(seq, chains, n_full) takes snapshots
        self.hit_filter = None
        self.on_register = None

    # ---- interning ----
    def _chain_id(self, parent: int, tokens: Tuple[int, ...]) -> int:
        key = (parent, tokens)
        cid = self._intern.get(key)
        if cid is None:
            cid = self._next_chain
            self._next_chain += 1
            self._intern[key] = cid
        return cid

    def _key_ids(self, seq: Sequence, lo: int, hi: int) -> Tuple[int, ...]:
        """Token ids for cache KEYS. Multimodal prompts substitute each
        image-pad run with content-hash pseudo-ids
        (multimodal/prepare.py builds seq.cache_key_ids) so two prompts
        with identical text but different pixels never alias a page."""
        base = seq.cache_key_ids
        if base is None:
            return tuple(seq.token_ids[lo:hi])
        if hi <= len(base):
            return tuple(base[lo:hi])
        mixed = list(base) + list(seq.token_ids[len(base):hi])
        return tuple(mixed[lo:hi])

    def _chains_up_to(self, seq: Sequence, n_full: int) -> List[int]:
        """Extend seq.page_hashes to cover the first ``n_full`` pages."""
        chains = seq.page_hashes
        parent = chains[-1] if chains else 0
        for i in range(len(chains), n_full):
            toks = self._key_ids(seq, i * self.page_size,
                                 (i + 1) * self.page_size)
            parent = self._chain_id(parent, toks)
            chains.append(parent)
        return chains[:n_full]

    def seq_page_chains(self, seq: Sequence) -> List[int]:
        """Chain ids for every FULL page of the prompt (cached on the seq)."""
        return self._chains_up_to(seq, seq.prompt_len // self.page_size)

    # ---- prefix lookup (called at admission, before first prefill chunk) ----
    def lookup_prefix(self, seq: Sequence) -> None:
        if seq.computed_token_num > 0 or seq.page_table:
            return
        chains = self.seq_page_chains(seq)
        self.lookup_tokens += seq.prompt_len
        hit_pages: List[int] = []
        for cid in chains:
            page = self.chain_to_page.get(cid)
            if page is None:
                break
            hit_pages.append(page)
        # Full-prompt hit rollback: keep at least one token to compute so the
        # forward pass produces a logit row (reference memory_manager.py:992).
        while hit_pages and len(hit_pages) * self.page_size >= seq.prompt_len:
            hit_pages.pop()
        if self.hit_filter is not None and hit_pages:
            keep = self.hit_filter(chains[:len(hit_pages)])
            hit_pages = hit_pages[:keep]
        for page in hit_pages:
            if self.page_ref[page] == 0:
                # resurrect from the evictable pool
                self.allocator.allocate_id(page)
            self.page_ref[page] += 1
        seq.page_table.extend(hit_pages)
        seq.num_cached_pages = len(hit_pages)
        seq.computed_token_num = len(hit_pages) * self.page_size
        self.hit_tokens += seq.computed_token_num

    def get_cache_hit_rate(self) -> float:
        return 100.0 * self.hit_tokens / max(1, self.lookup_tokens)

    # ---- allocation (fresh pages may evict cached content) ----
    def _allocate_fresh(self, n: int) -> List[int]:
        pages = self.allocator.allocate_many(n)
        for p in pages:
            cid = self.page_chain[p]
            if cid is not None:
                # evict the cached content this blank reuse destroys
                if self.chain_to_page.get(cid) == p:
                    del self.chain_to_page[cid]
                self.page_chain[p] = None
            self.page_ref[p] += 1
        return pages

    # ---- registration: publish full computed pages into the cache ----
    def register_computed_pages(self, seq: Sequence) -> None:
        """After a chunk commits, map newly-full PROMPT pages to chains.

        Output-token pages are also registered so multi-turn chat reuses
        generated context.
        """
        # computed_token_num can momentarily exceed len(token_ids) only if a
        # caller registers before appending the sampled token; clamp to the
        # tokens we actually have.
        n_full = min(seq.computed_token_num, len(seq.token_ids)) // self.page_size
        # overlap mode: never intern pages containing unresolved
        # placeholder tokens (negative ids) — reference hit exactly this
        # poisoning bug (memory_manager.py:1055-1079)
        limit = n_full * self.page_size
        for j, t in enumerate(seq.token_ids[:limit]):
            if t < 0:
                n_full = j // self.page_size
                break
        chains = self._chains_up_to(seq, n_full)
        for i in range(n_full):
            cid = chains[i]
            page = seq.page_table[i]
            if self.page_chain[page] != cid:
                self.page_chain[page] = cid
            # last writer wins; identical content either way
            self.chain_to_page[cid] = page
        if self.on_register is not None and n_full:
            self.on_register(seq, chains, n_full)

    def free_seq(self, seq: Sequence) -> None:
        for p in seq.page_table:
            self.page_ref[p] -= 1
            assert self.page_ref[p] >= 0
            if self.page_ref[p] == 0:
                # cached pages go back to the free list but keep content
                self.allocator.free(p)
        seq.page_table = []
        seq.num_cached_pages = 0
        for hook in self.free_hooks:
            hook(seq)
