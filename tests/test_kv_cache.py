"""Unit tests: IDAllocator, MemoryManager, PrefixMemoryManager."""

import pytest

from gllm_amd.core.kv_cache import MemoryManager, PrefixMemoryManager
from gllm_amd.sequence import SamplingParams, Sequence
from gllm_amd.utils.id_allocator import IDAllocator


def make_seq(sid, n_prompt, page_size=4):
    return Sequence(sid, list(range(10, 10 + n_prompt)),
                    SamplingParams(max_tokens=8))


def test_id_allocator_fifo_and_targeted():
    a = IDAllocator(4)
    assert a.allocate() == 0
    assert a.allocate() == 1
    a.free(0)
    assert a.allocate_id(0) == 0
    a.free(1)
    a.free(0)
    # FIFO: 2,3 first then 1, 0
    assert a.allocate_many(4) == [2, 3, 1, 0]
    with pytest.raises(RuntimeError):
        a.allocate()


def test_memory_manager_alloc_free():
    mm = MemoryManager(num_pages=8, page_size=4)
    seq = make_seq(1, 10)
    seq.to_compute_token_num = 10
    assert mm.pages_needed(seq) == 3
    mm.pre_allocate_page([seq])
    assert len(seq.page_table) == 3
    assert mm.get_num_free_pages() == 5
    slots = mm.slots_for(seq)
    assert len(slots) == 10
    assert slots[0] == seq.page_table[0] * 4
    assert slots[4] == seq.page_table[1] * 4
    mm.free_seq(seq)
    assert mm.get_num_free_pages() == 8


def test_prefix_cache_hit_and_rollback():
    mm = PrefixMemoryManager(num_pages=16, page_size=4)
    s1 = make_seq(1, 12)
    mm.lookup_prefix(s1)
    assert s1.computed_token_num == 0
    s1.to_compute_token_num = 12
    mm.pre_allocate_page([s1])
    s1.computed_token_num = 12
    s1.to_compute_token_num = 0
    s1.append_token(99)
    mm.register_computed_pages(s1)
    # identical prompt: pages 0..2 cached, but full-hit rolls back one page
    s2 = make_seq(2, 12)
    mm.lookup_prefix(s2)
    assert s2.computed_token_num == 8  # 3 full pages -> rollback to 2
    assert s2.page_table[:2] == s1.page_table[:2]
    # shared refcount
    assert mm.page_ref[s1.page_table[0]] == 2


def test_prefix_cache_partial_hit_and_eviction():
    mm = PrefixMemoryManager(num_pages=4, page_size=4)
    s1 = make_seq(1, 8)
    s1.to_compute_token_num = 8
    mm.lookup_prefix(s1)
    mm.pre_allocate_page([s1])
    s1.computed_token_num = 8
    mm.register_computed_pages(s1)
    mm.free_seq(s1)
    assert mm.get_num_free_pages() == 4
    # different tail, same first page
    s2 = Sequence(2, list(range(10, 14)) + [77, 78, 79, 80],
                  SamplingParams(max_tokens=4))
    mm.lookup_prefix(s2)
    assert s2.computed_token_num == 4  # first page hit only
    hit_rate = mm.get_cache_hit_rate()
    assert hit_rate > 0
    # exhaust the pool: cached pages get evicted for fresh allocation
    s2.to_compute_token_num = 4
    mm.pre_allocate_page([s2])
    s3 = make_seq(3, 8)
    s3.to_compute_token_num = 8
    mm.lookup_prefix(s3)
    mm.pre_allocate_page([s3])
    assert mm.get_num_free_pages() == 0


def test_prefix_cache_refcount_sharing_protects_pages():
    mm = PrefixMemoryManager(num_pages=8, page_size=4)
    s1 = make_seq(1, 8)
    s1.to_compute_token_num = 8
    mm.lookup_prefix(s1)
    mm.pre_allocate_page([s1])
    s1.computed_token_num = 8
    mm.register_computed_pages(s1)
    s2 = make_seq(2, 8)
    mm.lookup_prefix(s2)
    assert s2.computed_token_num == 4
    shared = s2.page_table[0]
    mm.free_seq(s1)
    # page still referenced by s2 -> not in free list
    assert not mm.allocator.is_free(shared)
    mm.free_seq(s2)
    assert mm.allocator.is_free(shared)
