"""Scheduler policy unit tests (pure Python, no GPU)."""

import pytest

from gllm_amd.config import EngineConfig
from gllm_amd.core.kv_cache import MemoryManager, PrefixMemoryManager
from gllm_amd.core.scheduler import Scheduler
from gllm_amd.sequence import SamplingParams, Sequence


def make_cfg(**kw):
    base = dict(model="x", page_size=4, maxp=32, maxd=16, minp=8, iterp=4,
                schedule_method="chunked_prefill", device="cpu",
                enable_prefix_caching=False)
    base.update(kw)
    return EngineConfig(**base)


def make_sched(cfg, num_pages=64):
    mm = MemoryManager(num_pages, cfg.page_size)
    return Scheduler(cfg, mm), mm


def add_seq(sched, sid, prompt_len, max_tokens=4):
    s = Sequence(sid, list(range(prompt_len)),
                 SamplingParams(max_tokens=max_tokens, temperature=0.0),
                 eos_token_id=None)
    sched.add_seqs([s])
    return s


def drive(sched, batch, tok=5):
    return sched.process_output(batch, [tok] * len(batch.items))


def test_prefill_then_decode_roundtrip():
    sched, mm = make_sched(make_cfg())
    s = add_seq(sched, 1, 10, max_tokens=3)
    b1 = sched.schedule_once()
    assert b1 is not None and len(b1.items) == 1
    it = b1.items[0]
    assert it.start == 0 and it.num_tokens == 10
    drive(sched, b1)
    assert s.computed_token_num == 10
    assert len(s.token_ids) == 11
    # decode steps
    for step in range(2):
        b = sched.schedule_once()
        assert len(b.items) == 1 and b.items[0].num_tokens == 1
        finished = drive(sched, b)
        if step == 1:
            assert finished and finished[0] is s
    assert s.finish_reason == "length"
    assert mm.get_num_free_pages() == 64


def test_chunked_prefill_splits_long_prompt():
    sched, _ = make_sched(make_cfg(maxp=8))
    s = add_seq(sched, 1, 20)
    starts = []
    while True:
        b = sched.schedule_once()
        if b is None:
            break
        starts.append((b.items[0].start, b.items[0].num_tokens))
        drive(sched, b)
        if s.computed_prompt:
            break
    assert starts == [(0, 8), (8, 8), (16, 4)]


def test_pp_inflight_chunks_same_seq():
    """With pp_size=2 a long prompt should occupy 2 micro-batches at once."""
    sched, _ = make_sched(make_cfg(maxp=8, pp_size=2))
    s = add_seq(sched, 1, 32)
    b1 = sched.schedule_once()
    b2 = sched.schedule_once()
    assert b1.items[0].start == 0 and b1.items[0].num_tokens == 8
    assert b2.items[0].start == 8 and b2.items[0].num_tokens == 8
    b3 = sched.schedule_once()
    assert b3 is None  # pipeline full
    drive(sched, b1)
    b3 = sched.schedule_once()
    assert b3.items[0].start == 16


def test_decode_budget_balancing_rotates():
    cfg = make_cfg(pp_size=4)
    sched, _ = make_sched(cfg)
    budgets = [sched.get_balanced_decode_token_budget(10) for _ in range(4)]
    assert sum(budgets) == 10
    assert all(b in (2, 3) for b in budgets)


def test_preemption_frees_pages_and_requeues():
    cfg = make_cfg(maxp=64, maxd=16)
    sched, mm = make_sched(cfg, num_pages=9)
    s1 = add_seq(sched, 1, 16, max_tokens=16)   # 4 pages
    s2 = add_seq(sched, 2, 16, max_tokens=16)   # 4 pages
    b = sched.schedule_once()
    drive(sched, b)
    while True:
        b = sched.schedule_once()
        if b is None:
            break
        drive(sched, b)
        if sched.num_preempt_seqs:
            break
        if all(x.is_finished for x in (s1, s2)):
            break
    assert sched.num_preempt_seqs >= 1
    # preempted seq went back to prefill queue with no pages
    preempted = [s for s in sched.seqs_to_prefill]
    assert len(preempted) >= 1
    assert preempted[0].computed_token_num == 0
    assert preempted[0].page_table == []


def test_token_throttling_budget_ramp():
    cfg = make_cfg(schedule_method="token_throttling", maxp=32, minp=4,
                   iterp=4, pp_size=2)  # world>1 engages WT
    sched, _ = make_sched(cfg, num_pages=256)
    for i in range(4):
        add_seq(sched, i, 40)
    b = sched.schedule_once()
    # WT budget = max(wait_tokens//iterp, minp) = max(160//4, 4) = 32 (capped by maxp*ratio)
    n_prefill = sum(it.num_tokens for it in b.items)
    assert 0 < n_prefill <= 32


def test_abort_queued_seq():
    sched, mm = make_sched(make_cfg())
    s = add_seq(sched, 1, 8)
    sched.abort_seqs([s.seq_id])
    assert sched.schedule_once() is None
    assert s.finish_reason == "abort"
    assert mm.get_num_free_pages() == 64


def test_determinism_two_replicas():
    """Two scheduler replicas fed identically must emit identical batches."""
    cfg = make_cfg(maxp=8, pp_size=2)
    a, _ = make_sched(cfg)
    b, _ = make_sched(cfg)
    for sched in (a, b):
        for i in range(3):
            add_seq(sched, i, 11 + 3 * i, max_tokens=4)
    for _ in range(20):
        ba, bb = a.schedule_once(), b.schedule_once()
        if ba is None:
            assert bb is None
            if not a.has_work():
                break
            continue
        assert [(it.seq.seq_id, it.start, it.num_tokens) for it in ba.items] \
            == [(it.seq.seq_id, it.start, it.num_tokens) for it in bb.items]
        a.process_output(ba, list(range(len(ba.items))))
        b.process_output(bb, list(range(len(bb.items))))


def test_pp_layer_range_and_assigned_layers():
    from gllm_amd.config import EngineConfig
    cfg = EngineConfig(model="x", pp_size=4)
    # even split with remainder on the LAST stages (token throttling
    # wants earlier stages lighter: they also run embed/sampling legs)
    assert [cfg.pp_layer_range(r, 10) for r in range(4)] == \
        [(0, 2), (2, 4), (4, 7), (7, 10)]
    cfg2 = EngineConfig(model="x", pp_size=3, assigned_layers="5,3,2")
    assert [cfg2.pp_layer_range(r, 10) for r in range(3)] == \
        [(0, 5), (5, 8), (8, 10)]
    import pytest as _pt
    cfg3 = EngineConfig(model="x", pp_size=2, assigned_layers="5,3")
    with _pt.raises(AssertionError):
        cfg3.pp_layer_range(0, 10)
