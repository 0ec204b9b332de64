"""Property-based paged-KV-cache invariants: any interleaving of
alloc/ensure/free keeps the page accounting exact — no page leaked, none
double-owned, capacity errors raised exactly at the documented bounds."""
import hypothesis.strategies as st
from hypothesis import given, settings

from ollamamq_amd.engine.kvcache import PagedKVCache


@settings(derandomize=True, max_examples=60, deadline=None)
@given(st.lists(st.tuples(st.sampled_from(["alloc", "grow", "free"]),
                          st.integers(0, 7), st.integers(1, 140)),
                min_size=1, max_size=60))
def test_page_accounting_invariants(ops):
    kv = PagedKVCache(1, 2, 128, page_size=16, n_pages=24, max_slots=4,
                      max_ctx=128)
    live = {}
    for op, sel, amount in ops:
        if op == "alloc":
            try:
                s = kv.alloc_slot()
                live[s] = 0
            except RuntimeError:
                assert len(live) == kv.max_slots
        elif op == "grow" and live:
            s = sorted(live)[sel % len(live)]
            want = min(live[s] + amount, 200)
            try:
                kv.ensure(s, want)
                assert want <= kv.max_ctx
                live[s] = want
            except RuntimeError:
                assert (want > kv.max_ctx
                        or (want + 15) // 16 - (live[s] + 15) // 16
                        > kv.free_page_count())
        elif op == "free" and live:
            s = sorted(live)[sel % len(live)]
            kv.free_slot(s)
            del live[s]
        # global invariants after every op
        owned = [p for s2 in range(kv.max_slots) for p in kv._slot_pages[s2]]
        assert len(owned) == len(set(owned)), "page double-owned"
        assert len(owned) + kv.free_page_count() == kv.n_pages, "page leak"
        for s2, n in live.items():
            assert len(kv._slot_pages[s2]) >= (n + 15) // 16

    for s in list(live):
        kv.free_slot(s)
    assert kv.free_page_count() == kv.n_pages
    assert len(kv._free_slots) == kv.max_slots
