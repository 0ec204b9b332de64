"""Property-based checks of scheduler ordering invariants (C7/C18).

candidate_order (csrc/dispatcher/scheduler.cpp, reference semantics
src/dispatcher.rs:507-564) must uphold, for ANY active-user set, VIP/boost
assignment and counter value:
  - the result is a permutation of the active users (no drops, no dupes)
  - an active VIP is always first
  - an active boost user is first (after VIP) on even counters
  - with no VIP/boost, relative order of distinct-processed users follows
    least-served-first up to rotation
"""
import hypothesis.strategies as st
from hypothesis import given, settings

from ollamamq_amd.dispatch import load

d = load()

user_st = st.text(alphabet="abcdefgh", min_size=1, max_size=4)
active_st = st.lists(
    st.tuples(user_st, st.integers(min_value=0, max_value=10 ** 6)),
    min_size=1, max_size=12,
    unique_by=lambda t: t[0])


@settings(derandomize=True, max_examples=300, deadline=None)
@given(active=active_st, counter=st.integers(min_value=0, max_value=10 ** 9),
       pick_vip=st.integers(min_value=-1, max_value=11),
       pick_boost=st.integers(min_value=-1, max_value=11))
def test_candidate_order_invariants(active, counter, pick_vip, pick_boost):
    names = [u for u, _ in active]
    vip = names[pick_vip % len(names)] if pick_vip >= 0 else "zz-no-vip"
    boost = names[pick_boost % len(names)] if pick_boost >= 0 else "zz-nb"
    if boost == vip:  # mutually exclusive by construction in AppState
        boost = "zz-nb"
    order = d.candidate_order(active, vip, boost, counter)

    assert sorted(order) == sorted(names)  # permutation, always

    if vip in names:
        assert order[0] == vip
    if boost in names and counter % 2 == 0:
        expect_at = 1 if vip in names else 0
        assert order[expect_at] == boost


@settings(derandomize=True, max_examples=200, deadline=None)
@given(active=active_st, counter=st.integers(min_value=0, max_value=10 ** 9))
def test_least_served_up_to_rotation(active, counter):
    order = d.candidate_order(active, "zz-no-vip", "zz-nb", counter)
    processed = dict(active)
    # the schedule is a rotation of the least-served-first sort: some split
    # point k exists where order[k:] + order[:k] is sorted by processed asc
    n = len(order)
    ok = any(
        all(processed[rot[i]] <= processed[rot[i + 1]]
            for i in range(n - 1))
        for k in range(n)
        for rot in [order[k:] + order[:k]])
    assert ok, (order, processed)
