"""Property-based sampler invariants (hypothesis): whatever the logits,
temperature, top-k and top-p, the sampler must return in-vocabulary ids,
respect the top-k support, and be deterministic under a fixed generator."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from ollamamq_amd.ops import reference as ref


@settings(derandomize=True, max_examples=40, deadline=None)
@given(st.integers(1, 8), st.integers(2, 200), st.integers(0, 10 ** 6),
       st.floats(0.05, 4.0), st.integers(0, 8),
       st.floats(0.05, 1.0))
def test_sampler_invariants(B, V, seed, temp, top_k, top_p):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(B, V, generator=g) * 4
    gs = torch.Generator().manual_seed(seed + 1)
    out = ref.sample(logits, temp, top_k, top_p, gs)
    assert out.shape == (B,)
    assert ((out >= 0) & (out < V)).all(), "out-of-vocabulary id"
    if top_k and top_k < V:
        # every sampled id must be among that row's top-k logits
        kth = logits.topk(top_k, dim=-1).values[:, -1]
        picked = logits.gather(1, out.unsqueeze(1)).squeeze(1)
        assert (picked >= kth - 1e-6).all(), "sampled outside top-k"
    # determinism under the same generator state
    gs2 = torch.Generator().manual_seed(seed + 1)
    out2 = ref.sample(logits, temp, top_k, top_p, gs2)
    assert torch.equal(out, out2)


@settings(derandomize=True, max_examples=25, deadline=None)
@given(st.integers(1, 6), st.integers(2, 100), st.integers(0, 10 ** 6))
def test_greedy_always_argmax(B, V, seed):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(B, V, generator=g)
    out = ref.sample(logits, 0.0, 0, 1.0)
    assert torch.equal(out, logits.argmax(dim=-1))


@settings(derandomize=True, max_examples=30, deadline=None)
@given(st.integers(2, 8), st.integers(8, 200), st.integers(0, 10 ** 6))
def test_mixed_per_request_params(B, V, seed):
    """rows carry DIFFERENT top_k/top_p in one batch;
    each row must honor its own filter (no engine-wide max/min bleed)."""
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(B, V, generator=g) * 4
    temps = torch.full((B,), 1.0)
    # row 0: top_k=1 (must be argmax); row 1: top_p tiny (must be argmax);
    # remaining rows: top_k=5 (must be within their own top-5)
    top_k = torch.tensor([1, 0] + [5] * (B - 2), dtype=torch.long)
    top_p = torch.tensor([1.0, 1e-9] + [1.0] * (B - 2))
    gs = torch.Generator().manual_seed(seed + 7)
    out = ref.sample(logits, temps, top_k, top_p, gs)
    am = logits.argmax(dim=-1)
    assert out[0] == am[0], "top_k=1 row must be greedy"
    assert out[1] == am[1], "tiny top_p row must be greedy"
    if B > 2 and V > 5:
        kth = logits.topk(5, dim=-1).values[:, -1]
        picked = logits.gather(1, out.unsqueeze(1)).squeeze(1)
        assert (picked[2:] >= kth[2:] - 1e-6).all(), \
            "row escaped its own top-5 (param bleed)"
    # determinism with per-row tensors too
    gs2 = torch.Generator().manual_seed(seed + 7)
    assert torch.equal(out, ref.sample(logits, temps, top_k, top_p, gs2))
