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
