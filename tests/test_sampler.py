"""Sampler semantics on the fp32 reference path (greedy, temperature,
top-k, top-p) — the compute contract behind Ollama's options
(temperature/top_k/top_p, reference README options passthrough)."""
import torch

from ollamamq_amd.ops import reference as ref


def test_greedy_matches_argmax():
    g = torch.Generator().manual_seed(0)
    logits = torch.randn(5, 100, generator=g)
    out = ref.sample(logits, 0.0, 0, 1.0)
    assert torch.equal(out, logits.argmax(dim=-1))


def test_temperature_sampling_respects_support():
    # with a huge logit gap, low temperature must pick the max ~always
    logits = torch.full((4, 50), -10.0)
    logits[:, 7] = 10.0
    g = torch.Generator().manual_seed(1)
    out = ref.sample(logits, 0.5, 0, 1.0, generator=g)
    assert (out == 7).all()


def test_top_k_masks_tail():
    g = torch.Generator().manual_seed(2)
    logits = torch.arange(20.0).repeat(8, 1)   # monotonically increasing
    out = ref.sample(logits, 1.0, 3, 1.0, generator=g)
    # only the top-3 ids (17, 18, 19) are eligible
    assert set(out.tolist()) <= {17, 18, 19}


def test_top_p_nucleus():
    g = torch.Generator().manual_seed(3)
    logits = torch.zeros(64, 10)
    logits[:, 0] = 8.0       # ~99.9% of the mass
    logits[:, 1] = 2.0
    out = ref.sample(logits, 1.0, 0, 0.5, generator=g)
    assert (out == 0).all()  # nucleus of 0.5 is just token 0


def test_mixed_greedy_and_stochastic_rows():
    g = torch.Generator().manual_seed(4)
    logits = torch.randn(6, 30, generator=g)
    temps = torch.tensor([0.0, 1.0, 0.0, 1.0, 0.0, 1.0])
    out = ref.sample(logits, temps, 0, 1.0, generator=g)
    am = logits.argmax(dim=-1)
    for i in (0, 2, 4):
        assert out[i] == am[i]
    assert out.shape == (6,)
