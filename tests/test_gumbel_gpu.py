"""Gumbel-max sampler kernel + in-graph sampled decode (1 GPU).

VERDICT r01 item 5: the stochastic sampler is a hand-written kernel and
sampled workloads replay hipGraphs.  Exactness: with a supplied uniform
noise buffer the kernel must equal the fp32 reference
argmax(l/T - log(-log(u))) bit-for-bit in its argmax decision; without
it, the counter-based noise must reproduce per (seed, position) and
match the softmax distribution.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _hip():
    from ollamamq_amd.ops import hip
    hip.require()
    return hip


def test_gumbel_matches_fp32_reference_with_noise():
    hip = _hip()
    g = torch.Generator().manual_seed(11)
    for B, V in [(4, 128256), (32, 128256), (1, 512), (7, 32000)]:
        logits = (torch.randn(B, V, generator=g) * 3).bfloat16().cuda()
        temps = (torch.rand(B, generator=g) * 1.5 + 0.05).cuda()
        noise = torch.rand(B, V, generator=g).clamp(1e-7, 1 - 1e-7).cuda()
        seeds = torch.zeros(B, dtype=torch.int64, device="cuda")
        ctrs = torch.zeros(B, dtype=torch.int32, device="cuda")
        out = hip.sample_gumbel(logits, temps, seeds, ctrs, noise=noise)
        ref = (logits.float() / temps[:, None]
               - torch.log(-torch.log(noise.float()))).argmax(-1)
        assert torch.equal(out.long().cpu(), ref.cpu())
        # greedy rows (T<=0) ride the same kernel
        temps0 = torch.zeros(B, device="cuda")
        out0 = hip.sample_gumbel(logits, temps0, seeds, ctrs, noise=noise)
        assert torch.equal(out0.long().cpu(),
                           logits.float().argmax(-1).cpu())


def test_gumbel_counter_noise_reproducible_and_distributed():
    hip = _hip()
    B, V = 8, 512
    g = torch.Generator().manual_seed(5)
    logits = (torch.randn(1, V, generator=g) * 2).bfloat16().cuda() \
        .expand(B, V).contiguous()
    temps = torch.full((B,), 1.0, device="cuda")
    seeds = torch.arange(B, dtype=torch.int64, device="cuda") + 99
    ctrs = torch.full((B,), 17, dtype=torch.int32, device="cuda")
    a = hip.sample_gumbel(logits, temps, seeds, ctrs)
    b = hip.sample_gumbel(logits, temps, seeds, ctrs)
    assert torch.equal(a, b), "same (seed, ctr) must reproduce"
    c = hip.sample_gumbel(logits, temps, seeds, ctrs + 1)
    assert not torch.equal(a, c), "ctr change must redraw"
    # distribution: many draws over a small vocab ~ softmax
    n = 4000
    l1 = (torch.randn(1, 64, generator=g) * 1.5).bfloat16().cuda()
    big = l1.expand(n, 64).contiguous()
    t1 = torch.ones(n, device="cuda")
    sd = torch.arange(n, dtype=torch.int64, device="cuda")
    ct = torch.zeros(n, dtype=torch.int32, device="cuda")
    draws = hip.sample_gumbel(big, t1, sd, ct).cpu()
    emp = torch.bincount(draws.long(), minlength=64).float() / n
    expect = torch.softmax(l1[0].float().cpu(), -1)
    # loose L1 bound: 4000 draws over 64 bins
    assert (emp - expect).abs().sum() < 0.25, \
        (emp - expect).abs().sum()


def test_sampled_decode_replays_graphs():
    """Temperature-only sampled sequences keep the pipelined graph path
    (engine._samp_class==1): graphs capture once and replay; tokens are
    diverse and seeded requests reproduce across batch compositions."""
    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import GenParams, LlamaEngine, PagedKVCache
    cfg = PRESETS["tiny"]
    model = LlamaModel(cfg, device="cuda:0", dtype=torch.bfloat16,
                       seed=1234)
    kv = PagedKVCache.for_model(cfg, n_pages=128, max_slots=6,
                                max_ctx=256, device="cuda:0",
                                dtype=torch.bfloat16)
    eng = LlamaEngine(model, kv, max_batch=4)
    prompts = [[1, 2, 3, 4, 5], [9, 8, 7, 6], [3, 3, 3]]
    sids = [eng.submit(p, GenParams(max_tokens=12, temperature=0.9,
                                    seed=100 + i))
            for i, p in enumerate(prompts)]
    seqs = [eng.seqs[s] for s in sids]
    for _ in range(64):
        eng.step()
        if not eng.has_work():
            break
    torch.cuda.synchronize()
    toks_a = [list(s.generated) for s in seqs]
    assert all(len(t) == 12 for t in toks_a)
    # the pipelined sampled path captured a class-1 graph
    assert any(k[1] == 1 for k in eng._graphs), eng._graphs.keys()
    # diversity: stochastic rows should not all be a constant token
    assert any(len(set(t)) > 1 for t in toks_a), toks_a
    # reproducibility: seeded request rerun in a DIFFERENT batch
    # composition produces the same tokens
    sid = eng.submit(prompts[0], GenParams(max_tokens=12, temperature=0.9,
                                           seed=100))
    solo = eng.seqs[sid]
    for _ in range(64):
        eng.step()
        if not eng.has_work():
            break
    torch.cuda.synchronize()
    assert list(solo.generated) == toks_a[0], \
        (solo.generated, toks_a[0])


def test_topk_topp_decode_in_graph(monkeypatch):
    """Opt-in class-2 graphs (OLLAMAMQ_GRAPH_TOPK=1): the captured tail
    filters the top-256 candidates per row's own params and
    Gumbel-samples the filtered set.  (Off by default: the captured
    torch.topk faults under real multi-graph 8B serving — see
    engine._samp_class; this test pins the tiny-model mechanics so the
    path stays alive for the root-cause hunt.)"""
    monkeypatch.setenv("OLLAMAMQ_GRAPH_TOPK", "1")
    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import GenParams, LlamaEngine, PagedKVCache
    cfg = PRESETS["tiny"]
    model = LlamaModel(cfg, device="cuda:0", dtype=torch.bfloat16,
                       seed=1234)
    kv = PagedKVCache.for_model(cfg, n_pages=128, max_slots=6,
                                max_ctx=256, device="cuda:0",
                                dtype=torch.bfloat16)
    eng = LlamaEngine(model, kv, max_batch=4)
    n = 14
    params = [GenParams(max_tokens=n, temperature=0.8, top_k=1),
              GenParams(max_tokens=n, temperature=0.9, top_k=5,
                        seed=42),
              GenParams(max_tokens=n)]                     # greedy row
    sids = [eng.submit([2, 7, 1, 8, 2, 8], p) for p in params]
    seqs = [eng.seqs[s] for s in sids]
    for _ in range(80):
        eng.step()
        if not eng.has_work():
            break
    torch.cuda.synchronize()
    toks = [list(s.generated) for s in seqs]
    assert all(len(t) == n for t in toks), toks
    assert any(k[1] == 2 for k in eng._graphs), eng._graphs.keys()
    # row 0: top_k=1 == greedy; row 2 greedy: compare to a pure-greedy run
    eng2 = LlamaEngine(model, kv, max_batch=4)
    g_sid = eng2.submit([2, 7, 1, 8, 2, 8], GenParams(max_tokens=n))
    gseq = eng2.seqs[g_sid]
    for _ in range(80):
        eng2.step()
        if not eng2.has_work():
            break
    torch.cuda.synchronize()
    assert toks[0] == list(gseq.generated), "top_k=1 must equal greedy"
    assert toks[2] == list(gseq.generated), "greedy row drifted"
    # row 1 top_k=5: every decode token must be in its step's top-5.
    # (weak check: diversity only — exact support check needs per-step
    # logits; covered by sampler unit tests)
    assert len(set(toks[1])) >= 1


def test_host_capped_sampling_exact_support():
    """hip.sample's fast host path (top_k <= 256): every draw inside its
    row's top-k, top_k=1 equals argmax, greedy rows exact, deterministic
    under a fixed generator, and falls back to the sort path for pure
    nucleus rows."""
    hip = _hip()
    g = torch.Generator().manual_seed(3)
    B, V = 12, 128256
    logits = (torch.randn(B, V, generator=g) * 2).bfloat16().cuda()
    temps = torch.tensor([0.0, 1.0] + [0.8] * (B - 2), device="cuda")
    tk = torch.tensor([0, 1] + [7] * (B - 2), dtype=torch.long,
                      device="cuda")
    tp = torch.full((B,), 0.95, device="cuda")
    gen = torch.Generator(device="cuda")
    gen.manual_seed(9)
    out = hip.sample(logits, temps, tk, tp, gen)
    am = logits.float().argmax(-1)
    assert out[0] == am[0], "greedy row must be argmax"
    assert out[1] == am[1], "top_k=1 must be argmax"
    kth = logits.float().topk(7, dim=-1).values[:, -1]
    picked = logits.float().gather(1, out.unsqueeze(1)).squeeze(1)
    assert (picked[2:] >= kth[2:] - 1e-3).all(), "escaped top-7"
    gen2 = torch.Generator(device="cuda")
    gen2.manual_seed(9)
    out2 = hip.sample(logits, temps, tk, tp, gen2)
    assert torch.equal(out.cpu(), out2.cpu())
    # pure-nucleus rows (top_k=0, top_p<1) take the exact sort fallback
    tk0 = torch.zeros(B, dtype=torch.long, device="cuda")
    out3 = hip.sample(logits, temps, tk0, tp, gen)
    assert out3.shape[0] == B
