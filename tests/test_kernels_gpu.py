"""Numerics: each gfx950 HIP kernel vs the plain-PyTorch fp32 reference.

Runs only on a real MI355X (pytest -m gpu via gpurun).  Inputs are random
bf16; the oracle computes in fp32 from the same bf16-rounded values, so
tolerances cover only the kernel's own bf16 output rounding + fp32
reduction-order differences.
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from ollamamq_amd.ops import reference as ref
from ollamamq_amd.engine.kvcache import PagedKVCache
from ollamamq_amd.ops.interface import AttnMeta


def dev():
    return "cuda:0"


def _hip():
    from ollamamq_amd.ops import hip
    hip.require()
    return hip


def rnd(*shape, scale=1.0, seed=0):
    g = torch.Generator(device="cpu")
    g.manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).bfloat16().to(dev())


def test_rmsnorm_residual():
    hip = _hip()
    for T, H in [(1, 4096), (33, 4096), (256, 512), (7, 8192)]:
        x = rnd(T, H, seed=T)
        res = rnd(T, H, seed=T + 1)
        w = rnd(H, scale=0.5, seed=T + 2) + 1.0
        y, r2 = hip.rmsnorm_residual(x, res, w.bfloat16(), 1e-5)
        y_ref, r_ref = ref.rmsnorm_residual(
            x.float().cpu(), res.float().cpu(), w.float().cpu(), 1e-5)
        torch.testing.assert_close(r2.float().cpu(), r_ref.float(),
                                   atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(y.float().cpu(), y_ref.float(),
                                   atol=5e-2, rtol=5e-2)
        # no-residual path
        y0, r0 = hip.rmsnorm_residual(x, None, w.bfloat16(), 1e-5)
        y0_ref, _ = ref.rmsnorm_residual(x.float().cpu(), None,
                                         w.float().cpu(), 1e-5)
        torch.testing.assert_close(y0.float().cpu(), y0_ref.float(),
                                   atol=5e-2, rtol=5e-2)


def test_rope():
    hip = _hip()
    T, Hq, Hk, D = 17, 4, 2, 128
    maxp = 128
    pos = torch.randint(0, maxp, (T,), dtype=torch.int32).to(dev())
    ang = torch.outer(torch.arange(maxp, dtype=torch.float32),
                      1.0 / 10000 ** (torch.arange(0, D, 2) / D))
    cos, sin = ang.cos().to(dev()), ang.sin().to(dev())
    q = rnd(T, Hq, D, seed=3)
    k = rnd(T, Hk, D, seed=4)
    q_ref, k_ref = q.float().cpu(), k.float().cpu()
    ref.rope(q_ref, k_ref, pos.cpu(), cos.cpu(), sin.cpu())
    hip.rope(q, k, pos, cos, sin)
    torch.testing.assert_close(q.float().cpu(), q_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(k.float().cpu(), k_ref, atol=2e-2, rtol=2e-2)


def make_caches(L=2, KVH=2, D=128, n_pages=64, slots=4, ctx=512):
    gc = PagedKVCache(L, KVH, D, page_size=16, n_pages=n_pages,
                      max_slots=slots, max_ctx=ctx, device=dev(),
                      dtype=torch.bfloat16)
    cc = PagedKVCache(L, KVH, D, page_size=16, n_pages=n_pages,
                      max_slots=slots, max_ctx=ctx, device="cpu",
                      dtype=torch.float32)
    return gc, cc


def fill_caches(gc, cc, lens, L=2, KVH=2, D=128, seed=9):
    """Append `lens[i]` tokens for slot i into both caches, layer by layer."""
    for i, n in enumerate(lens):
        sg, sc = gc.alloc_slot(), cc.alloc_slot()
        assert sg == sc
        gc.ensure(sg, n)
        cc.ensure(sc, n)
    cc.page_table.copy_(gc.page_table.cpu())
    from ollamamq_amd.ops import hip
    for li in range(L):
        for i, n in enumerate(lens):
            k = rnd(n, KVH, D, seed=seed + li * 97 + i)
            v = rnd(n, KVH, D, seed=seed + li * 97 + i + 31)
            slot = torch.full((n,), i, dtype=torch.int32, device=dev())
            pos = torch.arange(n, dtype=torch.int32, device=dev())
            hip.kv_append(gc, li, k, v, slot, pos)
            ref.kv_append(cc, li, k.float().cpu(), v.float().cpu(),
                          slot.cpu(), pos.cpu())


def test_kv_append_matches():
    _hip()
    gc, cc = make_caches()
    fill_caches(gc, cc, [20, 33, 5])
    torch.testing.assert_close(gc.k_pool.float().cpu(), cc.k_pool,
                               atol=0, rtol=0)
    torch.testing.assert_close(gc.v_pool.float().cpu(), cc.v_pool,
                               atol=0, rtol=0)


def _meta(device, slots, seq_lens, q_lens):
    cu = [0]
    for ql in q_lens:
        cu.append(cu[-1] + ql)
    return AttnMeta(
        mode="decode" if max(q_lens) == 1 else "prefill",
        slot_ids=torch.tensor(slots, dtype=torch.int32, device=device),
        seq_lens=torch.tensor(seq_lens, dtype=torch.int32, device=device),
        cu_q=torch.tensor(cu, dtype=torch.int32, device=device),
        logits_idx=None, max_q=max(q_lens), max_kv=max(seq_lens),
    )


def test_attention_decode():
    hip = _hip()
    Hq, KVH, D = 4, 2, 128
    lens = [1, 16, 17, 129]          # page boundaries + singleton
    gc, cc = make_caches(KVH=KVH)
    fill_caches(gc, cc, lens, KVH=KVH)
    S = len(lens)
    q = rnd(S, Hq, D, seed=77)
    meta_g = _meta(dev(), list(range(S)), lens, [1] * S)
    meta_c = _meta("cpu", list(range(S)), lens, [1] * S)
    out = hip.attention_decode(q, gc, 1, meta_g)
    out_ref = ref.attention(q.float().cpu(), cc, 1, meta_c)
    torch.testing.assert_close(out.float().cpu(), out_ref,
                               atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("Hq,KVH", [(2, 2), (4, 2), (8, 2), (16, 2), (14, 2)])
def test_attention_decode_gqa_groups(Hq, KVH):
    """All decode-kernel template instantiations (GQA group G=Hq/KVH in
    {1,2,4,8}: G>=4 takes the register-prefetch pipeline, G<4 the direct
    staging path) against the fp32 reference, ragged lengths."""
    hip = _hip()
    D = 128
    lens = [3, 64, 200, 516]
    gc, cc = make_caches(KVH=KVH, n_pages=256, ctx=1024)
    fill_caches(gc, cc, lens, KVH=KVH)
    S = len(lens)
    q = rnd(S, Hq, D, seed=100 + Hq)
    meta_g = _meta(dev(), list(range(S)), lens, [1] * S)
    meta_c = _meta("cpu", list(range(S)), lens, [1] * S)
    out = hip.attention_decode(q, gc, 1, meta_g)
    out_ref = ref.attention(q.float().cpu(), cc, 1, meta_c)
    torch.testing.assert_close(out.float().cpu(), out_ref,
                               atol=3e-2, rtol=3e-2)


def test_attention_decode_sliding_window():
    """Decode kernel with a sliding window vs the fp32 reference; lengths
    straddle the window so active chunk ranges differ per sequence."""
    hip = _hip()
    Hq, KVH, D, W = 8, 2, 128, 100
    lens = [40, 100, 101, 300]       # below, at, above, far above window
    gc, cc = make_caches(KVH=KVH, n_pages=128, ctx=512)
    fill_caches(gc, cc, lens, KVH=KVH)
    S = len(lens)
    q = rnd(S, Hq, D, seed=55)
    meta_g = _meta(dev(), list(range(S)), lens, [1] * S)
    meta_c = _meta("cpu", list(range(S)), lens, [1] * S)
    meta_g.window = W
    meta_c.window = W
    out = hip.attention_decode(q, gc, 0, meta_g)
    out_ref = ref.attention(q.float().cpu(), cc, 0, meta_c)
    torch.testing.assert_close(out.float().cpu(), out_ref,
                               atol=3e-2, rtol=3e-2)


def test_attention_prefill_sliding_window():
    """MFMA prefill with a window: long prompt, interior-chunk skip and
    the per-element window mask both exercised."""
    hip = _hip()
    Hq, KVH, D, W = 4, 2, 128, 70
    kv_lens = [200, 37]
    q_lens = [200, 37]
    gc, cc = make_caches(KVH=KVH, n_pages=128, ctx=512)
    fill_caches(gc, cc, kv_lens, KVH=KVH)
    T = sum(q_lens)
    q = rnd(T, Hq, D, seed=66)
    slots = list(range(len(kv_lens)))
    meta_g = _meta(dev(), slots, kv_lens, q_lens)
    meta_c = _meta("cpu", slots, kv_lens, q_lens)
    meta_g.window = W
    meta_c.window = W
    out = hip.attention_prefill(q, gc, 1, meta_g)
    out_ref = ref.attention(q.float().cpu(), cc, 1, meta_c)
    torch.testing.assert_close(out.float().cpu(), out_ref,
                               atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("preset", ["tiny-qwen", "tiny-swa"])
def test_family_forward_gpu_vs_cpu(preset):
    """Qwen2-bias / Mistral-window variants end to end on the HIP path:
    greedy tokens from a long prompt must track the CPU fp32 engine."""
    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams
    cfg = PRESETS[preset]
    prompt = list(range(2, 122))     # 120 tokens: > tiny-swa window (96)

    def run(device, dtype):
        model = LlamaModel(cfg, device=device, dtype=dtype, seed=11)
        kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=4,
                                    max_ctx=cfg.max_ctx, device=device,
                                    dtype=dtype)
        eng = LlamaEngine(model, kv, max_batch=4)
        sid = eng.submit(prompt, GenParams(max_tokens=8))
        seq = eng.seqs[sid]
        for _ in range(64):
            eng.step()
            if not eng.has_work():
                break
        return seq.generated

    gpu = run(dev(), torch.bfloat16)
    cpu = run("cpu", torch.float32)
    assert len(gpu) == len(cpu) == 8
    # greedy ties can flip late under bf16; the first tokens must agree
    assert gpu[:2] == cpu[:2], f"{gpu} vs {cpu}"


def test_attention_prefill_odd_gqa_group():
    """G=7 (Qwen2-7B's 28Q/4KV): MFMA prefill with clamped staging
    slots (NSLOT % NTHR != 0) must match the fp32 reference."""
    hip = _hip()
    Hq, KVH, D = 14, 2, 128
    kv_lens = [64, 130]
    q_lens = [64, 130]
    gc, cc = make_caches(KVH=KVH, n_pages=128, ctx=512)
    fill_caches(gc, cc, kv_lens, KVH=KVH)
    q = rnd(sum(q_lens), Hq, D, seed=21)
    slots = list(range(len(kv_lens)))
    meta_g = _meta(dev(), slots, kv_lens, q_lens)
    meta_c = _meta("cpu", slots, kv_lens, q_lens)
    out = hip.attention_prefill(q, gc, 0, meta_g)
    out_ref = ref.attention(q.float().cpu(), cc, 0, meta_c)
    torch.testing.assert_close(out.float().cpu(), out_ref,
                               atol=3e-2, rtol=3e-2)


def test_attention_prefill_varlen():
    hip = _hip()
    Hq, KVH, D = 4, 2, 128
    # continued prefill: slot 1 already has 40 tokens of context, the new
    # chunk is 23 query tokens at positions 40..62
    kv_lens = [37, 63, 130]
    q_lens = [37, 23, 130]
    gc, cc = make_caches(KVH=KVH)
    fill_caches(gc, cc, kv_lens, KVH=KVH)
    T = sum(q_lens)
    q = rnd(T, Hq, D, seed=5)
    slots = list(range(len(kv_lens)))
    meta_g = _meta(dev(), slots, kv_lens, q_lens)
    meta_c = _meta("cpu", slots, kv_lens, q_lens)
    out = hip.attention_prefill(q, gc, 0, meta_g)
    out_ref = ref.attention(q.float().cpu(), cc, 0, meta_c)
    torch.testing.assert_close(out.float().cpu(), out_ref,
                               atol=3e-2, rtol=3e-2)


def test_swiglu():
    hip = _hip()
    for T, F in [(5, 1024), (64, 14336)]:
        gu = rnd(T, 2 * F, seed=F)
        out = hip.swiglu(gu)
        out_ref = ref.swiglu(gu.float().cpu())
        torch.testing.assert_close(out.float().cpu(), out_ref,
                                   atol=2e-2, rtol=2e-2)


def test_argmax_sampler():
    hip = _hip()
    for B, V in [(1, 128256), (32, 128256), (7, 513)]:
        logits = rnd(B, V, scale=3.0, seed=V + B)
        out = hip.sample(logits, 0.0, 0, 1.0)
        expect = logits.float().argmax(dim=-1)
        assert torch.equal(out.cpu(), expect.cpu())


def test_forward_e2e_vs_cpu():
    """Whole tiny-model forward GPU-bf16 vs CPU-fp32: logits must agree to
    bf16 tolerance (cosine > 0.995 per row)."""
    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams
    cfg = PRESETS["tiny"]

    def run(device, dtype):
        model = LlamaModel(cfg, device=device, dtype=dtype, seed=11)
        kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=4,
                                    max_ctx=256, device=device, dtype=dtype)
        eng = LlamaEngine(model, kv, max_batch=4)
        # pipelined decode samples in-graph and never calls _sample; this
        # test captures logits through _sample, so use the plain path
        # (pipelined-vs-plain equivalence is covered by test_pipeline_gpu)
        eng.use_pipeline = False
        eng.submit(list(range(1, 33)), GenParams(max_tokens=4))
        eng.submit(list(range(40, 52)), GenParams(max_tokens=4))
        logits_log = []
        orig_sample = eng._sample

        def capture(seqs, logits):
            logits_log.append(logits.float().cpu())
            return orig_sample(seqs, logits)

        eng._sample = capture
        for _ in range(64):
            eng.step()
            if not eng.has_work():
                break
        return logits_log

    gpu_logits = run(dev(), torch.bfloat16)
    cpu_logits = run("cpu", torch.float32)
    assert len(gpu_logits) == len(cpu_logits)
    # Step 0 (prefill logits) sees identical inputs on both paths; later
    # steps may legitimately diverge once a greedy tie flips a token.
    a, b = gpu_logits[0], cpu_logits[0]
    cos = torch.nn.functional.cosine_similarity(a, b, dim=-1)
    assert cos.min() > 0.995, f"prefill logits diverged: {cos.min()}"


def test_fused_rope_append():
    """Fused rope+append == reference rope then append."""
    hip = _hip()
    T, Hq, KVH, D = 13, 4, 2, 128
    gc, cc = make_caches(KVH=KVH)
    sg, sc = gc.alloc_slot(), cc.alloc_slot()
    gc.ensure(sg, T)
    cc.ensure(sc, T)
    cc.page_table.copy_(gc.page_table.cpu())
    g = torch.Generator().manual_seed(8)
    qkv = torch.randn(T, (Hq + 2 * KVH) * D, generator=g).bfloat16().to(dev())
    q = qkv[:, :Hq * D].view(T, Hq, D)
    k = qkv[:, Hq * D:(Hq + KVH) * D].view(T, KVH, D)
    v = qkv[:, (Hq + KVH) * D:].view(T, KVH, D)
    pos = torch.arange(T, dtype=torch.int32, device=dev())
    slot = torch.zeros(T, dtype=torch.int32, device=dev())
    ang = torch.outer(torch.arange(64, dtype=torch.float32),
                      1.0 / 10000 ** (torch.arange(0, D, 2) / D))
    cos, sin = ang.cos().to(dev()), ang.sin().to(dev())
    q_ref = q.float().cpu().clone()
    k_ref = k.float().cpu().clone()
    v_ref = v.float().cpu().clone()
    ref.rope(q_ref, k_ref, pos.cpu(), cos.cpu(), sin.cpu())
    ref.kv_append(cc, 0, k_ref, v_ref, slot.cpu(), pos.cpu())
    hip.rope_append(gc, 0, q, k, v, pos, slot, cos, sin)
    torch.testing.assert_close(q.float().cpu(), q_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(k.float().cpu(), k_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(gc.k_pool.float().cpu(), cc.k_pool,
                               atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(gc.v_pool.float().cpu(), cc.v_pool,
                               atol=0, rtol=0)


def test_mixed_batch_continuous_batching():
    """A sequence admitted mid-decode (mixed prefill+decode forward) must
    not change the earlier sequence's greedy tokens, and the late joiner
    must produce the same tokens as when run alone."""
    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams
    cfg = PRESETS["tiny"]
    p1, p2 = list(range(1, 21)), list(range(30, 45))

    def fresh():
        model = LlamaModel(cfg, device="cuda:0", dtype=torch.bfloat16,
                           seed=21)
        kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=4,
                                    max_ctx=256, device="cuda:0",
                                    dtype=torch.bfloat16)
        return LlamaEngine(model, kv, max_batch=4)

    # run each alone
    alone = {}
    for name, p in (("s1", p1), ("s2", p2)):
        eng = fresh()
        sid = eng.submit(p, GenParams(max_tokens=8))
        seq = eng.seqs[sid]
        for _ in range(64):
            eng.step()
            if not eng.has_work():
                break
        alone[name] = seq.generated

    # staggered: submit s2 after s1 has decoded a few tokens -> the
    # admission step is a MIXED forward
    eng = fresh()
    sid1 = eng.submit(p1, GenParams(max_tokens=8))
    seq1 = eng.seqs[sid1]
    for _ in range(4):
        eng.step()
    sid2 = eng.submit(p2, GenParams(max_tokens=8))
    seq2 = eng.seqs[sid2]
    for _ in range(64):
        eng.step()
        if not eng.has_work():
            break
    assert not eng.has_work()
    assert seq1.generated == alone["s1"], "mixed step changed s1 tokens"
    assert seq2.generated == alone["s2"], "late joiner diverged"


def test_embed_gather():
    """Embedding gather kernel vs index_select (exact: it is a copy)."""
    hip = _hip()
    for T, V, H in [(1, 512, 4096), (32, 128256, 4096), (2048, 512, 512),
                    (7, 1000, 3584)]:
        g = torch.Generator().manual_seed(T)
        table = rnd(V, H, seed=V + T)
        toks = torch.randint(0, V, (T,), generator=g,
                             dtype=torch.int32).to(dev())
        out = hip.embedding(toks, table)
        expect = table.index_select(0, toks.long())
        assert torch.equal(out, expect)


def test_wstream_gemm():
    """Weight-streaming decode GEMM over packed weights vs hipBLASLt,
    across the real projection shapes (+ odd M, bias, uneven K ranges)."""
    hip = _hip()
    shapes = [
        (32, 6144, 4096),     # qkv 8B
        (32, 4096, 4096),     # o 8B
        (32, 28672, 4096),    # gate_up 8B
        (32, 4096, 14336),    # down 8B
        (7, 1024, 512),       # tiny + odd M
        (1, 4096, 4096),      # batch 1
        (33, 4096, 4096),     # MT=2 path
        (64, 4736, 3584),     # qwen-ish N/K (K%512 != 0 -> uneven waves)
        (48, 4096, 11008),    # llama2 down (K/64 = 172, uneven)
    ]
    for M, N, K in shapes:
        x = rnd(M, K, seed=M + N)
        w = rnd(N, K, seed=K + N)
        pk = hip.pack_weight(w)
        assert pk is not None
        y = hip.linear_packed(x, pk, None, N)
        expect = torch.nn.functional.linear(x, w)
        torch.testing.assert_close(y.float(), expect.float(),
                                   atol=8e-2, rtol=8e-2)
    # bias path (qwen qkv)
    M, N, K = 32, 4736, 3584
    x = rnd(M, K, seed=1)
    w = rnd(N, K, seed=2)
    b = rnd(N, seed=3)
    y = hip.linear_packed(x, hip.pack_weight(w), b, N)
    expect = torch.nn.functional.linear(x, w, b)
    torch.testing.assert_close(y.float(), expect.float(),
                               atol=8e-2, rtol=8e-2)


def test_wstream_pack_roundtrip_cpu_check():
    """The pack permutation is exactly invertible (layout sanity)."""
    hip = _hip()
    N, K = 64, 128
    w = rnd(N, K, seed=9)
    pk = hip.pack_weight(w)
    # invert: [t, b, j, h, r, e] -> [t, r, b, j, h, e]
    back = pk.view(N // 32, K // 64, 4, 2, 32, 8) \
             .permute(0, 4, 1, 2, 3, 5).reshape(N, K)
    assert torch.equal(back, w)


def test_wstream_gu_fused():
    """Fused gate_up+SwiGLU kernel vs F.linear + fp32-reference swiglu."""
    hip = _hip()
    for M, F, K in [(32, 14336, 4096), (1, 14336, 4096), (7, 512, 512),
                    (32, 9472, 3584), (64, 14336, 4096),
                    (47, 512, 512)]:
        x = rnd(M, K, seed=M + F)
        w = rnd(2 * F, K, seed=K + F)
        pk = hip.pack_weight_gu(w)
        assert pk is not None
        act = hip.linear_gu(x, pk, 2 * F)
        gu = torch.nn.functional.linear(x.float(), w.float())
        g, u = gu[:, :F], gu[:, F:]
        expect = torch.nn.functional.silu(g) * u
        torch.testing.assert_close(act.float(), expect,
                                   atol=8e-2, rtol=8e-2)


def _defrag(flat, M, K):
    """Read a 32-row frag-layout buffer back to [M, K] (test helper)."""
    got = torch.empty(M, K)
    fv = flat.reshape(-1).float().cpu()
    for m in range(M):
        for n in range(K):
            b, j, h, e = n // 64, (n // 16) % 4, (n // 8) % 2, n % 8
            u = ((b * 4 + j) * 64) + h * 32 + m
            got[m, n] = fv[u * 8 + e]
    return got


def test_frag_layout_roundtrip():
    """The frag-chain primitives agree on ONE layout: fragify output,
    XF GEMM input, yfrag GEMM output, fused-GU output and the decode
    attention fragout all use frag_off(m, k)."""
    hip = _hip()
    g = torch.Generator().manual_seed(3)
    M, K, N = 5, 512, 1024
    x = (torch.randn(M, K, generator=g) * 0.5).bfloat16().cuda()
    xf, sq = hip.fragify_sumsq(x)
    torch.testing.assert_close(sq.cpu(), (x.float().cpu() ** 2).sum(-1),
                               rtol=1e-2, atol=1e-2)
    w = (torch.randn(N, K, generator=g) * 0.1).bfloat16().cuda()
    pk = hip.pack_weight(w)
    y_std = hip.linear_packed(x, pk, None, N)
    # frag input == std input
    y_xf = hip.linear_packed(xf, pk, None, N, K=K, xlds=2)
    torch.testing.assert_close(y_xf[:M].float(), y_std.float(),
                               atol=5e-2, rtol=5e-2)
    # frag output round-trips
    y_fr = hip.linear_packed(xf, pk, None, N, K=K, xlds=2, yfrag=1)
    torch.testing.assert_close(_defrag(y_fr, M, N), y_std.float().cpu(),
                               atol=5e-2, rtol=5e-2)
    # fused GU with frag in/out
    F = 512
    wg = (torch.randn(2 * F, K, generator=g) * 0.1).bfloat16().cuda()
    gpk = hip.pack_weight_gu(wg)
    act_std = hip.linear_gu(x, gpk, 2 * F)
    act_fr = hip.linear_gu(xf, gpk, 2 * F, K=K, yfrag=1)
    torch.testing.assert_close(_defrag(act_fr, M, F),
                               act_std.float().cpu(), atol=5e-2,
                               rtol=5e-2)


def test_decode_attn_fragout():
    """attention_decode(fragout=True) matches the standard output."""
    hip = _hip()
    from ollamamq_amd.models import PRESETS
    from ollamamq_amd.engine.kvcache import PagedKVCache
    cfg = PRESETS["tiny"]
    kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=4, max_ctx=128,
                                device="cuda", dtype=torch.bfloat16)
    g = torch.Generator().manual_seed(5)
    B, L = 3, 9
    slots = [kv.alloc_slot() for _ in range(B)]
    for s_ in slots:
        kv.ensure(s_, L)
    kv.k_pool.normal_(0, 0.3)
    kv.v_pool.normal_(0, 0.3)
    q = (torch.randn(B, cfg.n_heads, 128, generator=g) * 0.3) \
        .bfloat16().cuda()
    meta = AttnMeta(
        mode="decode",
        slot_ids=torch.tensor(slots, dtype=torch.int32, device="cuda"),
        seq_lens=torch.tensor([L] * B, dtype=torch.int32, device="cuda"),
        cu_q=torch.arange(B + 1, dtype=torch.int32, device="cuda"),
        logits_idx=None, max_q=1, max_kv=L, window=0)
    a_std = hip.attention_decode(q, kv, 0, meta)
    a_fr = hip.attention_decode(q, kv, 0, meta, fragout=True)
    Kh = cfg.n_heads * 128
    torch.testing.assert_close(_defrag(a_fr, B, Kh),
                               a_std.view(B, Kh).float().cpu(),
                               atol=5e-2, rtol=5e-2)
