"""Model-family variants on the shared Llama block: Qwen2 (QKV bias) and
Mistral (sliding-window attention).  CPU fp32; the same presets run on
GPU in tests/test_kernels_gpu.py::test_family_forward_gpu_vs_cpu."""
import torch

from ollamamq_amd.models import LlamaModel, PRESETS
from ollamamq_amd.models.llama import LlamaLayer
from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams
from ollamamq_amd.ops import reference as ref
from ollamamq_amd.ops.interface import AttnMeta


def make_engine(preset, max_slots=4):
    cfg = PRESETS[preset]
    model = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=3)
    kv = PagedKVCache.for_model(cfg, n_pages=128, max_slots=max_slots,
                                max_ctx=cfg.max_ctx)
    return LlamaEngine(model, kv, max_batch=max_slots)


def run_all(eng, max_steps=400):
    for _ in range(max_steps):
        eng.step()
        if not eng.has_work():
            break
    assert not eng.has_work()


def test_qwen_bias_changes_output():
    """The bias must actually participate: zeroing it changes logits."""
    cfg = PRESETS["tiny-qwen"]
    assert cfg.qkv_bias
    m = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=3)
    assert all(l.bqkv is not None for l in m.layers)
    eng = make_engine("tiny-qwen")
    s = eng.submit([5, 6, 7, 8], GenParams(max_tokens=6))
    seq = eng.seqs[s]
    run_all(eng)
    base = list(seq.generated)

    eng2 = make_engine("tiny-qwen")
    for l in eng2.model.layers:
        l.bqkv.zero_()
    s2 = eng2.submit([5, 6, 7, 8], GenParams(max_tokens=6))
    seq2 = eng2.seqs[s2]
    run_all(eng2)
    assert len(base) == len(seq2.generated) == 6
    # same weights, bias zeroed: the function must differ
    assert base != seq2.generated


def test_qwen_bias_tp_shards_match_tp1():
    """cat of the TP=2 bias shards == the TP=1 bias (same seeded global)."""
    cfg = PRESETS["tiny-qwen"]
    gen = torch.Generator()
    gen.manual_seed(42)
    l1 = LlamaLayer(cfg, "cpu", torch.float32, gen, tp=1, rank=0)
    gen.manual_seed(42)
    r0 = LlamaLayer(cfg, "cpu", torch.float32, gen, tp=2, rank=0)
    gen.manual_seed(42)
    r1 = LlamaLayer(cfg, "cpu", torch.float32, gen, tp=2, rank=1)
    d, nh, nkv = cfg.head_dim, cfg.n_heads // 2, cfg.n_kv_heads // 2
    q = torch.cat([r0.bqkv[:nh * d], r1.bqkv[:nh * d]])
    k = torch.cat([r0.bqkv[nh * d:(nh + nkv) * d],
                   r1.bqkv[nh * d:(nh + nkv) * d]])
    v = torch.cat([r0.bqkv[(nh + nkv) * d:], r1.bqkv[(nh + nkv) * d:]])
    full = torch.cat([q, k, v])
    torch.testing.assert_close(full, l1.bqkv, atol=0, rtol=0)


def _dense_window_attention(q, k, v, window, kv_len, qlen):
    """Independent dense oracle for the sliding-window mask."""
    Hq, D = q.shape[1], q.shape[2]
    rep = Hq // k.shape[1]
    k = k.repeat_interleave(rep, dim=1).float()
    v = v.repeat_interleave(rep, dim=1).float()
    scores = torch.einsum("qhd,khd->hqk", q.float(), k) / (D ** 0.5)
    qpos = torch.arange(kv_len - qlen, kv_len)
    kpos = torch.arange(kv_len)
    mask = (kpos[None, :] > qpos[:, None]) | \
           (kpos[None, :] < qpos[:, None] - window + 1)
    scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
    return torch.einsum("hqk,khd->qhd", torch.softmax(scores, -1), v)


def test_reference_window_attention_vs_dense():
    torch.manual_seed(0)
    KVH, Hq, D, W = 2, 4, 128, 17
    kv_len, qlen = 60, 23
    cache = PagedKVCache(1, KVH, D, page_size=16, n_pages=8, max_slots=1,
                         max_ctx=128)
    slot = cache.alloc_slot()
    cache.ensure(slot, kv_len)
    k = torch.randn(kv_len, KVH, D)
    v = torch.randn(kv_len, KVH, D)
    ref.kv_append(cache, 0, k, v,
                  torch.zeros(kv_len, dtype=torch.int32),
                  torch.arange(kv_len, dtype=torch.int32))
    q = torch.randn(qlen, Hq, D)
    meta = AttnMeta(
        mode="prefill",
        slot_ids=torch.tensor([slot], dtype=torch.int32),
        seq_lens=torch.tensor([kv_len], dtype=torch.int32),
        cu_q=torch.tensor([0, qlen], dtype=torch.int32),
        logits_idx=None, max_q=qlen, max_kv=kv_len, window=W)
    out = ref.attention(q, cache, 0, meta)
    expect = _dense_window_attention(q, k, v, W, kv_len, qlen)
    torch.testing.assert_close(out.float(), expect, atol=1e-4, rtol=1e-4)


def test_swa_engine_differs_from_full_attention():
    """With a prompt longer than the window, the windowed model must
    diverge from an identical model with the window disabled."""
    cfg = PRESETS["tiny-swa"]
    assert cfg.sliding_window == 96
    prompt = list(torch.randint(0, 500, (150,),
                                generator=torch.Generator().manual_seed(1))
                  .tolist())
    eng = make_engine("tiny-swa")
    s = eng.submit(prompt, GenParams(max_tokens=8))
    seq = eng.seqs[s]
    run_all(eng)

    # same seed/arch but full attention
    import dataclasses
    full_cfg = dataclasses.replace(cfg, sliding_window=0)
    model = LlamaModel(full_cfg, device="cpu", dtype=torch.float32, seed=3)
    kv = PagedKVCache.for_model(full_cfg, n_pages=128, max_slots=4,
                                max_ctx=full_cfg.max_ctx)
    eng2 = LlamaEngine(model, kv, max_batch=4)
    s2 = eng2.submit(prompt, GenParams(max_tokens=8))
    seq2 = eng2.seqs[s2]
    run_all(eng2)
    assert len(seq.generated) == len(seq2.generated) == 8
    assert seq.generated != seq2.generated


def test_swa_engine_equals_full_when_short():
    """Prompts shorter than the window see no masking difference."""
    prompt = [9, 8, 7, 6, 5]
    eng = make_engine("tiny-swa")
    s = eng.submit(prompt, GenParams(max_tokens=30))
    seq = eng.seqs[s]
    run_all(eng)

    import dataclasses
    full_cfg = dataclasses.replace(PRESETS["tiny-swa"], sliding_window=0)
    model = LlamaModel(full_cfg, device="cpu", dtype=torch.float32, seed=3)
    kv = PagedKVCache.for_model(full_cfg, n_pages=128, max_slots=4,
                                max_ctx=full_cfg.max_ctx)
    eng2 = LlamaEngine(model, kv, max_batch=4)
    s2 = eng2.submit(prompt, GenParams(max_tokens=30))
    seq2 = eng2.seqs[s2]
    run_all(eng2)
    # total_len = 5 + 30 = 35 < 96: identical trajectories
    assert seq.generated == seq2.generated


def test_family_presets_shapes():
    q = PRESETS["qwen2-7b"]
    assert (q.hidden, q.n_heads, q.n_kv_heads, q.ffn, q.vocab) == \
        (3584, 28, 4, 18944, 152064) and q.qkv_bias
    m = PRESETS["mistral-7b"]
    assert (m.hidden, m.n_heads, m.n_kv_heads, m.vocab,
            m.sliding_window) == (4096, 32, 8, 32000, 4096)


def test_swa_mixed_step_late_admission():
    """A prompt admitted while another sequence decodes (mixed
    prefill+decode step) must produce the same tokens as the same
    admission pattern run with full-causal masking disabled only where
    lengths stay inside the window — i.e. consistency of the windowed
    mixed path with the windowed sequential path."""
    prompt_a = list(range(10, 130))     # 120 > window 96
    prompt_b = [7, 8, 9]

    def run(stagger):
        eng = make_engine("tiny-swa")
        sa = eng.submit(prompt_a, GenParams(max_tokens=12))
        seq_a = eng.seqs[sa]
        if stagger:
            for _ in range(4):
                eng.step()
        sb = eng.submit(prompt_b, GenParams(max_tokens=12))
        seq_b = eng.seqs[sb]
        run_all(eng)
        return seq_a.generated, seq_b.generated

    a1, b1 = run(stagger=True)    # b admitted mid-decode -> mixed step
    a2, b2 = run(stagger=False)   # both admitted together
    assert len(a1) == len(a2) == 12
    assert a1 == a2, "windowed decode depends on admission timing"
    assert b1 == b2


def test_llama2_mha_presets_run():
    """Llama-2 shape class: MHA (GQA group 1), non-power-of-2 hidden for
    13B; exercised via a tiny MHA config variant."""
    import dataclasses
    cfg = dataclasses.replace(PRESETS["tiny"], name="tiny-mha",
                              n_kv_heads=PRESETS["tiny"].n_heads)
    model = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=3)
    kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=2,
                                max_ctx=cfg.max_ctx)
    eng = LlamaEngine(model, kv, max_batch=2)
    s = eng.submit([1, 2, 3], GenParams(max_tokens=5))
    seq = eng.seqs[s]
    run_all(eng)
    assert len(seq.generated) == 5
    p2, p13 = PRESETS["llama2-7b"], PRESETS["llama2-13b"]
    assert p2.n_heads == p2.n_kv_heads == 32 and p2.ffn == 11008
    assert p13.hidden == 5120 and p13.n_kv_heads == 40


def test_llama31_rope_scaling_table():
    """Llama-3.1 NTK scaling: long-wavelength frequencies divided by the
    factor, short ones untouched, monotone ramp between.  (Checked on a
    tiny config carrying the 3.1 scaling tuple — the 8B preset only
    differs in size.)"""
    import math
    import dataclasses
    assert PRESETS["llama3.1-8b"].rope_scaling == (8.0, 1.0, 4.0, 8192)
    cfg = dataclasses.replace(PRESETS["tiny"], rope_theta=500000.0,
                              max_ctx=256, rope_scaling=(8.0, 1.0, 4.0, 8192))
    scaled = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=1)
    plain_cfg = dataclasses.replace(cfg, rope_scaling=None)
    plain = LlamaModel(plain_cfg, device="cpu", dtype=torch.float32, seed=1)
    # reconstruct inv-freqs from the angle tables at position 1
    inv_s = torch.atan2(scaled.rope_sin[1], scaled.rope_cos[1])
    inv_p = torch.atan2(plain.rope_sin[1], plain.rope_cos[1])
    wavelen = 2 * math.pi / inv_p
    long_wl = wavelen > 8192          # scaled down by factor 8
    short_wl = wavelen < 8192 / 4     # untouched
    torch.testing.assert_close(inv_s[long_wl], inv_p[long_wl] / 8.0,
                               atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(inv_s[short_wl], inv_p[short_wl],
                               atol=1e-6, rtol=1e-6)
    mid = ~(long_wl | short_wl)
    assert ((inv_s[mid] <= inv_p[mid] + 1e-6)
            & (inv_s[mid] >= inv_p[mid] / 8.0 - 1e-6)).all()
