"""Engine end-to-end on CPU: continuous batching, prefill+decode, KV reuse.

Mirrors the determinism obligations of the reference's scheduler tests
(reference src/dispatcher.rs:942-984 test style: pure in-process, no GPU).
"""
import pytest
import torch

from ollamamq_amd.models import LlamaModel, PRESETS
from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams


def make_engine(preset="tiny-cpu", n_pages=128, max_slots=8):
    cfg = PRESETS[preset]
    model = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=7)
    kv = PagedKVCache.for_model(cfg, n_pages=n_pages, max_slots=max_slots,
                                max_ctx=cfg.max_ctx)
    return LlamaEngine(model, kv, max_batch=max_slots)


def run_all(eng, max_steps=500):
    done = []
    for _ in range(max_steps):
        done.extend(eng.step())
        if not eng.has_work():
            break
    assert not eng.has_work(), "engine did not drain"
    return done


def test_single_sequence_greedy():
    eng = make_engine()
    sid = eng.submit([1, 2, 3, 4, 5], GenParams(max_tokens=8))
    seq = eng.seqs[sid]
    run_all(eng)
    assert len(seq.generated) == 8
    assert seq.finish_reason == "length"
    assert all(0 <= t < eng.model.cfg.vocab for t in seq.generated)


def test_greedy_deterministic_across_batching():
    """The same prompt decoded alone and batched with others must emit the
    same greedy tokens (continuous batching must not change numerics
    beyond fp32 reduction order — on CPU reference it is exact)."""
    prompt = [5, 9, 2, 7]
    eng1 = make_engine()
    s1 = eng1.submit(prompt, GenParams(max_tokens=6))
    seq_alone = eng1.seqs[s1]
    run_all(eng1)

    eng2 = make_engine()
    sids = [eng2.submit([3, 1, 4, 1, 5], GenParams(max_tokens=6)),
            eng2.submit(prompt, GenParams(max_tokens=6)),
            eng2.submit([9, 8], GenParams(max_tokens=6))]
    seq_b = eng2.seqs[sids[1]]
    run_all(eng2)
    assert seq_alone.generated == seq_b.generated


def test_kv_pages_freed():
    eng = make_engine()
    free0 = eng.kv.free_page_count()
    for _ in range(3):
        eng.submit(list(range(20)), GenParams(max_tokens=4))
    run_all(eng)
    assert eng.kv.free_page_count() == free0


def test_many_users_interleaved():
    eng = make_engine(max_slots=8)
    sids = [eng.submit([i + 1, i + 2, i + 3], GenParams(max_tokens=5))
            for i in range(16)]  # more users than slots -> queued admission
    seqs = [eng.seqs[s] for s in sids]
    done = run_all(eng, max_steps=2000)
    assert len(done) == 16
    for s in seqs:
        assert len(s.generated) == 5


def test_cancellation():
    eng = make_engine()
    sid1 = eng.submit([1, 2, 3], GenParams(max_tokens=50))
    sid2 = eng.submit([4, 5, 6], GenParams(max_tokens=5))
    seq1 = eng.seqs[sid1]
    for _ in range(3):
        eng.step()
    eng.cancel(sid1)
    run_all(eng)
    assert seq1.finish_reason == "cancelled"
    # cancelled seq released its pages; engine drained fully
    assert eng.kv.free_page_count() == eng.kv.n_pages


def test_streaming_callback_order():
    eng = make_engine()
    events = []
    eng.submit([1, 2], GenParams(max_tokens=4),
               on_token=lambda t, done: events.append((t, done)))
    run_all(eng)
    assert len(events) == 5
    assert all(not d for _, d in events[:-1])
    assert events[-1] == (-1, True)


def test_stop_token():
    eng = make_engine()
    # find which token greedy decode emits first, then use it as stop
    sid = eng.submit([1, 2, 3], GenParams(max_tokens=3))
    seq = eng.seqs[sid]
    run_all(eng)
    stop = seq.generated[0]
    sid2 = eng.submit([1, 2, 3], GenParams(max_tokens=10, stop_token=stop))
    seq2 = eng.seqs[sid2]
    run_all(eng)
    assert seq2.finish_reason == "stop"
    assert seq2.generated[-1] == stop


def test_per_request_seed_reproducible():
    """options.seed (Ollama parity): a seeded stochastic request yields
    the same tokens regardless of what else shares the batch."""
    from ollamamq_amd.engine import GenParams as GP

    def run(extra):
        eng = make_engine()
        if extra:
            eng.submit([9, 9, 9], GP(max_tokens=6, temperature=1.0))
        sid = eng.submit([1, 2, 3], GP(max_tokens=6, temperature=0.9,
                                       seed=1234))
        seq = eng.seqs[sid]
        run_all(eng)
        return seq.generated

    a = run(extra=False)
    b = run(extra=True)
    assert a == b and len(a) == 6

    eng = make_engine()
    sid = eng.seqs[eng.submit([1, 2, 3], GP(max_tokens=6, temperature=0.9,
                                            seed=77))]
    run_all(eng)
    assert sid.generated != a  # different seed, different trajectory


def test_step_failure_recovery_worker_sequence():
    """The worker loop's recovery after a failed step (engine/worker.py
    run loop): cancel all in-flight, reap with done markers, engine stays
    serviceable for new requests."""
    eng = make_engine()
    calls = []
    orig = eng.model.forward

    def flaky(*a, **k):
        calls.append(1)
        if len(calls) == 3:
            raise RuntimeError("injected forward failure")
        return orig(*a, **k)

    eng.model.forward = flaky
    done_markers = []
    sid = eng.submit(list(range(1, 9)),
                     GenParams(max_tokens=50))
    eng.seqs[sid].on_token = lambda t, d: done_markers.append(d)
    with pytest.raises(RuntimeError):
        for _ in range(100):
            eng.step()
    # worker-style recovery
    for seq in list(eng.waiting) + list(eng.running):
        seq.cancelled = True
    eng.step()
    assert done_markers and done_markers[-1] is True
    assert not eng.has_work() and eng.seqs == {}
    assert len(eng.kv._free_slots) == eng.kv.max_slots

    # still serves new work afterwards
    eng.model.forward = orig
    sid2 = eng.submit([4, 5, 6], GenParams(max_tokens=3))
    seq2 = eng.seqs[sid2]
    run_all(eng)
    assert len(seq2.generated) == 3


def test_long_run_resource_stability():
    """Waves of sequences through a small engine: every slot and KV page
    must return to the pool and the registry must empty each wave."""
    eng = make_engine(max_slots=4)
    for wave in range(5):
        sids = [eng.submit([wave + 1, i + 1, 3],
                           GenParams(max_tokens=4 + (i % 3)))
                for i in range(8)]
        run_all(eng)
        assert eng.seqs == {}
        assert len(eng.kv._free_slots) == eng.kv.max_slots
        assert eng.kv.free_page_count() == eng.kv.n_pages
        assert all(s not in eng.seqs for s in sids)


def make_small_ctx_engine(max_ctx=64):
    cfg = PRESETS["tiny-cpu"]
    model = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=7)
    kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=4,
                                max_ctx=max_ctx)
    return LlamaEngine(model, kv, max_batch=2)


def test_context_exhaustion_finishes_gracefully():
    """A sequence reaching the context window must finish with reason
    "length", not fault the whole batch step (Ollama parity: generation
    is bounded by num_ctx)."""
    eng = make_small_ctx_engine()
    sid = eng.submit(list(range(40)), GenParams(max_tokens=1 << 30))
    seq = eng.seqs[sid]
    done = run_all(eng, max_steps=100)
    assert done and done[0].finish_reason == "length"
    assert seq.total_len == eng.kv.max_ctx


def test_oversized_prompt_truncated_front():
    """Prompts longer than the context window keep their TAIL (what
    conditions generation) instead of raising at prefill."""
    eng = make_small_ctx_engine()
    sid = eng.submit(list(range(200)), GenParams(max_tokens=4))
    seq = eng.seqs[sid]
    assert len(seq.prompt) == eng.kv.max_ctx - 1
    assert seq.prompt[-1] == 199          # tail survives
    done = run_all(eng, max_steps=100)
    assert done[0].finish_reason == "length"


def test_zero_token_request():
    """max_tokens=0 (Ollama num_predict: 0) emits NO content tokens —
    only the done marker — in both the pure-prefill and mixed-batch
    admission paths."""
    for mixed in (False, True):
        eng = make_small_ctx_engine()
        if mixed:
            eng.submit([9, 9, 9], GenParams(max_tokens=6))
            eng.step()                    # one sequence already decoding
        emitted = []
        sid = eng.submit([1, 2, 3], GenParams(max_tokens=0),
                         on_token=lambda t, d: emitted.append((t, d)))
        for _ in range(20):
            done = eng.step()
            if any(s.seq_id == sid for s in done):
                break
        assert emitted == [(-1, True)], (mixed, emitted)
        run_all(eng)
