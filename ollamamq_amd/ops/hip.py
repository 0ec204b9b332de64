"""ctypes bindings for the gfx950 HIP kernels (csrc/kernels/kernels.hip).

The extension is built IN-TREE (csrc/kernels/_kernels_gfx950.so) so the
.so travels with the repo snapshot to GPU boxes.  On a GPU box a missing
extension is a hard error — ops never fall back silently to eager PyTorch
(the GPU tests must exercise the native path).
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

from . import reference as ref
from .interface import AttnMeta

_LIB_PATH = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "csrc", "kernels", "_kernels_gfx950.so",
)

_lib = None
_load_err: Optional[str] = None


def _try_load():
    global _lib, _load_err
    if _lib is not None or _load_err is not None:
        return
    try:
        lib = ctypes.CDLL(_LIB_PATH, mode=ctypes.RTLD_GLOBAL)
    except OSError as e:
        _load_err = f"cannot load {_LIB_PATH}: {e}"
        return
    vp, i, f = ctypes.c_void_p, ctypes.c_int, ctypes.c_float
    lib.rmsnorm_residual_bf16.argtypes = [vp, vp, vp, vp, vp, i, i, f, vp]
    i64 = ctypes.c_int64
    lib.rope_bf16.argtypes = [vp, vp, vp, vp, vp, i, i, i, i, i64, i64, vp]
    lib.kv_append_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp,
                                   i, i, i, i, i, i64, vp]
    lib.paged_attn_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp, vp,
                                    i, i, i, i, i, i, f, i64, vp]
    lib.swiglu_bf16.argtypes = [vp, vp, i, i, vp]
    lib.argmax_bf16.argtypes = [vp, vp, i, i, vp, vp, i, vp]
    lib.decode_attn_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp, vp,
                                     i, i, i, i, i, f, i64, i, i, i, vp,
                                     i, vp]
    lib.skinny_gemm_bf16.argtypes = [vp, vp, vp, vp, i, i, i, i64, i, vp]
    lib.skinny_direct_bf16.argtypes = [vp, vp, vp, vp, i, i, i, i64, i, vp]
    lib.rope_append_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp, vp, vp,
                                     i, i, i, i, i, i, i64, i64, i64, vp]
    lib.prefill_attn_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp, vp,
                                      i, i, i, i, i, f, i64, i, vp]
    lib.embed_gather_bf16.argtypes = [vp, vp, vp, i, i, vp]
    lib.wstream_gemm_bf16.argtypes = [vp, vp, vp, vp, vp, i, i, i, i64, i,
                                      i, i, vp, i, f, f, vp, vp, i, vp]
    lib.wstream_pure_bf16.argtypes = [vp, vp, i, i, i, vp]
    lib.wstream_gu_bf16.argtypes = [vp, vp, vp, i, i, i, i64, i,
                                    vp, i, f, f, i, vp]
    lib.row_sumsq_bf16.argtypes = [vp, vp, i, i, vp]
    lib.fragify_sumsq_bf16.argtypes = [vp, vp, vp, i, i, vp]
    lib.sample_gumbel_bf16.argtypes = [vp, vp, vp, vp, vp, vp, i, i,
                                       vp, vp, i, vp]
    lib.decode_pure_bf16.argtypes = [vp, vp, vp, vp, vp, vp, i, i, i, i,
                                     i, i, vp]
    lib.wstream_qkv_rope_bf16.argtypes = [vp, vp, vp, vp, i, i, i, i64,
                                          vp, i, f, f, vp, vp, vp, vp,
                                          vp, vp, vp, i, i, i, i, i, vp]
    for fn in ("rmsnorm_residual_bf16", "rope_bf16", "kv_append_bf16",
               "paged_attn_bf16", "swiglu_bf16", "argmax_bf16",
               "decode_attn_bf16", "skinny_gemm_bf16",
               "skinny_direct_bf16", "prefill_attn_bf16",
               "rope_append_bf16", "embed_gather_bf16",
               "wstream_gemm_bf16"):
        getattr(lib, fn).restype = ctypes.c_int
    _lib = lib


def _check(err: int, op: str):
    if err != 0:
        raise RuntimeError(f"HIP kernel launch failed in {op}: hipError {err}")


def available() -> bool:
    _try_load()
    return _lib is not None


def require():
    _try_load()
    if _lib is None:
        raise RuntimeError(
            "ollamamq_amd HIP extension missing on a GPU box: "
            f"{_load_err}. Build it with `python -m ollamamq_amd.build` "
            "(hipcc --offload-arch=gfx950)."
        )


def _stream():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _p(t: Optional[torch.Tensor]):
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


# ---------------------------------------------------------------------------

def embedding(tokens, table):
    """out[t] = table[tokens[t]]: one gather kernel, int32 ids, no casts."""
    T = tokens.shape[0]
    V, H = table.shape
    if H % 8 or table.dtype != torch.bfloat16:
        return table.index_select(0, tokens.long())
    tok32 = tokens if tokens.dtype == torch.int32 else tokens.int()
    out = torch.empty((T, H), dtype=table.dtype, device=table.device)
    _check(_lib.embed_gather_bf16(_p(out), _p(table), _p(tok32), T, H,
                                  _stream()), "embed_gather")
    return out


def rmsnorm_residual(x, residual, weight, eps):
    T, H = x.shape
    y = torch.empty_like(x)
    res_out = torch.empty_like(x)
    _check(_lib.rmsnorm_residual_bf16(_p(y), _p(res_out), _p(x),
                                      _p(residual), _p(weight), T, H,
                                      float(eps), _stream()), "rmsnorm")
    return y, res_out


def _row_stride(t, D):
    # tolerate fused-QKV views: [T, H, D] with stride (S, D, 1)
    assert t.stride(2) == 1 and t.stride(1) == D, "head dim must be packed"
    return t.stride(0)


def rope(q, k, positions, cos, sin):
    T, Hq, D = q.shape
    Hk = k.shape[1]
    assert D == 128, "rope kernel specialized for head_dim=128"
    pos32 = positions if positions.dtype == torch.int32 else positions.int()
    _check(_lib.rope_bf16(_p(q), _p(k), _p(pos32), _p(cos), _p(sin),
                          T, Hq, Hk, D, _row_stride(q, D), _row_stride(k, D),
                          _stream()), "rope")


def kv_append(cache, layer, k, v, slot_ids, positions):
    T, KVH, D = k.shape
    kp, vp = _layer_ptrs(cache, layer)
    pos32 = positions if positions.dtype == torch.int32 else positions.int()
    slot32 = slot_ids if slot_ids.dtype == torch.int32 else slot_ids.int()
    assert _row_stride(k, D) == _row_stride(v, D)
    _check(_lib.kv_append_bf16(kp, vp, _p(k), _p(v), _p(slot32), _p(pos32),
                               _p(cache.page_table), T, KVH, D,
                               cache.page_size, cache.page_table.shape[1],
                               _row_stride(k, D), _stream()), "kv_append")


def rope_append(cache, layer, q, k, v, positions, slot_ids, cos, sin):
    """Fused RoPE (q,k in place) + paged append of rotated k and v."""
    T, Hq, D = q.shape
    KVH = k.shape[1]
    assert D == 128
    kp, vp = _layer_ptrs(cache, layer)
    pos32 = positions if positions.dtype == torch.int32 else positions.int()
    slot32 = slot_ids if slot_ids.dtype == torch.int32 else slot_ids.int()
    _check(_lib.rope_append_bf16(
        _p(q), _p(k), _p(v), kp, vp, _p(pos32), _p(slot32),
        _p(cache.page_table), _p(cos), _p(sin), T, Hq, KVH, D,
        cache.page_size, cache.page_table.shape[1],
        _row_stride(q, D), _row_stride(k, D), _row_stride(v, D),
        _stream()), "rope_append")


def _layer_ptrs(cache, layer):
    stride = cache.k_pool.stride(0) * cache.k_pool.element_size()
    return (ctypes.c_void_p(cache.k_pool.data_ptr() + layer * stride),
            ctypes.c_void_p(cache.v_pool.data_ptr() + layer * stride))


PREFILL_QT = 32


def _attn_plan(cache, meta, qt):
    """Device tile arrays for the attention launch, cached on the meta
    object (one build per forward, shared by every layer)."""
    key = f"_plan_{qt}"
    plan = getattr(meta, key, None)
    if plan is not None:
        return plan
    dev = meta.slot_ids.device
    if qt == 1:
        n = meta.slot_ids.shape[0]
        tile_slot = meta.slot_ids.int()
        tile_q0 = torch.arange(n, dtype=torch.int32, device=dev)
        tile_pos0 = (meta.seq_lens - 1).int()
        tile_rows = torch.ones(n, dtype=torch.int32, device=dev)
    else:
        slots = meta.slot_ids.tolist()
        lens = meta.seq_lens.tolist()
        cu = meta.cu_q.tolist()
        ts, tq, tp, tr = [], [], [], []
        for i, slot in enumerate(slots):
            qlen = cu[i + 1] - cu[i]
            start_pos = lens[i] - qlen
            for t0 in range(0, qlen, qt):
                rows = min(qt, qlen - t0)
                ts.append(slot)
                tq.append(cu[i] + t0)
                tp.append(start_pos + t0)
                tr.append(rows)
        tile_slot = torch.tensor(ts, dtype=torch.int32, device=dev)
        tile_q0 = torch.tensor(tq, dtype=torch.int32, device=dev)
        tile_pos0 = torch.tensor(tp, dtype=torch.int32, device=dev)
        tile_rows = torch.tensor(tr, dtype=torch.int32, device=dev)
    plan = (tile_slot, tile_q0, tile_pos0, tile_rows)
    setattr(meta, key, plan)
    return plan


def _attention(q, cache, layer, meta, qt):
    T, Hq, D = q.shape
    assert D == 128
    out = torch.empty_like(q)
    kp, vp = _layer_ptrs(cache, layer)
    tile_slot, tile_q0, tile_pos0, tile_rows = _attn_plan(cache, meta, qt)
    n_tiles = tile_slot.shape[0]
    if n_tiles == 0:
        return out
    _check(_lib.paged_attn_bf16(
        _p(out), _p(q), kp, vp, _p(cache.page_table),
        _p(tile_slot), _p(tile_q0), _p(tile_pos0), _p(tile_rows),
        n_tiles, qt, Hq, cache.n_kv_heads, cache.page_size,
        cache.page_table.shape[1], 1.0 / (D ** 0.5), _row_stride(q, D),
        _stream()), "paged_attn")
    return out


_scratch = {}


def _decode_scratch(S, Hq, split, dev):
    key = (S, Hq, split, str(dev))
    t = _scratch.get(key)
    if t is None:
        t = (torch.empty(S * Hq * split * 128, dtype=torch.float32,
                         device=dev),
             torch.empty(S * Hq * split * 2, dtype=torch.float32,
                         device=dev))
        _scratch[key] = t
    return t


def _combine_sem(S, KVH, dev):
    """Ticket counters for the in-launch split-K combine: zero-initialized
    once; the last-arriving block resets its counter, so the buffer is
    always all-zero between launches (guide G16 counter recipe)."""
    key = ("sem", S, KVH, str(dev))
    t = _scratch.get(key)
    if t is None:
        t = torch.zeros(S * KVH, dtype=torch.int32, device=dev)
        _scratch[key] = t
    return t


# staged KV chunk length (tokens) per pipeline stage; 32/64 compiled
# (the kernel scores one key per wave64 lane, so 64 is the ceiling).
# 64 wins for GQA groups >= 2 (sweep in NOTES.md); G=1 (MHA, Llama-2)
# takes 32 so the register-prefetch pipeline fits (NPF=8) — measured
# 142.9 -> 101.4 us at ctx 512 (see NOTES.md)
_CHUNK_ENV = os.environ.get("OLLAMAMQ_DECODE_CHUNK")


def _decode_chunk(G):
    if _CHUNK_ENV:
        return min(64, int(_CHUNK_ENV))
    return 32 if G == 1 else 64


def attention_decode(q, cache, layer, meta, fragout=False):
    """Flash-decoding: KV-split partials + exact online-softmax combine.

    The split factor targets ≥1024 workgroups so the memory-bound KV sweep
    fills the 256-CU chip (a bare (seq, kv-head) grid at batch 32 is 1
    workgroup/CU and runs at ~10% of HBM bandwidth).
    fragout: emit the output in the fused-chain 32-row frag layout (the
    o GEMM streams it linearly like packed weights)."""
    S, Hq, D = q.shape
    assert D == 128
    out = torch.empty((((S + 31) // 32) * 32 * Hq * D,)
                      if fragout else (S, Hq, D),
                      dtype=q.dtype, device=q.device)
    kp, vp = _layer_ptrs(cache, layer)
    kvh = cache.n_kv_heads
    # target ~1024 blocks; under graph capture max_kv is the pool's
    # max_ctx, so an unconditional max-split would freeze in dozens of
    # empty segments per sequence (measured: 5.55k -> 4.3k tok/s)
    want = max(1, 1024 // max(1, S * kvh))
    env = os.environ.get("OLLAMAMQ_DECODE_SPLIT")
    if env:                       # tuning override (tools/perf_decode.py)
        want = int(env)
    max_seg = max(1, (meta.max_kv + 63) // 64)
    split = int(min(want, max_seg, 32))
    o_part, ml_part = (None, None)
    op = mp = ctypes.c_void_p(0)
    sem = ctypes.c_void_p(0)
    if split > 1:
        o_part, ml_part = _decode_scratch(S, Hq, split, q.device)
        op, mp = _p(o_part), _p(ml_part)
        # in-launch combine measured WORSE (41.8 vs 29.3 us @ctx512 B32;
        # bench 5131 vs 5580): all S*split*KVH blocks pay the agent-scope
        # release fence (buffer_wbl2, ~1.7-6.5 us at 4 blocks/CU) to save
        # one 5 us combine launch. Kept for re-evaluation on shapes with
        # fewer, longer blocks.
        if os.environ.get("OLLAMAMQ_FUSED_COMBINE") == "1":
            sem = _p(_combine_sem(S, kvh, q.device))
    slot32 = meta.slot_ids.int() if meta.slot_ids.dtype != torch.int32 \
        else meta.slot_ids
    len32 = meta.seq_lens.int() if meta.seq_lens.dtype != torch.int32 \
        else meta.seq_lens
    _check(_lib.decode_attn_bf16(
        _p(out), op, mp, _p(q), kp, vp, _p(cache.page_table),
        _p(slot32), _p(len32), S, Hq, kvh, cache.page_size,
        cache.page_table.shape[1], 1.0 / (D ** 0.5), _row_stride(q, D),
        split, meta.window, _decode_chunk(Hq // kvh), sem,
        1 if fragout else 0, _stream()),
        "decode_attn")
    return out


def attention_mixed(q, cache, layer, meta):
    """Mixed batch: decode kernel over the first n_decode rows, MFMA
    prefill kernel over the remaining chunk rows — both write into one
    output tensor (sub-views share storage)."""
    T, Hq, D = q.shape
    nd = meta.n_decode
    out = torch.empty((T, Hq, D), dtype=q.dtype, device=q.device)
    # ---- decode part ----
    sub = getattr(meta, "_mixed_dec", None)
    if sub is None:
        sub = AttnMeta(
            mode="decode", slot_ids=meta.slot_ids[:nd],
            seq_lens=meta.seq_lens[:nd],
            cu_q=meta.cu_q[:nd + 1], logits_idx=None, max_q=1,
            max_kv=meta.max_kv, window=meta.window)
        meta._mixed_dec = sub
    if nd:
        out[:nd] = attention_decode(q[:nd], cache, layer, sub)
    # ---- prefill part (tiles over sequences nd..) ----
    plan = getattr(meta, "_mixed_plan", None)
    if plan is None:
        slots = meta.slot_ids[nd:].tolist()
        lens = meta.seq_lens[nd:].tolist()
        cu = meta.cu_q[nd:].tolist()
        dev = q.device
        ts, tq, tp, tr = [], [], [], []
        for i, slot in enumerate(slots):
            qlen = cu[i + 1] - cu[i]
            start_pos = lens[i] - qlen
            for t0 in range(0, qlen, PREFILL_QT):
                ts.append(slot)
                tq.append(cu[i] + t0)
                tp.append(start_pos + t0)
                tr.append(min(PREFILL_QT, qlen - t0))
        plan = (torch.tensor(ts, dtype=torch.int32, device=dev),
                torch.tensor(tq, dtype=torch.int32, device=dev),
                torch.tensor(tp, dtype=torch.int32, device=dev),
                torch.tensor(tr, dtype=torch.int32, device=dev))
        meta._mixed_plan = plan
    tile_slot, tile_q0, tile_pos0, tile_rows = plan
    n_tiles = tile_slot.shape[0]
    if n_tiles:
        kp, vp = _layer_ptrs(cache, layer)
        _check(_lib.prefill_attn_bf16(
            _p(out), _p(q), kp, vp, _p(cache.page_table),
            _p(tile_slot), _p(tile_q0), _p(tile_pos0), _p(tile_rows),
            n_tiles, Hq, cache.n_kv_heads, cache.page_size,
            cache.page_table.shape[1], 1.0 / (D ** 0.5),
            _row_stride(q, D), meta.window, _stream()), "prefill_attn")
    return out


def attention_prefill(q, cache, layer, meta):
    """MFMA flash prefill (prefill_attn.hip); set OLLAMAMQ_VALU_PREFILL=1
    to fall back to the VALU paged_attn path (A/B + debugging)."""
    if os.environ.get("OLLAMAMQ_VALU_PREFILL") == "1" \
            and not meta.window:   # VALU fallback predates sliding window
        return _attention(q, cache, layer, meta, 16)
    T, Hq, D = q.shape
    assert D == 128
    out = torch.empty((T, Hq, D), dtype=q.dtype, device=q.device)
    kp, vp = _layer_ptrs(cache, layer)
    tile_slot, tile_q0, tile_pos0, tile_rows = _attn_plan(cache, meta,
                                                          PREFILL_QT)
    n_tiles = tile_slot.shape[0]
    if n_tiles == 0:
        return out
    _check(_lib.prefill_attn_bf16(
        _p(out), _p(q), kp, vp, _p(cache.page_table),
        _p(tile_slot), _p(tile_q0), _p(tile_pos0), _p(tile_rows),
        n_tiles, Hq, cache.n_kv_heads, cache.page_size,
        cache.page_table.shape[1], 1.0 / (D ** 0.5), _row_stride(q, D),
        meta.window, _stream()), "prefill_attn")
    return out


# default 0 = library GEMMs everywhere: the fragment-direct kernel beats
# hipBLASLt on qkv/o in MICROBENCHES (L3-warm weights re-read 100x) but
# loses ~0.3 ms/step in the real serving loop where weights stream cold
# from HBM — honest A/B in bench.py decided this (5473 vs 5221 tok/s).
# The hand-written variants stay available via these env gates.
SKINNY_MAX_N = int(os.environ.get("OLLAMAMQ_SKINNY_MAX_N", "0"))
SKINNY_MAX_K = int(os.environ.get("OLLAMAMQ_SKINNY_MAX_K", "0"))

_gemm_scratch = {}


def _skinny_ksplit(N, K):
    """grid k-split: target >=768 blocks (~3/CU) so independent blocks
    hide each other's staging latency; kseg must stay a multiple of 256
    (even chunk count for the 2-deep pipeline)."""
    blocks = N // 32
    ks = 1
    while blocks * ks < 768 and ks < 8 and K % (256 * 2 * ks) == 0:
        ks *= 2
    return ks


def pack_weight(w):
    """Pre-pack a [N, K] bf16 weight into MFMA fragment order for the
    weight-streaming decode GEMM (wstream_gemm.hip): element W[n][k] with
    n = t*32+r, k = b*64 + j*16 + h*8 + e goes to 16-byte unit
    ((t*NB + b)*4 + j)*64 + (h*32 + r).  The kernel's loads then walk the
    buffer LINEARLY (full 128 B line per instruction, nt-tagged) — the fix
    for the 4x cold-line over-fetch of fragment-direct loads (NOTES r01).
    Returns None when the shape doesn't qualify."""
    N, K = w.shape
    if N % 32 or K % 64 or w.dtype != torch.bfloat16 \
            or not w.is_contiguous():
        return None
    p = w.reshape(N // 32, 32, K // 64, 4, 2, 8) \
         .permute(0, 2, 3, 4, 1, 5).contiguous()
    return p.view(-1)


def pack_weight_gu(w):
    """GU-interleaved pack of a fused [2F, K] gate_up weight: tile t holds
    gate rows f=t*16..t*16+16 then the matching 16 up rows, so the fused
    kernel's 32-column output tile pairs each gate with its up and the
    epilogue can apply SwiGLU in-register (wstream_gemm.hip GU=1)."""
    N, K = w.shape
    F = N // 2
    if N % 32 or F % 16 or K % 64 or w.dtype != torch.bfloat16 \
            or not w.is_contiguous():
        return None
    p = w.reshape(2, F // 16, 16, K // 64, 4, 2, 8) \
         .permute(1, 3, 4, 5, 0, 2, 6).contiguous()
    return p.view(-1)


def linear_gu(x, packed, N, rstd=None, rstd_nt=0, inv_h=0.0, eps=0.0,
              K=None, yfrag=0, M_frag=32):
    """act = swiglu(x @ Wgu^T) fused; N = 2F total weight rows.
    With rstd: x is the raw residual and the epilogue applies the
    rmsnorm scale before SwiGLU (norm weight folded into the pack).
    K given => x is a frag-layout buffer (xlds=2); yfrag => the
    activation is emitted in frag layout for the down GEMM."""
    F = N // 2
    if K is None:
        M, K = x.shape
        xs = x.stride(0)
        xlds = (1 if N * K * 2 > (64 << 20) else 0) if M <= 32 else 0
    else:
        M, xs, xlds = M_frag, 0, 2
    act = torch.empty((M, F) if not yfrag
                      else (((M + 31) // 32) * 32 * F,),
                      dtype=x.dtype, device=x.device)
    _check(_lib.wstream_gu_bf16(
        _p(act), _p(x), _p(packed), M, N, K, xs, xlds,
        _p(rstd), rstd_nt, float(inv_h), float(eps), yfrag,
        _stream()), "wstream_gu")
    return act


def _rope_pair_order(heads, d=128):
    """Output-row order pairing each head's RoPE halves: 16 lo dims then
    their 16 hi (d+64) dims, so every 32-row tile holds full rotation
    pairs for the fused qkv epilogue."""
    idx = []
    for h in range(heads):
        base = h * d
        for d0 in (0, 16, 32, 48):
            idx.extend(range(base + d0, base + d0 + 16))
            idx.extend(range(base + 64 + d0, base + 64 + d0 + 16))
    return idx


def qkv_rope_order(nl, nkl, d=128):
    """Row permutation for the whole fused qkv weight [q|k|v]."""
    q = _rope_pair_order(nl, d)
    k = [nl * d + i for i in _rope_pair_order(nkl, d)]
    v = [(nl + nkl) * d + i for i in range(nkl * d)]
    return q + k + v


def pack_weight_qkv_rope(w, nl, nkl):
    """Pair-ordered pack of the fused qkv weight for the RoPE epilogue."""
    order = torch.tensor(qkv_rope_order(nl, nkl), device=w.device)
    return pack_weight(w.index_select(0, order).contiguous())


def linear_qkv_rope(x, packed, N, bias_rp, cache, layer, positions,
                    slots, cos, sin, nl, nkl, rstd, rstd_nt, inv_h, eps,
                    K=None, M_real=32):
    """Fused rmsnorm -> qkv GEMM -> RoPE -> paged KV append: q returns
    rotated in the output's standard layout; k/v land in the pool."""
    if K is None:
        M, K = x.shape
        xs, xf = x.stride(0), 0
    else:
        M, xs, xf = M_real, 0, 1
    y = torch.empty((M, N), dtype=x.dtype, device=x.device)
    kp, vp = _layer_ptrs(cache, layer)
    pos32 = positions if positions.dtype == torch.int32 else positions.int()
    slot32 = slots if slots.dtype == torch.int32 else slots.int()
    _check(_lib.wstream_qkv_rope_bf16(
        _p(y), _p(x), _p(packed), _p(bias_rp), M, N, K, xs,
        _p(rstd), rstd_nt, float(inv_h), float(eps),
        _p(cos), _p(sin), _p(pos32), _p(slot32), _p(cache.page_table),
        kp, vp, nl, nkl, cache.page_size, cache.page_table.shape[1],
        xf, _stream()), "wstream_qkv_rope")
    return y


USE_WSTREAM = os.environ.get("OLLAMAMQ_NO_WSTREAM") != "1"


def _wstream_ksplit(N, K):
    """Measured rules (profiles/r02_gemm_sweep.md + in-graph A/B):
    * small weights (< 64 MB: qkv/o) are ramp/latency-floor-bound —
      fewer blocks with longer streams win (ks=1 beat every split);
    * big weights need >= 256 blocks to reach the full-chip stream rate
      (down at 128 blocks ran at half the chip: 41 vs 26 us);
    * uneven wave ranges (NB % 8ks != 0) cost ~10% tail imbalance
      (down ks3 measured worse than both ks2 and ks4);
    * very long (>16 iters) or very short (<4) per-wave streams lose."""
    tiles = N // 32
    nb = K // 64
    big = N * K * 2 > (64 << 20)
    best, best_cost = 1, 1 << 30
    for ks in range(1, 9):
        iters = nb // (ks * 8)
        if iters < 1:
            break
        cost = ks
        if nb % (ks * 8):
            cost += 16
        if big and tiles * ks < 256:
            cost += 24
        if iters < 4:
            cost += (4 - iters) * 8
        if iters > 16:
            cost += (iters - 16) * 2
        if cost < best_cost:
            best, best_cost = ks, cost
    return best


_WS_DEPTH = int(os.environ.get("OLLAMAMQ_WS_DEPTH", "1"))
_WS_XLDS = int(os.environ.get("OLLAMAMQ_WS_XLDS", "1"))
_WS_KS = os.environ.get("OLLAMAMQ_WS_KS")   # sweep override
# N above which the library GEMM wins (measured r02 sweep,
# profiles/r02_gemm_sweep.md: wstream beats hipBLASLt on the
# ramp-dominated small-N decode shapes qkv/o/down; on gate_up/logits the
# lib's big-grid kernels are closer to the stream ceiling than our
# x-load overhead allows — revisit after the x staging rework)
_WS_MAX_N = int(os.environ.get("OLLAMAMQ_WS_MAX_N", "16384"))


def linear_packed(x, packed, bias, N, ks=None, depth=None, xlds=None,
                  rstd=None, rstd_nt=0, inv_h=0.0, eps=0.0,
                  res=None, sq_out=None, y=None, yfrag=0, K=None,
                  M_frag=32):
    """y = x @ W^T via the weight-streaming kernel over pre-packed W.

    Fused-chain extras (decode, M<=32): `rstd`/`rstd_nt`/`inv_h`/`eps`
    apply rmsnorm scaling of the raw-residual input x (norm weight must
    be folded into the pack); `res` adds the residual stream into the
    output (pass y=res for the in-place residual update) and `sq_out`
    emits per-tile sum-of-squares partials for the next GEMM."""
    if K is None:
        M, K = x.shape
        xs = x.stride(0)
    else:                       # frag-layout x: flat buffer, xlds==2
        M = M_frag
        xs = 0
    if y is None:
        y = torch.empty((((M + 31) // 32) * 32 * N,) if yfrag
                        else (M, N), dtype=x.dtype, device=x.device)
    if ks is None:
        per_k = os.environ.get(f"OLLAMAMQ_WS_KS_{K}")   # per-shape tuning
        if per_k:
            ks = int(per_k)
        else:
            ks = int(_WS_KS) if _WS_KS else _wstream_ksplit(N, K)
    if rstd is not None:
        ks = 1            # rstd fusion requires ks == 1
    part = ctypes.c_void_p(0)
    if ks > 1:
        key = ("ws", M, N, ks, str(x.device))
        t = _gemm_scratch.get(key)
        if t is None:
            t = torch.empty(ks * M * N, dtype=torch.float32,
                            device=x.device)
            _gemm_scratch[key] = t
        part = _p(t)
    if xlds is None:
        # measured (profiles/r02_gemm_sweep.md): LDS x-staging pays on
        # long streams (down/gate_up/logits), the direct fragment load
        # wins on the short ramp-bound ones (qkv/o)
        xlds = (_WS_XLDS if M <= 32 and N * K * 2 > (64 << 20) else 0)
    _check(_lib.wstream_gemm_bf16(
        _p(y), part, _p(x), _p(packed), _p(bias), M, N, K, xs,
        ks, depth if depth is not None else _WS_DEPTH, xlds,
        _p(rstd), rstd_nt, float(inv_h), float(eps), _p(res), _p(sq_out),
        yfrag, _stream()), "wstream_gemm")
    return y


def fragify_sumsq(x, xf=None, sq=None):
    """Standard [M, H] -> 32-row frag layout + per-row sum of squares
    (seeds the fused decode chain after the embedding gather)."""
    M, H = x.shape
    if xf is None:
        xf = torch.empty(((M + 31) // 32) * 32 * H, dtype=x.dtype,
                         device=x.device)
    if sq is None:
        sq = torch.empty(M, dtype=torch.float32, device=x.device)
    _check(_lib.fragify_sumsq_bf16(_p(xf), _p(sq), _p(x), M, H,
                                   _stream()), "fragify_sumsq")
    return xf, sq


def row_sumsq(x, out=None):
    """sq[m] = sum(x[m]**2) fp32 — seeds the fused-rmsnorm chain."""
    M, H = x.shape
    if out is None:
        out = torch.empty(M, dtype=torch.float32, device=x.device)
    _check(_lib.row_sumsq_bf16(_p(out), _p(x), M, H, _stream()),
           "row_sumsq")
    return out


def wstream_pure(packed, N, K, ks):
    """Bandwidth diagnostic: the GEMM's weight stream alone."""
    sink = torch.zeros(1, dtype=torch.float32, device=packed.device)
    _check(_lib.wstream_pure_bf16(_p(sink), _p(packed), N, K, ks,
                                  _stream()), "wstream_pure")
    return sink


def linear(x, weight, bias=None, packed=None):
    M, K = x.shape
    N = weight.shape[0]
    if packed is not None and USE_WSTREAM and M <= 64 \
            and (N <= _WS_MAX_N or N >= 65536) \
            and x.dtype == torch.bfloat16 and x.stride(1) == 1:
        return linear_packed(x, packed, bias, N)
    # gated to the shapes where the hand-written kernel beats hipBLASLt
    # (measured tools/perf_gemm.py); widen via env as the kernel improves
    if bias is None and M <= 32 and K % 256 == 0 and N % 32 == 0 \
            and N <= SKINNY_MAX_N and K <= SKINNY_MAX_K \
            and x.dtype == torch.bfloat16 and weight.stride(1) == 1:
        y = torch.empty((M, N), dtype=x.dtype, device=x.device)
        assert x.stride(1) == 1
        direct = os.environ.get("OLLAMAMQ_SKINNY_STAGED") != "1"
        if direct:
            # per-wave k extent must be a multiple of 64
            ks = 1
            blocks = N // 32
            while blocks * ks < 768 and ks < 8 and K % (ks * 2 * 512) == 0:
                ks *= 2
            if K % (ks * 512) != 0:
                return torch.nn.functional.linear(x, weight)
        else:
            ks = _skinny_ksplit(N, K)
        part = ctypes.c_void_p(0)
        if ks > 1:
            key = (M, N, ks, str(x.device))
            t = _gemm_scratch.get(key)
            if t is None:
                t = torch.empty(ks * M * N, dtype=torch.float32,
                                device=x.device)
                _gemm_scratch[key] = t
            part = _p(t)
        fn = _lib.skinny_direct_bf16 if direct else _lib.skinny_gemm_bf16
        _check(fn(_p(y), part, _p(x), _p(weight),
                  M, N, K, x.stride(0), ks, _stream()), "skinny_gemm")
        return y
    return torch.nn.functional.linear(x, weight, bias)


def swiglu(gate_up):
    T, F2 = gate_up.shape
    F = F2 // 2
    out = torch.empty((T, F), dtype=gate_up.dtype, device=gate_up.device)
    _check(_lib.swiglu_bf16(_p(out), _p(gate_up), T, F, _stream()),
           "swiglu")
    return out


def sample_gumbel(logits, temps, seeds, ctrs, noise=None, out=None):
    """Exact temperature sampling (Gumbel-max): token ~ softmax(l/T) per
    row; rows with temps<=0 are greedy argmax.  Counter-based noise from
    (seeds[row], ctrs[row], v) — graph-replay-safe and reproducible per
    request.  noise: optional [B,V] uniform(0,1) override (tests)."""
    B, V = logits.shape
    l = logits if logits.dtype == torch.bfloat16 else logits.bfloat16()
    if out is None:
        out = torch.empty(B, dtype=torch.int32, device=logits.device)
    sp = min(32, max(1, 768 // max(1, B)))
    pb = pi = ctypes.c_void_p(0)
    if sp > 1:
        key = ("gmb", B, sp, str(logits.device))
        t = _scratch.get(key)
        if t is None:
            t = (torch.empty(B * sp, dtype=torch.float32,
                             device=logits.device),
                 torch.empty(B * sp, dtype=torch.int32,
                             device=logits.device))
            _scratch[key] = t
        pb, pi = _p(t[0]), _p(t[1])
    _check(_lib.sample_gumbel_bf16(
        _p(out), _p(l), _p(temps), _p(seeds), _p(ctrs), _p(noise),
        B, V, pb, pi, sp, _stream()), "sample_gumbel")
    return out


def sample(logits, temperature, top_k, top_p, generator=None):
    if torch.is_tensor(temperature):
        all_greedy = bool((temperature <= 0).all())
    else:
        all_greedy = temperature <= 0
    if all_greedy:
        B, V = logits.shape
        out = torch.empty(B, dtype=torch.int32, device=logits.device)
        l = logits if logits.dtype == torch.bfloat16 else logits.bfloat16()
        # split-V: B blocks alone underfill the chip at decode batch sizes
        sp = min(32, max(1, 768 // max(1, B)))
        pb = pi = ctypes.c_void_p(0)
        if sp > 1:
            key = ("amax", B, sp, str(logits.device))
            t = _scratch.get(key)
            if t is None:
                t = (torch.empty(B * sp, dtype=torch.float32,
                                 device=logits.device),
                     torch.empty(B * sp, dtype=torch.int32,
                                 device=logits.device))
                _scratch[key] = t
            pb, pi = _p(t[0]), _p(t[1])
        _check(_lib.argmax_bf16(_p(out), _p(l), B, V, pb, pi, sp,
                                _stream()), "argmax")
        return out.long()
    # stochastic host path.  When every stochastic row carries
    # 0 < top_k <= 256 the sampling is EXACT over the top-256 candidates
    # (its nucleus is a subset of its top-k; probabilities use a
    # full-vocabulary logsumexp, so the top-p mass cut is the true one)
    # and costs a topk + tiny multinomial instead of the full-vocab sort
    # (~0.5 ms/step at V=128256).  Anything else falls back to the
    # fp32 reference (exact, sort-based).
    B, V = logits.shape
    dev = logits.device
    if not torch.is_tensor(temperature):
        temperature = torch.full((B,), float(temperature), device=dev)
    tk = top_k if torch.is_tensor(top_k) else torch.full(
        (B,), int(top_k or 0), dtype=torch.long, device=dev)
    tp = top_p if torch.is_tensor(top_p) else torch.full(
        (B,), float(top_p if top_p else 1.0), device=dev)
    tk = tk.long()
    stoch = temperature > 0
    C = min(256, V)
    capped_ok = bool(((tk > 0) & (tk <= C) | ~stoch).all())
    if not capped_ok:
        return ref.sample(logits, temperature, tk, tp, generator)
    l = logits.float()
    t = torch.where(stoch, temperature, torch.ones_like(temperature))
    lt = l / t.unsqueeze(1)
    v, idx = torch.topk(lt, C, dim=-1)
    lse = torch.logsumexp(lt, dim=-1, keepdim=True)
    p = torch.exp(v - lse)                      # exact full-vocab probs
    ar = torch.arange(C, device=dev)
    keep = ar.unsqueeze(0) < tk.clamp(min=1, max=C).unsqueeze(1)
    cum = p.cumsum(dim=-1)
    keep &= (cum - p) < tp.unsqueeze(1)
    keep[:, 0] = True
    pm = p * keep
    pick = torch.multinomial(pm / pm.sum(-1, keepdim=True), 1,
                             generator=generator).squeeze(1)
    out_s = idx.gather(1, pick.unsqueeze(1)).squeeze(1)
    if bool(stoch.all()):
        return out_s
    # greedy rows override with the true argmax
    return torch.where(stoch, out_s, l.argmax(-1))


def decode_pure(cache, layer, meta, split, chunk=64):
    """Staging-only diagnostic for decode attention (see decode_attn.hip
    k_decode_pure)."""
    sink = torch.zeros(1, dtype=torch.float32, device=cache.k_pool.device)
    kp, vp = _layer_ptrs(cache, layer)
    S = meta.slot_ids.shape[0]
    _check(_lib.decode_pure_bf16(
        _p(sink), kp, vp, _p(cache.page_table), _p(meta.slot_ids.int()),
        _p(meta.seq_lens.int()), S, cache.n_kv_heads, cache.page_size,
        cache.page_table.shape[1], split, chunk, _stream()),
        "decode_pure")
    return sink
