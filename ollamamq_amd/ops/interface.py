"""Op dispatch: hand-written gfx950 HIP kernels on GPU, fp32 torch on CPU.

Policy (task requirement): on a GPU box the HIP extension MUST be present —
ops fail loudly rather than falling back to eager PyTorch, so a passing GPU
test is evidence the native path ran.  Set OLLAMAMQ_FORCE_REF=1 to force the
reference path explicitly (used only by numerics tests to produce oracles).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch

from . import reference as ref


def _use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if os.environ.get("OLLAMAMQ_FORCE_REF") == "1":
        return False
    from . import hip  # deferred: imports/builds the extension
    hip.require()      # raises if the extension is missing on a GPU box
    return True


@dataclass
class AttnMeta:
    """Per-forward attention metadata (flat varlen layout).

    mode: "prefill" (multi-token queries), "decode" (1 token/seq) or
    "mixed" (the first n_decode sequences are single-token decodes, the
    rest are prefill chunks — one forward shares the weight pass).
    slot_ids : [S] int32  kv-cache slot of each sequence in batch order
    seq_lens : [S] int32  total KV length per slot AFTER the kv_append
    cu_q     : [S+1] int32 exclusive prefix sum of per-seq query counts
    logits_idx: [S] int64 flat indices of each sequence's last token
                (None = all tokens get logits)
    """
    mode: str
    slot_ids: torch.Tensor
    seq_lens: torch.Tensor
    cu_q: torch.Tensor
    logits_idx: Optional[torch.Tensor]
    max_q: int
    max_kv: int
    n_decode: int = 0
    # sliding-window attention (Mistral-style): query at position p attends
    # keys [max(0, p - window + 1), p]; 0 = full causal
    window: int = 0


def embedding(tokens: torch.Tensor, table: torch.Tensor) -> torch.Tensor:
    if _use_hip(table):
        from . import hip
        return hip.embedding(tokens, table)
    return table.index_select(0, tokens.long())


def rmsnorm_residual(x, residual, weight, eps):
    if _use_hip(x):
        from . import hip
        return hip.rmsnorm_residual(x, residual, weight, eps)
    return ref.rmsnorm_residual(x, residual, weight, eps)


def rope(q, k, positions, cos, sin):
    if _use_hip(q):
        from . import hip
        return hip.rope(q, k, positions, cos, sin)
    return ref.rope(q, k, positions, cos, sin)


def rope_append(cache, layer, q, k, v, positions, slot_ids, cos, sin):
    """Fused RoPE + KV append (one pass); falls back to the two reference
    ops on CPU."""
    if _use_hip(q):
        from . import hip
        return hip.rope_append(cache, layer, q, k, v, positions, slot_ids,
                               cos, sin)
    ref.rope(q, k, positions, cos, sin)
    ref.kv_append(cache, layer, k, v, slot_ids, positions)


def kv_append(cache, layer, k, v, slot_ids, positions):
    if _use_hip(k):
        from . import hip
        return hip.kv_append(cache, layer, k, v, slot_ids, positions)
    return ref.kv_append(cache, layer, k, v, slot_ids, positions)


def attention(q, cache, layer, meta: AttnMeta):
    if _use_hip(q):
        from . import hip
        if meta.mode == "decode":
            return hip.attention_decode(q, cache, layer, meta)
        if meta.mode == "mixed":
            return hip.attention_mixed(q, cache, layer, meta)
        return hip.attention_prefill(q, cache, layer, meta)
    # the fp32 reference handles any varlen layout uniformly
    return ref.attention(q, cache, layer, meta)


def linear(x, weight, bias=None, packed=None):
    """y = x @ weight.T (+ bias).  Library GEMM for prefill shapes; on the
    decode path (M<=64) a hand-written weight-streaming MFMA kernel runs
    over the pre-packed copy when `packed` is provided (hipBLASLt measured
    1.8-4.6 TB/s vs the ~6.3 TB/s streaming roofline on these shapes)."""
    if _use_hip(x):
        from . import hip
        return hip.linear(x, weight, bias, packed)
    return torch.nn.functional.linear(x, weight, bias)


def pack_weight(w):
    """MFMA-fragment-order packed copy for the decode GEMM (GPU only).
    Every projection is packed: the fused decode chain streams all of
    them (the per-shape lib-vs-stream policy only applies to the generic
    non-chain dispatch in hip.linear)."""
    if w.is_cuda:
        from . import hip
        if hip.available():
            return hip.pack_weight(w)
    return None


def swiglu(gate_up):
    if _use_hip(gate_up):
        from . import hip
        return hip.swiglu(gate_up)
    return ref.swiglu(gate_up)


def gateup_swiglu(x, weight, packed_gu=None):
    """act = swiglu(x @ Wgu^T).  On the decode path (M<=32) with a
    GU-packed copy this is ONE fused weight-streaming kernel; otherwise
    the library GEMM + the swiglu kernel.  (An MT2 variant for 33..64
    exists but measured BELOW the library at 64 users — its direct
    fragment x-loads double the scattered line traffic; see NOTES.)"""
    if packed_gu is not None and x.shape[0] <= 32 and _use_hip(x):
        from . import hip
        if hip.USE_WSTREAM and x.dtype == torch.bfloat16 \
                and x.stride(1) == 1:
            return hip.linear_gu(x, packed_gu, weight.shape[0])
    return swiglu(linear(x, weight))


def pack_weight_gu(w):
    """GU-interleaved packed copy for the fused gate_up+SwiGLU kernel."""
    if w.is_cuda:
        from . import hip
        if hip.available():
            return hip.pack_weight_gu(w)
    return None


# ---- fused-rmsnorm decode chain (GPU-only; the model gates on
# fused_chain availability — tp=1, decode, batch<=32) -----------------

def row_sumsq(x, out=None):
    """sq[m] = sum(x[m]**2) in fp32 (seeds the chain after embedding)."""
    if _use_hip(x):
        from . import hip
        return hip.row_sumsq(x, out)
    s = (x.float() ** 2).sum(-1)
    if out is not None:
        out.copy_(s)
        return out
    return s


def linear_fused(x, packed, N, bias=None, rstd=None, rstd_nt=0,
                 inv_h=0.0, eps=0.0, res=None, sq_out=None, y=None,
                 yfrag=0, K=None, xlds=None, M_frag=32):
    """Weight-streaming GEMM with the fused-chain epilogue/prologue:
    optional rmsnorm scaling of the raw-residual input (norm weight
    folded into the pack), in-place residual add (pass y=res) and
    sum-of-squares partial emission for the next GEMM's rstd.
    K given => x is a 32-row frag-layout buffer streamed linearly;
    yfrag => the output is emitted in frag layout."""
    from . import hip
    return hip.linear_packed(x, packed, bias, N, rstd=rstd,
                             rstd_nt=rstd_nt, inv_h=inv_h, eps=eps,
                             res=res, sq_out=sq_out, y=y, yfrag=yfrag,
                             K=K, xlds=xlds, M_frag=M_frag)


def fragify_sumsq(x, xf=None, sq=None):
    """Standard [M, H] -> 32-row frag layout + per-row sum of squares."""
    from . import hip
    return hip.fragify_sumsq(x, xf, sq)


def attention_decode_frag(q, cache, layer, meta):
    """Decode attention emitting its output in frag layout."""
    from . import hip
    return hip.attention_decode(q, cache, layer, meta, fragout=True)


def gu_fused(x, packed, N, rstd, rstd_nt, inv_h, eps, K=None, yfrag=0,
             M_frag=32):
    """rmsnorm -> gate_up GEMM -> SwiGLU, one kernel."""
    from . import hip
    return hip.linear_gu(x, packed, N, rstd=rstd, rstd_nt=rstd_nt,
                         inv_h=inv_h, eps=eps, K=K, yfrag=yfrag,
                         M_frag=M_frag)


def pack_weight_qkv_rope(w, nl, nkl):
    """Pair-ordered qkv pack for the fused RoPE/KV-append epilogue."""
    if w.is_cuda and w.shape[1] % 64 == 0:
        from . import hip
        if hip.available():
            return hip.pack_weight_qkv_rope(w, nl, nkl)
    return None


def qkv_rope_bias_order(bias, nl, nkl):
    """Reorder a qkv bias to match the pair-ordered pack."""
    from . import hip
    order = torch.tensor(hip.qkv_rope_order(nl, nkl), device=bias.device)
    return bias.index_select(0, order).contiguous()


def qkv_rope_fused(x, packed, N, bias_rp, cache, layer, positions, slots,
                   cos, sin, nl, nkl, rstd, rstd_nt, inv_h, eps,
                   K=None, M_real=32):
    """rmsnorm -> qkv GEMM -> RoPE -> paged KV append, one kernel."""
    from . import hip
    return hip.linear_qkv_rope(x, packed, N, bias_rp, cache, layer,
                               positions, slots, cos, sin, nl, nkl,
                               rstd, rstd_nt, inv_h, eps, K=K,
                               M_real=M_real)


def sample(logits, temperature, top_k, top_p, generator=None):
    # Sampler: torch ops compose on-GPU; custom fused kernel in ops/hip.py
    # handles the greedy + temperature paths.
    if _use_hip(logits):
        from . import hip
        return hip.sample(logits, temperature, top_k, top_p, generator)
    return ref.sample(logits, temperature, top_k, top_p, generator)
