"""Plain-PyTorch fp32 reference implementations of every hot op.

These are the numerics oracle for the gfx950 HIP kernels (tests compare the
HIP path against these at fp32) and the CPU execution path for the
dispatcher/scheduler test suite, which runs with no GPU (the
reference's whole test strategy runs on loopback without real backends).
"""
from __future__ import annotations

import torch


def rmsnorm_residual(x, residual, weight, eps):
    """y = rmsnorm(x + residual) * weight; returns (y, x + residual).

    residual may be None (plain rmsnorm). Always accumulates in fp32.
    """
    if residual is not None:
        residual = (x.float() + residual.float())
    else:
        residual = x.float()
    var = residual.pow(2).mean(dim=-1, keepdim=True)
    y = residual * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype), residual.to(x.dtype)


def rope(q, k, positions, cos, sin):
    """In-place NeoX-style half-rotation RoPE on q [T,Hq,D] and k [T,Hk,D].

    cos/sin: [max_ctx, D/2] host-precomputed tables (guide Appendix B:
    on-device trig turns a memory-bound op VALU-bound).
    """
    c = cos[positions.long()].unsqueeze(1)  # [T,1,D/2]
    s = sin[positions.long()].unsqueeze(1)
    for t in (q, k):
        d2 = t.shape[-1] // 2
        # NB: .float() on an fp32 tensor is a VIEW — compute both halves
        # before mutating t, or the lower-half store corrupts x1.
        x1 = t[..., :d2].float()
        x2 = t[..., d2:].float()
        lo = x1 * c - x2 * s
        hi = x2 * c + x1 * s
        t[..., :d2] = lo.to(t.dtype)
        t[..., d2:] = hi.to(t.dtype)


def kv_append(cache, layer, k, v, slot_ids, positions):
    """Scatter k/v [T, KVH, D] into the paged pool at (slot, position)."""
    ps = cache.page_size
    pages = cache.page_table[slot_ids.long(), (positions // ps).long()]
    offs = (positions % ps).long()
    cache.k_pool[layer, pages.long(), :, offs] = k.to(cache.k_pool.dtype)
    cache.v_pool[layer, pages.long(), :, offs] = v.to(cache.v_pool.dtype)


def _gather_kv(cache, layer, slot, kv_len):
    """Return K,V [kv_len, KVH, D] for one slot from the paged pool."""
    ps = cache.page_size
    n_pages = (kv_len + ps - 1) // ps
    pages = cache.page_table[slot, :n_pages].long()
    k = cache.k_pool[layer, pages]  # [n_pages, KVH, ps, D]
    v = cache.v_pool[layer, pages]
    k = k.permute(0, 2, 1, 3).reshape(n_pages * ps, -1, cache.head_dim)
    v = v.permute(0, 2, 1, 3).reshape(n_pages * ps, -1, cache.head_dim)
    return k[:kv_len], v[:kv_len]


def attention(q, cache, layer, meta):
    """Causal paged attention, prefill (varlen) and decode alike.

    q: [T, Hq, D] flat over sequences; meta gives per-sequence q extents.
    Query token at absolute position p attends KV positions [0, p].
    GQA: Hq queries share Hq/Hkv groups. fp32 math throughout.
    """
    T, Hq, D = q.shape
    out = torch.empty_like(q)
    scale = 1.0 / (D ** 0.5)
    cu = meta.cu_q.tolist()
    slots = meta.slot_ids.tolist()
    lens = meta.seq_lens.tolist()
    for i, slot in enumerate(slots):
        q_i = q[cu[i]:cu[i + 1]].float()          # [qlen, Hq, D]
        qlen = q_i.shape[0]
        kv_len = lens[i]
        k, v = _gather_kv(cache, layer, slot, kv_len)
        k = k.float()
        v = v.float()
        rep = Hq // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)       # [kv, Hq, D]
        v = v.repeat_interleave(rep, dim=1)
        scores = torch.einsum("qhd,khd->hqk", q_i, k) * scale
        # causal: query j sits at absolute position kv_len - qlen + j
        qpos = torch.arange(kv_len - qlen, kv_len, device=q.device)
        kpos = torch.arange(kv_len, device=q.device)
        mask = kpos[None, :] > qpos[:, None]
        if meta.window and meta.window > 0:
            # sliding window: query p attends [max(0, p-window+1), p]
            mask |= kpos[None, :] < qpos[:, None] - meta.window + 1
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hqk,khd->qhd", p, v)
        out[cu[i]:cu[i + 1]] = o.to(q.dtype)
    return out


def swiglu(gate_up):
    """[T, 2F] fused gate/up -> silu(gate) * up, fp32 math."""
    f = gate_up.shape[-1] // 2
    g = gate_up[..., :f].float()
    u = gate_up[..., f:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def sample(logits, temperature, top_k, top_p, generator=None):
    """Per-row sampling: greedy when temperature<=0, else temp/top-k/top-p.

    logits: [B, V]; temperature/top_k/top_p are python scalars (uniform
    across the batch) OR 1-D [B] tensors — each row is filtered by ITS OWN
    top_k/top_p (per-request semantics: mixed sampling params in one batch
    never bleed into each other).  top_k 0 and top_p >= 1 disable the
    respective filter for that row.  Returns [B] int64 token ids.
    """
    B, V = logits.shape
    dev = logits.device
    logits = logits.float()
    if not torch.is_tensor(temperature):
        temperature = torch.full((B,), float(temperature), device=dev)
    if not torch.is_tensor(top_k):
        top_k = torch.full((B,), int(top_k or 0), dtype=torch.long,
                           device=dev)
    if not torch.is_tensor(top_p):
        top_p = torch.full((B,), float(top_p if top_p else 1.0), device=dev)
    top_k = top_k.long()
    greedy = temperature <= 0
    out = torch.empty(B, dtype=torch.long, device=dev)
    if greedy.any():
        out[greedy] = logits[greedy].argmax(dim=-1)
    rest = ~greedy
    if rest.any():
        l = logits[rest] / temperature[rest].unsqueeze(1)
        tk = top_k[rest]
        tp = top_p[rest]
        if bool(((tk > 0) & (tk < V)).any()) or bool((tp < 1.0).any()):
            # one descending sort serves both filters; applying top-k as a
            # post-softmax mask + renormalize is identical to the -inf
            # pre-softmax mask (softmax restricted to a subset)
            sv, si = l.sort(dim=-1, descending=True)
            sp = torch.softmax(sv, dim=-1)
            ar = torch.arange(V, device=dev)
            kk = torch.where(tk > 0, tk.clamp(max=V), torch.full_like(tk, V))
            keep = ar.unsqueeze(0) < kk.unsqueeze(1)          # per-row top-k
            cum = sp.cumsum(dim=-1)
            keep &= (cum - sp) < tp.unsqueeze(1)   # keep token crossing mass
            keep[:, 0] = True                      # never empty a row
            sp = sp * keep
            sp = sp / sp.sum(dim=-1, keepdim=True)
            pick = torch.multinomial(sp, 1, generator=generator).squeeze(1)
            out[rest] = si.gather(1, pick.unsqueeze(1)).squeeze(1)
        else:
            probs = torch.softmax(l, dim=-1)
            out[rest] = torch.multinomial(
                probs, 1, generator=generator).squeeze(1)
    return out
