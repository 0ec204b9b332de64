"""Strided-view handling: kernels must operate in place on fused-QKV views
(the model never copies q/k/v out of the projection output)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from ollamamq_amd.ops import reference as ref
from ollamamq_amd.engine.kvcache import PagedKVCache


def test_rope_on_fused_qkv_views():
    from ollamamq_amd.ops import hip
    hip.require()
    T, Hq, Hk, D = 9, 4, 2, 128
    g = torch.Generator().manual_seed(1)
    qkv = torch.randn(T, (Hq + 2 * Hk) * D, generator=g).bfloat16().cuda()
    q = qkv[:, :Hq * D].view(T, Hq, D)
    k = qkv[:, Hq * D:(Hq + Hk) * D].view(T, Hk, D)
    assert not q.is_contiguous()
    pos = torch.arange(T, dtype=torch.int32).cuda() + 3
    ang = torch.outer(torch.arange(64, dtype=torch.float32),
                      1.0 / 10000 ** (torch.arange(0, D, 2) / D))
    cos, sin = ang.cos().cuda(), ang.sin().cuda()
    q_ref = q.float().cpu().clone()
    k_ref = k.float().cpu().clone()
    ref.rope(q_ref, k_ref, pos.cpu(), cos.cpu(), sin.cpu())
    v_before = qkv[:, (Hq + Hk) * D:].clone()
    hip.rope(q, k, pos, cos, sin)
    torch.testing.assert_close(q.float().cpu(), q_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(k.float().cpu(), k_ref, atol=2e-2, rtol=2e-2)
    # v region untouched
    assert torch.equal(qkv[:, (Hq + Hk) * D:], v_before)


def test_kv_append_strided_source():
    from ollamamq_amd.ops import hip
    hip.require()
    T, Hq, KVH, D = 21, 4, 2, 128
    g = torch.Generator().manual_seed(2)
    qkv = torch.randn(T, (Hq + 2 * KVH) * D, generator=g).bfloat16().cuda()
    k = qkv[:, Hq * D:(Hq + KVH) * D].view(T, KVH, D)
    v = qkv[:, (Hq + KVH) * D:].view(T, KVH, D)
    gc = PagedKVCache(1, KVH, D, page_size=16, n_pages=8, max_slots=2,
                      max_ctx=128, device="cuda", dtype=torch.bfloat16)
    cc = PagedKVCache(1, KVH, D, page_size=16, n_pages=8, max_slots=2,
                      max_ctx=128, device="cpu", dtype=torch.float32)
    s = gc.alloc_slot(); cc.alloc_slot()
    gc.ensure(s, T); cc.ensure(s, T)
    cc.page_table.copy_(gc.page_table.cpu())
    slot = torch.zeros(T, dtype=torch.int32, device="cuda")
    pos = torch.arange(T, dtype=torch.int32, device="cuda")
    hip.kv_append(gc, 0, k, v, slot, pos)
    ref.kv_append(cc, 0, k.float().cpu(), v.float().cpu(),
                  slot.cpu(), pos.cpu())
    torch.testing.assert_close(gc.k_pool.float().cpu(), cc.k_pool,
                               atol=0, rtol=0)
    torch.testing.assert_close(gc.v_pool.float().cpu(), cc.v_pool,
                               atol=0, rtol=0)
