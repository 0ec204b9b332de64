"""Microbench: MFMA prefill attention vs the VALU paged_attn path.

    python tools/perf_prefill.py [total_tokens] [seq_len]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from ollamamq_amd.engine.kvcache import PagedKVCache
from ollamamq_amd.ops.interface import AttnMeta
from ollamamq_amd.ops import reference as ref
from ollamamq_amd.ops import hip


def main():
    hip.require()
    total = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    seq = int(sys.argv[2]) if len(sys.argv) > 2 else 512
    Hq, KVH, D, L = 32, 8, 128, 1
    B = total // seq
    dev = "cuda"
    cache = PagedKVCache(L, KVH, D, page_size=16, n_pages=B * (seq // 16 + 2),
                         max_slots=B, max_ctx=seq + 64, device=dev,
                         dtype=torch.bfloat16)
    for i in range(B):
        s = cache.alloc_slot()
        cache.ensure(s, seq)
    cache.k_pool.normal_()
    cache.v_pool.normal_()
    q = torch.randn(total, Hq, D, device=dev).bfloat16()
    cu = torch.arange(B + 1, dtype=torch.int32, device=dev) * seq
    meta = AttnMeta(
        mode="prefill",
        slot_ids=torch.arange(B, dtype=torch.int32, device=dev),
        seq_lens=torch.full((B,), seq, dtype=torch.int32, device=dev),
        cu_q=cu, logits_idx=None, max_q=seq, max_kv=seq)

    flops = 4.0 * B * seq * seq / 2 * D * Hq

    def bench(fn, label):
        for _ in range(3):
            out = fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        n = 20
        for _ in range(n):
            out = fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / n
        print(f"{label}: {dt*1e3:8.3f} ms  {flops/dt/1e12:7.1f} TF/s")
        return out

    o_mfma = bench(lambda: hip.attention_prefill(q, cache, 0, meta), "mfma")
    os.environ["OLLAMAMQ_VALU_PREFILL"] = "1"
    o_valu = bench(lambda: hip.attention_prefill(q, cache, 0, meta), "valu")
    del os.environ["OLLAMAMQ_VALU_PREFILL"]
    diff = (o_mfma.float() - o_valu.float()).abs().max().item()
    print("max |mfma - valu| =", diff)
    # numerics vs fp32 reference on a small slice
    meta_c = AttnMeta(
        mode="prefill",
        slot_ids=meta.slot_ids[:1].cpu(), seq_lens=meta.seq_lens[:1].cpu(),
        cu_q=torch.tensor([0, seq], dtype=torch.int32),
        logits_idx=None, max_q=seq, max_kv=seq)
    cc = PagedKVCache(L, KVH, D, page_size=16, n_pages=seq // 16 + 2,
                      max_slots=1, max_ctx=seq + 64, device="cpu",
                      dtype=torch.float32)
    s = cc.alloc_slot()
    cc.ensure(s, seq)
    cc.page_table.zero_()
    cc.page_table[0, :seq // 16 + 1] = torch.arange(seq // 16 + 1,
                                                    dtype=torch.int32)
    # mirror slot 0's pages
    pages = cache.page_table[0, :seq // 16 + 1].long().cpu()
    cc.k_pool[0, :len(pages)] = cache.k_pool[0, pages].float().cpu()
    cc.v_pool[0, :len(pages)] = cache.v_pool[0, pages].float().cpu()
    o_ref = ref.attention(q[:seq].float().cpu(), cc, 0, meta_c)
    err = (o_mfma[:seq].float().cpu() - o_ref).abs().max().item()
    print("max |mfma - fp32 ref| =", err)
    assert err < 5e-2, "numerics mismatch"


if __name__ == "__main__":
    main()
