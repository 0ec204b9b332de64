"""Microbench: decode attention old (unsplit) vs flash-decoding path.

Run on a GPU box:  python tools/perf_decode.py [ctx] [batch]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ollamamq_amd.engine.kvcache import PagedKVCache
from ollamamq_amd.ops.interface import AttnMeta
from ollamamq_amd.ops import hip


def main():
    hip.require()
    ctx = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
    B = int(sys.argv[2]) if len(sys.argv) > 2 else 32
    Hq_arg = int(sys.argv[3]) if len(sys.argv) > 3 else 32
    KVH_arg = int(sys.argv[4]) if len(sys.argv) > 4 else 8
    # 8 layers cycled per call: the working set (8 × B×KVH×ctx×512B) blows
    # past the 256 MiB L3 like the real 32-layer model does — timing one
    # resident layer would measure L3, not HBM (the perf_gemm.py lesson)
    Hq, KVH, D, L = Hq_arg, KVH_arg, 128, 8
    dev = "cuda"
    n_pages = B * (ctx // 16 + 2)
    cache = PagedKVCache(L, KVH, D, page_size=16, n_pages=n_pages,
                         max_slots=B, max_ctx=ctx + 64, device=dev,
                         dtype=torch.bfloat16)
    for i in range(B):
        s = cache.alloc_slot()
        cache.ensure(s, ctx)
    cache.k_pool.normal_()
    cache.v_pool.normal_()
    q = torch.randn(B, Hq, D, device=dev).bfloat16()
    meta = AttnMeta(
        mode="decode",
        slot_ids=torch.arange(B, dtype=torch.int32, device=dev),
        seq_lens=torch.full((B,), ctx, dtype=torch.int32, device=dev),
        cu_q=torch.arange(B + 1, dtype=torch.int32, device=dev),
        logits_idx=None, max_q=1, max_kv=ctx)

    kv_bytes = B * KVH * ctx * D * 2 * 2

    def bench(fn, label):
        for _ in range(5):
            out = fn(0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        n = 64
        for i in range(n):
            out = fn(i % L)     # cycle layers: cold KV every call
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / n
        print(f"{label}: {dt*1e6:8.1f} us  {kv_bytes/dt/1e12:6.2f} TB/s")
        return out

    o_new = bench(lambda l: hip.attention_decode(q, cache, l, meta),
                  "split ")
    for sp in (1, 2, 4, 8):
        bench(lambda l, sp=sp: hip.decode_pure(cache, l, meta, sp),
              f"pure{sp}")
    o_old = bench(lambda l: hip._attention(q, cache, l, meta, 1), "legacy")
    diff = (o_new.float() - o_old.float()).abs().max().item()
    print("max |new-old| =", diff)
    assert diff < 3e-2, "split path disagrees with legacy"


if __name__ == "__main__":
    main()
