"""Decode-shape microbench for the small per-step kernels (rmsnorm,
swiglu, rope_append, argmax): isolates each op at batch B with cold
activations (cycled buffers) to see how far each sits above its
latency floor.  Run on a GPU box: python tools/perf_elementwise.py [B]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ollamamq_amd.engine.kvcache import PagedKVCache
from ollamamq_amd.ops import hip


def bench(label, fn, n=200, warm=20):
    for _ in range(warm):
        fn(0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        fn(i)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / n
    print(f"{label:>22}: {dt*1e6:7.2f} us")


def main():
    hip.require()
    B = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    H, F, V = 4096, 14336, 128256
    R = 8  # buffer ring so activations are not L2-resident re-reads
    dev = "cuda"
    xs = [torch.randn(B, H, device=dev).bfloat16() for _ in range(R)]
    rs = [torch.randn(B, H, device=dev).bfloat16() for _ in range(R)]
    w = torch.randn(H, device=dev).bfloat16()
    gus = [torch.randn(B, 2 * F, device=dev).bfloat16() for _ in range(R)]
    logits = [torch.randn(B, V, device=dev).bfloat16() for _ in range(R)]

    bench("rmsnorm_residual", lambda i: hip.rmsnorm_residual(
        xs[i % R], rs[i % R], w, 1e-5))
    bench("swiglu", lambda i: hip.swiglu(gus[i % R]))
    bench("argmax (split-V)", lambda i: hip.sample(
        logits[i % R], 0.0, 0, 1.0))

    Hq, KVH, D = 32, 8, 128
    cache = PagedKVCache(1, KVH, D, page_size=16, n_pages=B * 40,
                         max_slots=B, max_ctx=640, device=dev,
                         dtype=torch.bfloat16)
    for i in range(B):
        s = cache.alloc_slot()
        cache.ensure(s, 520)
    qs = [torch.randn(B, Hq, D, device=dev).bfloat16() for _ in range(R)]
    ks = [torch.randn(B, KVH, D, device=dev).bfloat16() for _ in range(R)]
    vs = [torch.randn(B, KVH, D, device=dev).bfloat16() for _ in range(R)]
    pos = torch.full((B,), 500, dtype=torch.int32, device=dev)
    slot = torch.arange(B, dtype=torch.int32, device=dev)
    ang = torch.outer(torch.arange(640, dtype=torch.float32),
                      1.0 / 10000 ** (torch.arange(0, D, 2) / D))
    cos, sin = ang.cos().to(dev), ang.sin().to(dev)
    bench("rope_append (fused)", lambda i: hip.rope_append(
        cache, 0, qs[i % R], ks[i % R], vs[i % R], pos, slot, cos, sin))


if __name__ == "__main__":
    main()
