"""Microbench: weight-streaming packed GEMM vs hipBLASLt on the 8B decode
shapes, with a ksplit x prefetch-depth sweep and a pure-stream arm (same
grid/addressing, loads only) that shows each geometry's load-path ceiling.

Run on a GPU box: python tools/perf_gemm.py [M] [--sweep]

COLD-weight protocol: cycles enough weight copies that the 256 MiB L3
never serves a re-read (the serving loop streams weights from HBM once
per step; r01's fragment-direct kernel won warm and lost cold)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from ollamamq_amd.ops import hip


def bench(fn, args, ncopy, n=50):
    for i in range(5):
        fn(args[i % ncopy])
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        fn(args[i % ncopy])
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    hip.require()
    M = int(sys.argv[1]) if len(sys.argv) > 1 and sys.argv[1].isdigit() \
        else 32
    sweep = "--sweep" in sys.argv
    shapes = [("qkv", 6144, 4096), ("o", 4096, 4096),
              ("gate_up", 28672, 4096), ("down", 4096, 14336),
              ("logits", 128256, 4096)]
    for name, N, K in shapes:
        x = torch.randn(M, K).bfloat16().cuda()
        ncopy = max(2, (600 << 20) // (N * K * 2) + 1)
        ws = [torch.randn(N, K).bfloat16().cuda() for _ in range(ncopy)]
        pks = [hip.pack_weight(w) for w in ws]
        wb = N * K * 2

        t_lib = bench(lambda w: torch.nn.functional.linear(x, w), ws,
                      ncopy)
        print(f"{name:8s} N={N:6d} K={K:6d}: lib {t_lib*1e6:7.1f}us "
              f"{wb/t_lib/1e12:5.2f}TB/s")
        ks_list = ([1, 2, 4, 8] if sweep
                   else [hip._wstream_ksplit(N, K)])
        for ks in ks_list:
            if (K // 64) // (ks * 8) < 1:
                continue
            t_pure = bench(lambda p, k=ks: hip.wstream_pure(p, N, K, k),
                           pks, ncopy)
            line = (f"    ks={ks}: pure {t_pure*1e6:7.1f}us "
                    f"{wb/t_pure/1e12:5.2f}TB/s")
            for tag, kw in (("d1", dict(depth=1, xlds=0)),
                            ("xl", dict(depth=1, xlds=1))):
                t = bench(lambda p, k=ks, kw=kw:
                          hip.linear_packed(x, p, None, N, ks=k, **kw),
                          pks, ncopy)
                line += (f" | {tag} {t*1e6:7.1f}us "
                         f"{wb/t/1e12:5.2f}TB/s")
            print(line, flush=True)
        del ws, pks
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
