"""Microbench: weight-streaming packed GEMM vs hipBLASLt (torch F.linear)
on the 8B decode shapes.  Run on a GPU box: python tools/perf_gemm.py [M]

COLD-weight protocol: cycles enough weight copies that the 256 MiB L3
never serves a re-read — the serving loop streams each layer's weights
from HBM once per step, and an L3-warm microbench lies about that regime
(r01's fragment-direct kernel won warm and lost cold; bench.py A/B is
the final word)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from ollamamq_amd.ops import hip


def main():
    hip.require()
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    shapes = [("qkv", 6144, 4096), ("o", 4096, 4096),
              ("gate_up", 28672, 4096), ("down", 4096, 14336),
              ("logits", 128256, 4096)]
    for name, N, K in shapes:
        x = torch.randn(M, K).bfloat16().cuda()
        ncopy = max(2, (400 << 20) // (N * K * 2) + 1)
        ws = [torch.randn(N, K).bfloat16().cuda() for _ in range(ncopy)]
        pks = [hip.pack_weight(w) for w in ws]
        wb = N * K * 2

        def bench(fn, args):
            for i in range(5):
                fn(args[i % ncopy])
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            n = 50
            for i in range(n):
                fn(args[i % ncopy])
            torch.cuda.synchronize()
            return (time.perf_counter() - t0) / n

        t_ws = bench(lambda p: hip.linear_packed(x, p, None, N), pks)
        t_lib = bench(lambda w: torch.nn.functional.linear(x, w), ws)
        from ollamamq_amd.ops.hip import _wstream_ksplit
        print(f"{name:8s} N={N:6d} K={K:6d} ks={_wstream_ksplit(N, K)} "
              f"copies={ncopy}: "
              f"wstream {t_ws*1e6:7.1f}us {wb/t_ws/1e12:5.2f}TB/s | "
              f"lib {t_lib*1e6:7.1f}us {wb/t_lib/1e12:5.2f}TB/s")
        del ws, pks
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
