"""Microbench: skinny MFMA GEMM vs hipBLASLt (torch F.linear) on the 8B
decode shapes.  Run on a GPU box: python tools/perf_gemm.py [M]"""
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
        # COLD weights: cycle enough copies that the 256 MiB L3 never
        # serves a re-read (the serving loop streams each layer's weights
        # from HBM once per step — an L3-warm microbench lies about this
        # regime; bench.py A/B proved it)
        ncopy = max(2, (400 << 20) // (N * K * 2) + 1)
        ws = [torch.randn(N, K).bfloat16().cuda() for _ in range(ncopy)]
        wb = N * K * 2

        def bench(fn):
            for i in range(5):
                fn(ws[i % ncopy])
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            n = 50
            for i in range(n):
                fn(ws[i % ncopy])
            torch.cuda.synchronize()
            return (time.perf_counter() - t0) / n

        t_hip = bench(lambda w: hip.linear(x, w))
        t_lib = bench(lambda w: torch.nn.functional.linear(x, w))
        print(f"{name:8s} N={N:6d} K={K:6d} copies={ncopy}: "
              f"ours {t_hip*1e6:7.1f}us {wb/t_hip/1e12:5.2f}TB/s | "
              f"lib {t_lib*1e6:7.1f}us {wb/t_lib/1e12:5.2f}TB/s")
        for w in ws:
            del w
        del ws
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
