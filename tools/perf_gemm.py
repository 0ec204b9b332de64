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
        w = torch.randn(N, K).bfloat16().cuda()
        wb = N * K * 2

        def bench(fn):
            for _ in range(10):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            n = 100
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            return (time.perf_counter() - t0) / n

        t_hip = bench(lambda: hip.linear(x, w))
        t_lib = bench(lambda: torch.nn.functional.linear(x, w))
        xt = x.t().contiguous()
        t_tr = bench(lambda: (w @ xt).t())
        print(f"{name:8s} N={N:6d} K={K:6d}: "
              f"ours {t_hip*1e6:7.1f}us {wb/t_hip/1e12:5.2f}TB/s | "
              f"lib {t_lib*1e6:7.1f}us {wb/t_lib/1e12:5.2f}TB/s | "
              f"libT {t_tr*1e6:7.1f}us {wb/t_tr/1e12:5.2f}TB/s")


if __name__ == "__main__":
    main()
