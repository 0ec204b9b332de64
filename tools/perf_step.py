"""Split the decode step into GPU time vs host overhead.

Runs the 8B engine to steady state, then times (a) full engine.step()
iterations and (b) bare graph replays of the same captured step — the gap
is Python/host bookkeeping that the GPU waits on.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from ollamamq_amd.models import LlamaModel, PRESETS
from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams


def main():
    model_name = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    users = int(sys.argv[2]) if len(sys.argv) > 2 else 32
    cfg = PRESETS[model_name]
    dev = "cuda"
    n_pages = (users + 2) * ((512 + 200 + 15) // 16 + 2)
    model = LlamaModel(cfg, device=dev, dtype=torch.bfloat16, seed=1,
                       fast_init=True)
    kv = PagedKVCache.for_model(cfg, n_pages=n_pages, max_slots=users + 2,
                                max_ctx=1024, device=dev,
                                dtype=torch.bfloat16)
    eng = LlamaEngine(model, kv, max_batch=users)
    g = torch.Generator().manual_seed(3)
    for _ in range(users):
        eng.submit(torch.randint(0, cfg.vocab, (512,), generator=g).tolist(),
                   GenParams(max_tokens=10 ** 9))
    guard = 0
    while eng.waiting and guard < 1000:
        eng.step()
        guard += 1
    for _ in range(5):
        eng.step()
    torch.cuda.synchronize()

    n = 30
    t0 = time.perf_counter()
    for _ in range(n):
        eng.step()
    torch.cuda.synchronize()
    full = (time.perf_counter() - t0) / n

    entry = eng._graphs.get(users)
    assert entry is not None
    t0 = time.perf_counter()
    for _ in range(n):
        entry["graph"].replay()
    torch.cuda.synchronize()
    replay = (time.perf_counter() - t0) / n

    # sampler-only cost (argmax + tolist sync)
    logits = entry["logits"]
    t0 = time.perf_counter()
    for _ in range(n):
        eng._sample(eng.running, logits)
    torch.cuda.synchronize()
    sample = (time.perf_counter() - t0) / n

    print(f"full step : {full*1e3:7.3f} ms")
    print(f"graph GPU : {replay*1e3:7.3f} ms")
    print(f"sample+sync:{sample*1e3:7.3f} ms")
    print(f"host gap  : {(full-replay-sample)*1e3:7.3f} ms")


if __name__ == "__main__":
    main()
