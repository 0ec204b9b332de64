"""TP worker group lockstep on CPU/gloo: rank 0 serves, both ranks step the
same engine replica; generated tokens must match the single-rank engine."""
import os

import torch
import torch.multiprocessing as mp

from ollamamq_amd.engine import GenParams

PROMPT = [3, 1, 4, 1, 5, 9]
N = 5


def _single():
    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import LlamaEngine, PagedKVCache
    cfg = PRESETS["tiny"]
    model = LlamaModel(cfg, device="cpu", dtype=torch.float32, seed=1234)
    kv = PagedKVCache.for_model(cfg, n_pages=32, max_slots=4, max_ctx=128)
    eng = LlamaEngine(model, kv, max_batch=4)
    sid = eng.submit(PROMPT, GenParams(max_tokens=N))
    seq = eng.seqs[sid]
    for _ in range(64):
        eng.step()
        if not eng.has_work():
            break
    return seq.generated


def _rank(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    from ollamamq_amd.engine.tp_worker import TPWorker
    dist.init_process_group("gloo", rank=rank, world_size=world)
    w = TPWorker(rank, world, 0, max_batch=4, default_ctx=128)
    if rank == 0:
        err = w.load("tiny", 128)
        assert err is None, err
        toks = []
        import threading
        done = threading.Event()

        def on_token(t, fin):
            if fin:
                done.set()
            else:
                toks.append(t)

        w.generate("tiny", PROMPT,
                   GenParams(max_tokens=N), on_token)
        assert done.wait(timeout=120)
        q.put(toks)
    else:
        # follower: the engine loop thread does the work; idle here until
        # the parent kills the process
        import time
        time.sleep(120)


def test_tp_worker_lockstep_matches_single():
    expected = _single()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank, args=(r, 2, 29587, q), daemon=True)
             for r in range(2)]
    for p in procs:
        p.start()
    toks = q.get(timeout=300)
    for p in procs:
        p.terminate()
        p.join(timeout=30)
    assert toks == expected, f"{toks} != {expected}"
