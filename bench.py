#!/usr/bin/env python3
"""Flagship serving benchmark — driver contract (see BASELINE.json).

Measures the BASELINE.json headline: aggregate tokens/sec serving Llama-3-8B
to 32 concurrent users per GPU backend (weak scaling: each of the N ranks is
one independent MI355X backend running the in-process HIP engine, exactly
config 3 of BASELINE.json).  A "step" is one continuous-batching decode
iteration: every resident sequence advances one token through the full
hand-written HIP path (fused RMSNorm, RoPE, paged attention, SwiGLU,
sampler) + hipBLASLt projections.  Warmup admits + prefills the users, so
the timed region is pure steady-state serving; p50 queue-wait (submit ->
first token) is measured during warmup and reported in config.

Launch (driver):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

The reference publishes no number for this metric =>
vs_baseline is null.
"""
import argparse
import json
import os
import statistics
import sys
import threading
import time

import torch


def run_stack(args):
    """Full-stack serving benchmark: native dispatcher + UDS engine
    workers + HTTP streaming clients.  The timed
    region crosses the whole reference-parity path — HTTP ingress,
    per-user queues, scheduler, UDS hop, engine, token stream-out — and
    p50 queue-wait is the DISPATCHER's own /admin/stats definition
    (enqueue -> dispatch), not engine TTFT."""
    import socket
    import subprocess
    import urllib.request

    has_gpu = torch.cuda.is_available()
    model = args.model or ("llama3-8b" if has_gpu else "tiny-cpu")
    users = args.users if has_gpu else min(args.users, 4)
    port = 11640 + (os.getpid() % 199)
    # generation-heavy requests: with short answers the measurement is
    # dominated by per-request prefill + batch churn rather than serving
    # throughput (the reference metric is generation tokens/sec)
    max_tokens = 160 if has_gpu else 12
    prompt_len = min(args.prompt_len, 256) if has_gpu else 16
    warm_s = max(4, args.warmup)
    meas_s = max(8, args.steps // 3) if has_gpu else max(6, args.steps // 6)

    cmd = [sys.executable, "-m", "ollamamq_amd.launch",
           "--gpus", str(args.gpus), "--model", model,
           "--max-ctx", str(args.max_ctx),
           "--max-batch", str(max(32, users)),
           "--port", str(port), "--no-tui",
           "-c", "/nonexistent/appconf.yaml"]
    node = subprocess.Popen(cmd, stdout=subprocess.DEVNULL,
                            stderr=subprocess.DEVNULL,
                            start_new_session=True)
    base = f"http://127.0.0.1:{port}"
    try:
        deadline = time.time() + 600
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(base + "/health",
                                            timeout=2) as r:
                    if r.read() == b"OK":
                        break
            except OSError:
                time.sleep(0.5)
        else:
            raise SystemExit("stack did not come up")

        stop = threading.Event()
        token_ts = []          # (time, 1) per streamed token
        ttfts = []             # client-side submit -> first token (ms)
        ts_lock = threading.Lock()

        def user_loop(uid):
            body = json.dumps({
                "model": model, "prompt": "x" * prompt_len,
                "options": {"num_predict": max_tokens}, "stream": True,
            }).encode()
            while not stop.is_set():
                try:
                    req = urllib.request.Request(
                        base + "/api/generate", data=body,
                        headers={"Content-Type": "application/json",
                                 "X-User-ID": f"bench-user-{uid}"})
                    local = []
                    t_sub = time.monotonic()
                    first = None
                    with urllib.request.urlopen(req, timeout=120) as r:
                        for line in r:
                            if b'"done": false' in line or \
                                    b'"done":false' in line:
                                now = time.monotonic()
                                if first is None:
                                    first = (now - t_sub) * 1e3
                                local.append(now)
                    with ts_lock:
                        token_ts.extend(local)
                        if first is not None:
                            ttfts.append(first)
                except OSError:
                    time.sleep(0.2)

        threads = [threading.Thread(target=user_loop, args=(u,),
                                    daemon=True) for u in range(users)]
        for t in threads:
            t.start()
        time.sleep(warm_s)
        t0 = time.monotonic()
        time.sleep(meas_s)
        t1 = time.monotonic()
        stop.set()
        with urllib.request.urlopen(base + "/admin/stats", timeout=10) \
                as r:
            stats = json.loads(r.read())
        time.sleep(0.5)
        with ts_lock:
            n_tok = sum(1 for ts in token_ts if t0 <= ts <= t1)
        value = n_tok / (t1 - t0)
        out = {
            "metric": "agg_tokens_per_sec",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round((t1 - t0) * 1e3 / max(1, n_tok // users),
                                 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if has_gpu else "fp32",
            "data": "synthetic prompts, random-init weights",
            "config": {
                "model": model,
                "mode": "full-stack (HTTP -> dispatcher -> UDS workers)",
                "global_batch": users,
                "users_per_gpu": users // max(1, args.gpus),
                "seq_len": prompt_len,
                "max_ctx": args.max_ctx,
                "parallelism": f"dp{args.gpus} via native dispatcher",
                "p50_queue_wait_ms":
                    stats.get("queue_wait", {}).get("p50_ms"),
                "p99_queue_wait_ms":
                    stats.get("queue_wait", {}).get("p99_ms"),
                "client_ttft_p50_ms":
                    round(statistics.median(ttfts), 1) if ttfts else None,
                "requests_processed": stats.get("processed"),
                "measure_s": round(t1 - t0, 1),
            },
        }
        print(json.dumps(out), flush=True)
    finally:
        try:
            os.killpg(node.pid, 15)
        except (ProcessLookupError, PermissionError):
            node.terminate()
        node.wait(timeout=30)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--users", type=int, default=32,
                    help="concurrent users per GPU backend")
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--model", type=str, default=None)
    ap.add_argument("--max-ctx", type=int, default=4096)
    ap.add_argument("--prefill-chunk", type=int, default=4096)
    ap.add_argument("--stack", action="store_true",
                    help="benchmark through the full dispatcher stack "
                         "(HTTP + scheduler + UDS workers) instead of "
                         "the engine-only kernel metric")
    args = ap.parse_args()

    if args.stack:
        run_stack(args)
        return

    from ollamamq_amd.models import LlamaModel, PRESETS
    from ollamamq_amd.engine import LlamaEngine, PagedKVCache, GenParams

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    has_gpu = torch.cuda.is_available()
    if has_gpu:
        # bind the device BEFORE the NCCL/RCCL communicator is created
        torch.cuda.set_device(local_rank)
    dist = world > 1
    if dist:
        torch.distributed.init_process_group(
            backend="nccl" if has_gpu else "gloo")

    if has_gpu:
        device = f"cuda:{local_rank}"
        dtype = torch.bfloat16
        model_name = args.model or "llama3-8b"
        users = args.users
        prompt_len = args.prompt_len
    else:
        device = "cpu"
        dtype = torch.float32
        model_name = args.model or "tiny-cpu"
        users = min(args.users, 4)
        prompt_len = min(args.prompt_len, 32)

    cfg = PRESETS[model_name]
    gen_budget = args.warmup + args.steps + 8
    ctx = min(args.max_ctx, cfg.max_ctx)
    assert prompt_len + gen_budget <= ctx, "context too small for bench"
    n_pages = (users + 2) * ((prompt_len + gen_budget + 15) // 16 + 2)

    t_load0 = time.monotonic()
    model = LlamaModel(cfg, device=device, dtype=dtype, seed=1234,
                       fast_init=has_gpu)
    kv = PagedKVCache.for_model(
        cfg, n_pages=n_pages, max_slots=users + 2, max_ctx=ctx,
        device=device, dtype=dtype)
    eng = LlamaEngine(model, kv, max_batch=users,
                      prefill_chunk=args.prefill_chunk)
    load_s = time.monotonic() - t_load0

    # --- submit synthetic users -------------------------------------------
    g = torch.Generator().manual_seed(42 + rank)
    submit_t = {}
    ttft = {}
    for u in range(users):
        prompt = torch.randint(0, cfg.vocab, (prompt_len,), generator=g).tolist()
        t_sub = time.monotonic()
        sid = eng.submit(prompt, GenParams(max_tokens=10 ** 9))
        submit_t[sid] = t_sub

    # --- warmup: prefill everyone + W decode steps ------------------------
    guard = 0
    while eng.waiting and guard < 10000:
        eng.step()
        guard += 1
    for sid, seq in list(eng.seqs.items()):
        if seq.first_token_at is not None and sid in submit_t:
            ttft[sid] = (seq.first_token_at - submit_t[sid]) * 1e3
    for _ in range(args.warmup):
        eng.step()
    assert len(eng.running) == users, \
        f"rank {rank}: {len(eng.running)} running != {users}"

    # --- timed region: K decode steps -------------------------------------
    if has_gpu:
        torch.cuda.synchronize()
    if dist:
        torch.distributed.barrier()
    t0 = time.monotonic()
    for _ in range(args.steps):
        eng.step()
    if has_gpu:
        torch.cuda.synchronize()
    elapsed = time.monotonic() - t0
    if dist:
        # nccl (=RCCL) requires device tensors; gloo (CPU tests) takes CPU
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=f"cuda:{local_rank}" if has_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
        torch.distributed.barrier()

    tokens_total = users * args.steps * world
    value = tokens_total / elapsed
    p50_wait = statistics.median(ttft.values()) if ttft else None

    if rank == 0:
        out = {
            "metric": "agg_tokens_per_sec",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if has_gpu else "fp32",
            "data": "synthetic prompts, random-init weights",
            "config": {
                "model": cfg.name,
                "global_batch": users * world,
                "users_per_gpu": users,
                "seq_len": prompt_len,
                "max_ctx": ctx,
                "parallelism": f"dp{world} (independent GPU backends)",
                "p50_queue_wait_ms": round(p50_wait, 1) if p50_wait else None,
                "weights_gb_per_gpu": round(model.weight_bytes() / 2 ** 30, 2),
                "load_s": round(load_s, 1),
            },
        }
        print(json.dumps(out), flush=True)
    if dist:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
