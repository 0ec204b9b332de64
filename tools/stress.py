"""Multi-user randomized stress driver (reference test_dispatcher.sh
parity: 50 users × 1-12 randomized requests over 4 endpoints × 2 models,
10% early client cancels) against a running ollamamq-amd node.

    python tools/stress.py --base http://127.0.0.1:11435 \
        --users 50 --models llama3-8b

Prints aggregate tokens/sec and queue-wait percentiles at the end — the
BASELINE.json serving metric, measured through the full HTTP stack.
"""
import argparse
import concurrent.futures as cf
import json
import os
import random
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import httpx

ENDPOINTS = ["/api/generate", "/api/chat", "/v1/chat/completions",
             "/v1/completions"]


def one_request(base, user, model, rng, cancel_pct, max_tokens,
                sampled_pct=0.0):
    # ~5% of traffic hits the embedding endpoints (mixed workload)
    if rng.random() < 0.05:
        ep = rng.choice(["/api/embed", "/v1/embeddings"])
        body = {"model": model,
                "input": ["".join(rng.choice("abcde ") for _ in range(20))]}
        t0 = time.monotonic()
        try:
            r = httpx.post(base + ep, json=body,
                           headers={"X-User-ID": user}, timeout=60)
            return {"user": user, "status": r.status_code, "tokens": 0,
                    "ttft": None, "dur": time.monotonic() - t0}
        except Exception as e:
            return {"user": user, "status": -1, "error": str(e),
                    "tokens": 0, "ttft": None, "dur": 0.0}
    ep = rng.choice(ENDPOINTS)
    openai = ep.startswith("/v1/")
    prompt = "".join(rng.choice("abcdefghij ") for _ in range(rng.randint(8, 200)))
    if "chat" in ep:
        msg = {"role": "user", "content": prompt}
        # reference-stress parity: ~5% multimodal requests carry a
        # base64 image; text models accept and ignore the bytes
        if rng.random() < 0.05:
            import base64
            msg["images"] = [base64.b64encode(
                rng.randbytes(256)).decode()]
        body = {"model": model, "messages": [msg]}
    else:
        body = {"model": model, "prompt": prompt}
    if openai:
        body["max_tokens"] = max_tokens
        body["stream"] = True
    else:
        body["options"] = {"num_predict": max_tokens}
    # sampled share: exercises the in-graph Gumbel (temperature-only)
    # and capped top-k/top-p decode tails under real serving churn
    if rng.random() < sampled_pct:
        opts = body.setdefault("options", {})
        opts["temperature"] = round(rng.uniform(0.4, 1.2), 2)
        if rng.random() < 0.5:
            opts["top_k"] = rng.choice([5, 20, 40])
        if rng.random() < 0.5:
            opts["top_p"] = round(rng.uniform(0.7, 0.98), 2)
        if rng.random() < 0.3:
            opts["seed"] = rng.randint(1, 10 ** 6)
        if openai:
            body["temperature"] = opts["temperature"]
    cancel = rng.random() < cancel_pct
    t0 = time.monotonic()
    ttft = None
    n_tokens = 0
    try:
        with httpx.stream("POST", base + ep, json=body,
                          headers={"X-User-ID": user},
                          timeout=300.0) as r:
            if r.status_code != 200:
                return {"user": user, "status": r.status_code,
                        "tokens": 0, "ttft": None, "dur": 0.0}
            for chunk in r.iter_text():
                if ttft is None and chunk.strip():
                    ttft = time.monotonic() - t0
                n_tokens += chunk.count('"content"') + chunk.count('"response"') \
                    + chunk.count('"text"')
                if cancel and time.monotonic() - t0 > 0.3:
                    break  # simulated client disconnect
            return {"user": user, "status": 200, "tokens": n_tokens,
                    "ttft": ttft, "dur": time.monotonic() - t0,
                    "cancelled": cancel}
    except Exception as e:
        return {"user": user, "status": -1, "error": str(e), "tokens": 0,
                "ttft": None, "dur": time.monotonic() - t0}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--base", default="http://127.0.0.1:11435")
    ap.add_argument("--users", type=int, default=50)
    ap.add_argument("--models", default="llama3-8b")
    ap.add_argument("--max-tokens", type=int, default=24)
    ap.add_argument("--cancel-pct", type=float, default=0.10)
    ap.add_argument("--par", type=int, default=32)
    ap.add_argument("--sampled-pct", type=float, default=0.0,
                    help="fraction of requests with stochastic sampling "
                         "params (temperature/top-k/top-p/seed)")
    args = ap.parse_args()
    models = args.models.split(",")

    r = httpx.get(args.base + "/health", timeout=5.0)
    assert r.status_code == 200, "node not healthy"

    jobs = []
    for u in range(args.users):
        rng = random.Random(1000 + u)
        for _ in range(rng.randint(1, 12)):
            jobs.append((f"user{u:02d}", rng.choice(models), rng))
    random.Random(7).shuffle(jobs)

    t0 = time.monotonic()
    results = []
    with cf.ThreadPoolExecutor(args.par) as ex:
        futs = [ex.submit(one_request, args.base, u, m, rng,
                          args.cancel_pct, args.max_tokens,
                          args.sampled_pct)
                for u, m, rng in jobs]
        for f in cf.as_completed(futs):
            results.append(f.result())
    wall = time.monotonic() - t0

    ok = [r for r in results if r["status"] == 200]
    errs = [r for r in results if r["status"] not in (200,)]
    tokens = sum(r["tokens"] for r in ok)
    ttfts = sorted(r["ttft"] for r in ok if r["ttft"] is not None)

    def pct(p):
        return ttfts[min(len(ttfts) - 1, int(p * len(ttfts)))] * 1e3 \
            if ttfts else None

    print(json.dumps({
        "requests": len(results), "ok": len(ok), "errors": len(errs),
        "cancelled": sum(1 for r in ok if r.get("cancelled")),
        "wall_s": round(wall, 2),
        "agg_tokens_per_sec": round(tokens / wall, 1),
        "queue_wait_ms": {"p50": pct(0.5) and round(pct(0.5), 1),
                          "p90": pct(0.9) and round(pct(0.9), 1),
                          "p99": pct(0.99) and round(pct(0.99), 1)},
    }, indent=2))
    if errs:
        print("sample errors:", errs[:3], file=sys.stderr)


if __name__ == "__main__":
    main()
