"""Node launcher: one command brings up the whole serving node.

    python -m ollamamq_amd.launch --gpus 8 --model llama3-8b --port 11435

spawns one engine worker process per MI355X (the torch.distributed process
model: process-per-GPU), waits for their sockets, then runs the native C++
dispatcher (ollamamq-server) with every worker attached as an in-process
backend.  Equivalent role to the reference's `ollama-mq` binary + external
Ollama servers — except the "servers" are this framework's own GPU engines.

TP mode (`--tp N --model llama3-70b`): the N GPUs form ONE logical backend;
rank 0 owns the socket, ranks 1..N-1 join the RCCL group and follow rank 0's
engine steps (see engine/tp_worker.py).
"""
from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time

HERE = os.path.dirname(os.path.abspath(__file__))
SERVER_BIN = os.path.join(HERE, "csrc", "dispatcher", "ollamamq-server")


def wait_sockets(paths, timeout=600):
    t0 = time.time()
    missing = set(paths)
    while missing and time.time() - t0 < timeout:
        for p in list(missing):
            if os.path.exists(p):
                missing.discard(p)
        time.sleep(0.5)
    return not missing


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--tp", type=int, default=1,
                    help="GPUs per logical backend (tensor parallel)")
    ap.add_argument("--model", type=str, default="llama3-8b")
    ap.add_argument("--max-ctx", type=int, default=4096)
    # 288 GB HBM3E leaves KV room far beyond batch 32; measured
    # continuous-batching throughput: 32/64/128 users = 5.5k/9.5k/13.2k
    # tok/s on one MI355X — 64 is the serving default
    ap.add_argument("--max-batch", type=int, default=64)
    ap.add_argument("--port", type=int, default=11435)
    ap.add_argument("--host", type=str, default="127.0.0.1")
    ap.add_argument("-t", "--timeout", type=int, default=0,
                    help="request timeout seconds (dispatcher -t)")
    ap.add_argument("--no-tui", action="store_true")
    ap.add_argument("--sock-dir", type=str, default="/tmp")
    ap.add_argument("-c", "--model-config", type=str, default="appconf.yaml")
    ap.add_argument("--extra-backends", type=str, default="",
                    help="comma list of external HTTP backends to add")
    ap.add_argument("--workers", type=str, default="",
                    help="heterogeneous fleet spec overriding "
                         "--gpus/--tp/--model: comma list of "
                         "MODEL[*COUNT][/tpN], e.g. "
                         "'llama3-8b*4,llama3-70b/tp4' = four 1-GPU 8B "
                         "backends + one TP4 70B backend (BASELINE.json "
                         "config 5).  Each worker advertises ONLY its "
                         "model, so /api/tags routing steers requests "
                         "(reference src/dispatcher.rs:599-620).")
    args = ap.parse_args()

    if not os.path.exists(SERVER_BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)

    procs = []
    socks = []
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")

    # Build the backend plan: homogeneous (--gpus/--tp/--model) or a
    # heterogeneous --workers spec; each entry = (model, tp_degree).
    plan = []
    if args.workers:
        for item in args.workers.split(","):
            item = item.strip()
            if not item:
                continue
            count, tp = 1, 1
            if "/tp" in item:
                item, tp_s = item.rsplit("/tp", 1)
                tp = int(tp_s)
            if "*" in item:
                item, cnt_s = item.rsplit("*", 1)
                count = int(cnt_s)
            plan.extend([(item.strip(), tp)] * count)
    else:
        assert args.gpus % args.tp == 0, "--gpus must be a multiple of --tp"
        plan = [(args.model, args.tp)] * (args.gpus // args.tp)

    gpu_base = 0
    for b, (model, tp) in enumerate(plan):
        sock = os.path.join(args.sock_dir, f"omq_worker{b}.sock")
        try:
            os.unlink(sock)
        except FileNotFoundError:
            pass
        socks.append(sock)
        if tp == 1:
            cmd = [sys.executable, "-m", "ollamamq_amd.engine.worker",
                   "--socket", sock, "--gpu", str(gpu_base),
                   "--model", model, "--max-ctx", str(args.max_ctx),
                   "--max-batch", str(args.max_batch)]
            if args.workers:
                cmd += ["--models", model]
            procs.append(subprocess.Popen(cmd, env=env))
        else:
            cmd = [sys.executable, "-m", "torch.distributed.run",
                   "--nnodes=1", f"--nproc-per-node={tp}",
                   "--master-addr", "127.0.0.1",
                   "--master-port", str(29600 + b),
                   "-m", "ollamamq_amd.engine.tp_worker",
                   "--socket", sock, "--gpu-base", str(gpu_base),
                   "--model", model, "--max-ctx", str(args.max_ctx),
                   "--max-batch", str(args.max_batch)]
            # RCCL small-message tuning for the TP decode all-reduces
            # (16-64 KiB latency-bound messages over the
            # 7-link point-to-point xGMI mesh).  OLLAMAMQ_RCCL_ALGO /
            # _PROTO map onto NCCL_ALGO / NCCL_PROTO for the worker
            # group only; unset, RCCL's own size-based tuning applies
            # (forcing LL globally would hurt the prefill-size
            # all-reduces, so there is no hard default here).
            tenv = dict(env)
            for src, dst in (("OLLAMAMQ_RCCL_ALGO", "NCCL_ALGO"),
                             ("OLLAMAMQ_RCCL_PROTO", "NCCL_PROTO"),
                             ("OLLAMAMQ_RCCL_NCHANNELS",
                              "NCCL_MIN_NCHANNELS")):
                if os.environ.get(src):
                    tenv.setdefault(dst, os.environ[src])
            procs.append(subprocess.Popen(cmd, env=tenv))
        gpu_base += tp
    n_backends = len(plan)

    print(f"waiting for {n_backends} worker socket(s)...", flush=True)
    if not wait_sockets(socks):
        for p in procs:
            p.terminate()
        raise SystemExit("workers failed to come up")

    server_cmd = [SERVER_BIN, "-p", str(args.port), "-H", args.host,
                  "-w", ",".join(socks), "-c", args.model_config]
    if args.extra_backends:
        server_cmd += ["-o", args.extra_backends]
    if args.timeout > 0:
        server_cmd += ["-t", str(args.timeout)]
    if args.no_tui:
        server_cmd.append("--no-tui")
    server = subprocess.Popen(server_cmd)
    procs.append(server)

    def shutdown(*_):
        for p in procs:
            p.terminate()
        sys.exit(0)

    signal.signal(signal.SIGINT, shutdown)
    signal.signal(signal.SIGTERM, shutdown)
    server.wait()
    shutdown()


if __name__ == "__main__":
    main()
