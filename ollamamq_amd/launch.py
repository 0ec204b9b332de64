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
    ap.add_argument("--no-tui", action="store_true")
    ap.add_argument("--sock-dir", type=str, default="/tmp")
    ap.add_argument("-c", "--model-config", type=str, default="appconf.yaml")
    ap.add_argument("--extra-backends", type=str, default="",
                    help="comma list of external HTTP backends to add")
    args = ap.parse_args()

    if not os.path.exists(SERVER_BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)

    procs = []
    socks = []
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")

    assert args.gpus % args.tp == 0, "--gpus must be a multiple of --tp"
    n_backends = args.gpus // args.tp
    for b in range(n_backends):
        sock = os.path.join(args.sock_dir, f"omq_worker{b}.sock")
        try:
            os.unlink(sock)
        except FileNotFoundError:
            pass
        socks.append(sock)
        if args.tp == 1:
            cmd = [sys.executable, "-m", "ollamamq_amd.engine.worker",
                   "--socket", sock, "--gpu", str(b),
                   "--model", args.model, "--max-ctx", str(args.max_ctx),
                   "--max-batch", str(args.max_batch)]
            procs.append(subprocess.Popen(cmd, env=env))
        else:
            gpus = range(b * args.tp, (b + 1) * args.tp)
            cmd = [sys.executable, "-m", "torch.distributed.run",
                   "--nnodes=1", f"--nproc-per-node={args.tp}",
                   "--master-addr", "127.0.0.1",
                   "--master-port", str(29600 + b),
                   "-m", "ollamamq_amd.engine.tp_worker",
                   "--socket", sock, "--gpu-base", str(min(gpus)),
                   "--model", args.model, "--max-ctx", str(args.max_ctx),
                   "--max-batch", str(args.max_batch)]
            procs.append(subprocess.Popen(cmd, env=env))

    print(f"waiting for {n_backends} worker socket(s)...", flush=True)
    if not wait_sockets(socks):
        for p in procs:
            p.terminate()
        raise SystemExit("workers failed to come up")

    server_cmd = [SERVER_BIN, "-p", str(args.port), "-H", args.host,
                  "-w", ",".join(socks), "-c", args.model_config]
    if args.extra_backends:
        server_cmd += ["-o", args.extra_backends]
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
