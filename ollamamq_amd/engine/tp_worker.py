"""Tensor-parallel worker group: N GPUs as ONE logical backend.

Launched by ollamamq_amd.launch via torch.distributed.run (one process per
GPU over RCCL/xGMI).  Rank 0 owns the dispatcher socket and the request
surface; every rank runs an identical engine replica in lockstep, so the
2 all-reduces per layer and the vocab-parallel all-gather line up by
construction (the engine is deterministic given the same op order;
sequence ids and sampling generators are seeded identically).

Steady-state protocol (the part the reference has no analog for — its
"collective" is a tokio Notify inside one process, reference
src/dispatcher.rs:170-171): the op stream (submit/cancel/load/unload) is
synchronized with ONE 8-byte tensor broadcast per sync point, and sync
points happen every SYNC_EVERY engine steps — NOT every step.  Between
sync points all ranks free-run SYNC_EVERY lockstep decode steps with no
host-side exchange at all: because every rank is a replica, has_work()
and sequence finishes agree without communication.  The pickled
broadcast_object_list is only paid when ops actually arrived (rare — at
request cadence, not token cadence).

Decode steps run inside hipGraphs with the RCCL collectives captured
in-graph (engine.use_graphs stays ON at TP>1): each replay issues the
~7·n_layers kernels AND the 2·n_layers all-reduces from the graph,
eliminating per-step launch + collective-issue overhead.  Graph capture
happens in lockstep on all ranks (same op order ⇒ same capture step),
and the pre-capture eager warmups initialize the RCCL communicator
before capture begins (a capture-time communicator init is illegal).

CPU-testable with gloo (tests/test_tp_worker.py drives a 2-rank group).
"""
from __future__ import annotations

import argparse
import os
import threading
import time

import torch
import torch.distributed as dist

from ..models import LlamaModel, PRESETS
from .engine import GenParams, LlamaEngine
from .kvcache import PagedKVCache
from .tokenizer import ByteTokenizer
from . import worker as worker_mod

# Engine steps between op-stream sync points.  Ops arriving mid-window
# wait at most SYNC_EVERY steps (~45 ms at 5.6 ms/step) — request-level
# latency, invisible next to prefill.  Smaller values add one tiny
# broadcast per window; at 1 the protocol degenerates to per-step sync.
SYNC_EVERY = 8


class TPWorker(worker_mod.Worker):
    """Rank-0 worker whose engine loop drives the whole TP group."""

    def __init__(self, rank: int, world: int, gpu_base: int,
                 max_batch: int = 32, default_ctx: int = 4096):
        self.rank = rank
        self.world = world
        self.gpu_base = gpu_base
        self._pending_ops = []
        self._ops_mu = threading.Lock()
        # op-count sync word: device-resident under RCCL so the idle-tick
        # broadcast never touches host memory paths
        flag_dev = (f"cuda:{gpu_base + rank}"
                    if torch.cuda.is_available() else "cpu")
        self._flag = torch.zeros(1, dtype=torch.int64, device=flag_dev)
        super().__init__(gpu_base + rank, max_batch, default_ctx)

    # --- model lifecycle (applied on every rank via the op stream) -----
    def _do_load(self, model: str, num_ctx: int):
        cfg = PRESETS.get(model)
        if cfg is None:
            return f"unknown model: {model}"
        if cfg.n_heads % self.world or \
                (cfg.n_kv_heads % self.world and self.world > cfg.n_kv_heads):
            return f"model {model} heads not divisible by TP={self.world}"
        ctx = min(num_ctx or self.default_ctx, cfg.max_ctx)
        try:
            if torch.cuda.is_available():
                torch.cuda.set_device(self.gpu)
            m = LlamaModel(cfg, device=self.device, dtype=self.dtype,
                           seed=1234, tp_rank=self.rank, tp_size=self.world,
                           fast_init=self.device != "cpu")
            pages = (self.max_batch + 2) * ((ctx + 15) // 16 + 2)
            kv = PagedKVCache.for_model(
                cfg, tp_size=self.world, n_pages=pages,
                max_slots=self.max_batch + 2, max_ctx=ctx,
                device=self.device, dtype=self.dtype)
            eng = LlamaEngine(m, kv, max_batch=self.max_batch)
            # lockstep capture: every rank reaches warm_graphs at the same
            # op, so the in-graph RCCL collectives pair up across ranks
            eng.warm_graphs()
            self.engines[model] = eng
            self.tokenizers[model] = ByteTokenizer(cfg.vocab)
            self.loaded_ctx[model] = ctx
            return None
        except torch.cuda.OutOfMemoryError:
            return "out of HBM: model + KV pool do not fit"

    # --- op-stream plumbing (rank-0 request surface) -------------------
    # Signatures MUST match worker.Worker: Conn._request calls
    # generate(..., num_ctx=...) and unload(..., only_if_idle=True)
    # (worker.py:342-345,380).
    def load(self, model, num_ctx=0):
        # queued into the op stream so every rank allocates together
        return self._rpc(("load", model, num_ctx))

    def unload(self, model, only_if_idle=False):
        return self._rpc(("unload", model, only_if_idle))

    def generate(self, model, prompt_tokens, params, on_token, num_ctx=0):
        done = threading.Event()
        box = {}

        def record(result):
            box["r"] = result
            done.set()

        with self._ops_mu:
            self._pending_ops.append(
                (("submit", model, prompt_tokens, params, num_ctx),
                 record, on_token))
        self.work_ev.set()
        done.wait(timeout=900)
        r = box.get("r")
        if isinstance(r, str):
            raise RuntimeError(r)
        return r

    def cancel(self, model, sid):
        with self._ops_mu:
            self._pending_ops.append((("cancel", model, sid), None, None))
        self.work_ev.set()

    def _rpc(self, op):
        done = threading.Event()
        box = {}

        def record(result):
            box["r"] = result
            done.set()

        with self._ops_mu:
            self._pending_ops.append((op, record, None))
        self.work_ev.set()
        done.wait(timeout=900)
        return box.get("r")

    # --- lockstep engine loop -----------------------------------------
    def _apply_op(self, op, on_token):
        kind = op[0]
        if kind == "load":
            return self._do_load(op[1], op[2])
        if kind == "unload":
            model, only_if_idle = op[1], op[2]
            eng = self.engines.get(model)
            if eng is None:
                return f"model not loaded: {model}"
            if only_if_idle and eng.has_work():
                return "model busy"
            self.engines.pop(model, None)
            self.tokenizers.pop(model, None)
            self.loaded_ctx.pop(model, None)
            del eng
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
            return None
        if kind == "submit":
            _, model, prompt, params, num_ctx = op
            if model not in self.engines:
                err = self._do_load(model, num_ctx)
                if err:
                    return err
            return self.engines[model].submit(prompt, params, on_token)
        if kind == "cancel":
            eng = self.engines.get(op[1])
            if eng:
                eng.cancel(op[2])
            return None
        return f"bad op {kind}"

    def _run_steps(self, budget: int) -> bool:
        """Free-run up to `budget` lockstep engine steps (no collectives
        beyond the ones inside forward).  All ranks take the same number
        of steps because the engines are replicas — has_work() and
        finishes agree by determinism, not by communication."""
        did = False
        for _ in range(budget):
            stepped = False
            for eng in list(self.engines.values()):
                if eng.has_work():
                    eng.step()
                    stepped = True
            if not stepped:
                break
            did = True
        return did

    def _engine_loop(self):
        while True:
            if self.rank == 0:
                with self._ops_mu:
                    batch = self._pending_ops
                    self._pending_ops = []
                ops = [b[0] for b in batch]
                if self.world > 1:
                    self._flag[0] = len(ops)
                    dist.broadcast(self._flag, src=0)
                    if ops:
                        dist.broadcast_object_list([ops], src=0)
                results = []
                with self.lock:
                    for op, record, on_token in batch:
                        results.append((record,
                                        self._apply_op(op, on_token)))
                    did = self._run_steps(SYNC_EVERY)
                for record, r in results:
                    if record:
                        record(r)
                if not did and not ops:
                    self.work_ev.wait(timeout=0.02)
                    self.work_ev.clear()
            else:
                dist.broadcast(self._flag, src=0)
                n = int(self._flag.item())
                ops = []
                if n:
                    payload = [None]
                    dist.broadcast_object_list(payload, src=0)
                    ops = payload[0]
                with self.lock:
                    for op in ops:
                        self._apply_op(op, None)
                    self._run_steps(SYNC_EVERY)


def follower_loop(w: TPWorker):
    # ranks > 0 just run the engine loop thread forever
    while True:
        time.sleep(3600)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--socket", required=True)
    ap.add_argument("--gpu-base", type=int, default=0)
    ap.add_argument("--model", type=str, default=None)
    ap.add_argument("--max-ctx", type=int, default=4096)
    ap.add_argument("--max-batch", type=int, default=32)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        # bind the device BEFORE the RCCL communicator is created
        torch.cuda.set_device(args.gpu_base + rank)
    if world > 1:
        dist.init_process_group(backend)

    w = TPWorker(rank, world, args.gpu_base,
                 max_batch=args.max_batch, default_ctx=args.max_ctx)
    if args.model and rank == 0:
        err = w.load(args.model, args.max_ctx)
        if err:
            raise SystemExit(f"preload failed: {err}")
    if rank == 0:
        worker_mod.serve(args.socket, w)
    else:
        follower_loop(w)


if __name__ == "__main__":
    main()
