"""Tensor-parallel worker group: N GPUs as ONE logical backend.

Launched by ollamamq_amd.launch via torch.distributed.run (one process per
GPU over RCCL/xGMI).  Rank 0 owns the dispatcher socket and the request
surface; every rank runs an identical engine replica in lockstep: rank 0
broadcasts the op-stream (submissions / cancels / loads) each loop tick
and all ranks call engine.step() together, so the 2 all-reduces per layer
and the vocab-parallel all-gather line up by construction (the engine is
deterministic given the same op order; sequence ids and sampling
generators are seeded identically).

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


class TPWorker(worker_mod.Worker):
    """Rank-0 worker whose engine loop drives the whole TP group."""

    def __init__(self, rank: int, world: int, gpu_base: int,
                 max_batch: int = 32, default_ctx: int = 4096):
        self.rank = rank
        self.world = world
        self.gpu_base = gpu_base
        self._pending_ops = []
        self._ops_mu = threading.Lock()
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
            if self.world > 1:
                eng.use_graphs = False  # graphs+RCCL: enable once validated
            self.engines[model] = eng
            self.tokenizers[model] = ByteTokenizer(cfg.vocab)
            self.loaded_ctx[model] = ctx
            return None
        except torch.cuda.OutOfMemoryError:
            return "out of HBM: model + KV pool do not fit"

    # --- op-stream plumbing -------------------------------------------
    def load(self, model, num_ctx=0):
        # queued into the op stream so every rank allocates together
        return self._rpc(("load", model, num_ctx))

    def unload(self, model):
        return self._rpc(("unload", model))

    def generate(self, model, prompt_tokens, params, on_token):
        done = threading.Event()
        box = {}

        def record(result):
            box["r"] = result
            done.set()

        with self._ops_mu:
            self._pending_ops.append(
                (("submit", model, prompt_tokens, params), record, on_token))
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
            eng = self.engines.pop(op[1], None)
            self.tokenizers.pop(op[1], None)
            self.loaded_ctx.pop(op[1], None)
            if eng is None:
                return f"model not loaded: {op[1]}"
            del eng
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
            return None
        if kind == "submit":
            _, model, prompt, params = op
            if model not in self.engines:
                err = self._do_load(model, 0)
                if err:
                    return err
            return self.engines[model].submit(prompt, params, on_token)
        if kind == "cancel":
            eng = self.engines.get(op[1])
            if eng:
                eng.cancel(op[2])
            return None
        return f"bad op {kind}"

    def _engine_loop(self):
        while True:
            if self.rank == 0:
                with self._ops_mu:
                    batch = self._pending_ops
                    self._pending_ops = []
                ops = [b[0] for b in batch]
                with self.lock:
                    work = any(e.has_work() for e in self.engines.values())
                do_step = work or bool(ops)
                payload = [ops, do_step]
                if self.world > 1:
                    dist.broadcast_object_list(payload, src=0)
                results = []
                with self.lock:
                    for op, record, on_token in batch:
                        results.append((record,
                                        self._apply_op(op, on_token)))
                    if do_step:
                        for eng in list(self.engines.values()):
                            if eng.has_work():
                                eng.step()
                for record, r in results:
                    if record:
                        record(r)
                if not do_step:
                    self.work_ev.wait(timeout=0.02)
                    self.work_ev.clear()
            else:
                payload = [None, None]
                dist.broadcast_object_list(payload, src=0)
                ops, do_step = payload
                with self.lock:
                    for op in ops:
                        self._apply_op(op, None)
                    if do_step:
                        for eng in list(self.engines.values()):
                            if eng.has_work():
                                eng.step()


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
