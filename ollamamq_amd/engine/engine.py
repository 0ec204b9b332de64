"""Continuous-batching inference engine for one (logical) MI355X backend.

This is the compute side of what the reference merely proxies: where
ollamaMQ forwards a request over HTTP and pumps bytes back
(reference src/dispatcher.rs:742-778), this engine admits the request into a
resident batch, runs prefill, then emits one token per decode step from the
hand-written HIP path, streaming tokens through a callback that the server
layer turns into Ollama JSON-lines / OpenAI SSE chunks.

Scheduling inside the engine: new sequences are prefilled as a varlen batch
(chunked), then join the decode batch; one decode step advances every
running sequence by one token (the reference's one-request-per-backend cap,
src/dispatcher.rs:589, is the *scheduler's* default; the engine itself
batches whatever it is given).
"""
from __future__ import annotations

import itertools
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

import torch

from ..ops.interface import AttnMeta
from ..ops import interface as ops
from .kvcache import PagedKVCache


@dataclass
class GenParams:
    max_tokens: int = 128
    temperature: float = 0.0
    top_k: int = 0
    top_p: float = 1.0
    stop_token: Optional[int] = None
    seed: Optional[int] = None


@dataclass
class Sequence:
    seq_id: int
    prompt: List[int]
    params: GenParams
    on_token: Optional[Callable[[int, bool], None]] = None
    slot: int = -1
    generated: List[int] = field(default_factory=list)
    state: str = "waiting"            # waiting -> prefill -> running -> done
    prefill_done: int = 0             # tokens of prompt already prefilled
    submitted_at: float = field(default_factory=time.monotonic)
    first_token_at: Optional[float] = None
    finished_at: Optional[float] = None
    finish_reason: Optional[str] = None
    cancelled: bool = False
    _chunk: int = 0                   # prompt tokens taken this step

    @property
    def total_len(self) -> int:
        return len(self.prompt) + len(self.generated)


class LlamaEngine:
    def __init__(
        self,
        model,
        kv_cache: PagedKVCache,
        max_batch: int = 64,
        prefill_chunk: int = 2048,
    ):
        self.model = model
        self.kv = kv_cache
        self.max_batch = max_batch
        self.prefill_chunk = prefill_chunk
        self.waiting: List[Sequence] = []
        self.running: List[Sequence] = []
        self.seqs: Dict[int, Sequence] = {}
        self._ids = itertools.count(1)
        self.dev = model.device
        self._gen = None
        if self.dev.type == "cuda":
            self._gen = torch.Generator(device=self.dev)
            self._gen.manual_seed(0xC0FFEE)
        self.steps = 0
        self.tokens_out = 0
        # hipGraph-captured decode steps (one graph per exact batch size,
        # captured lazily; VRAM cost ~75 MB/size on 8B, bounded by
        # max_batch — measured stable across a 6-min soak after capture):
        # decode is ~7 kernels × n_layers of launches; replay collapses the
        # launch gaps (MI355X_MICROARCH "launches-baseline": ≈1.2 µs/boundary)
        import os as _os
        self.use_graphs = (self.dev.type == "cuda"
                           and _os.environ.get("OLLAMAMQ_NO_GRAPH") != "1")
        self._graphs: Dict[tuple, dict] = {}   # (batch bucket, class)
        # Pipelined decode: the graph samples (greedy argmax) and advances
        # tok/pos/lens on-device, so steady-state steps skip the H2D input
        # fill AND the host token sync — the host consumes tokens one step
        # late through a pinned-buffer/event ring while the GPU runs the
        # next step.  Drained on any batch-composition change.
        self.use_pipeline = (self.use_graphs
                             and _os.environ.get("OLLAMAMQ_NO_PIPELINE")
                             != "1")
        self._pipe = None   # {"key": seq-id tuple, "inflight": (ev,buf,seqs)}
        # Decode batches are PADDED UP to a bucket size so the graph set
        # is small and fixed: lazy mid-serving captures cost ~75 MB and
        # ~150 ms each (the r02 soak caught a 328 MB tail drift from
        # stragglers).  Padding is nearly free — the step is
        # weight-streaming-bound, pad rows attend over a 1-token dummy
        # slot and their lens/pos are advance-masked in-graph.
        self._buckets = sorted({b for b in
                                (1, 2, 3, 4, 6, 8, 12, 16, 24, 32,
                                 48, 64, 96, 128, max_batch)
                                if b <= max_batch})
        self._dummy_slot = -1
        # opt-in: capture the top-k/top-p tail in-graph (see _samp_class)
        self._GRAPH_TOPK = _os.environ.get("OLLAMAMQ_GRAPH_TOPK") == "1"

    # -- submission --------------------------------------------------------
    def submit(self, prompt: List[int], params: GenParams,
               on_token=None) -> int:
        if len(prompt) == 0:
            prompt = [0]
        # Ollama-parity: a prompt longer than the context window is
        # truncated from the front (the tail conditions generation) so the
        # request degrades instead of faulting the batch at prefill
        limit = max(1, self.kv.max_ctx - 1)
        if len(prompt) > limit:
            prompt = prompt[-limit:]
        sid = next(self._ids)
        seq = Sequence(sid, list(prompt), params, on_token)
        # per-sequence noise seed for the in-graph Gumbel sampler:
        # request seed (reproducible regardless of batch composition,
        # the counter is the sequence's own length) or a per-sequence
        # engine default (deterministic across TP ranks: sid matches)
        seq.noise_seed = (int(params.seed) if params.seed is not None
                          else (0x5EED0000 ^ (sid * 0x9E3779B1)) &
                               0x7FFFFFFFFFFFFFFF)
        self.seqs[sid] = seq
        self.waiting.append(seq)
        return sid

    def cancel(self, seq_id: int) -> None:
        seq = self.seqs.get(seq_id)
        if seq is not None and seq.state != "done":
            seq.cancelled = True

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def n_active(self) -> int:
        return len(self.waiting) + len(self.running)

    # -- stepping ----------------------------------------------------------
    def step(self) -> List[Sequence]:
        """Advance the engine by one iteration; returns sequences finished.

        When both new prompts and running sequences exist, prefill and
        decode steps alternate so a long prompt cannot starve the decode
        batch (inter-token latency stays bounded by ~one prefill chunk).
        """
        self._reap_cancelled()
        admitted = self._admit() if self.waiting else []
        if admitted and self.running:
            # mixed batch: every running sequence decodes one token AND
            # the new prompts prefill a chunk in the SAME forward — one
            # weight pass serves both, so admission never stalls decode
            self._mixed_step(admitted)
            # admitted seqs that finished prefill were moved into
            # self.running by _mixed_step — don't postprocess them twice
            # (a double _finish would emit the done marker twice)
            finished = self._postprocess(
                self.running + [s for s in admitted
                                if s not in self.running])
        elif admitted:
            self._prefill_step(admitted)
            finished = self._postprocess(admitted)
        elif self.running:
            self._decode_step()
            finished = self._postprocess(self.running)
        else:
            return []
        self.steps += 1
        for seq in finished:
            self._finish(seq)
        return finished

    def embed(self, prompt: List[int]):
        """Synchronous embedding: final-norm hidden state of the last
        prompt token (no generation, KV slot freed immediately)."""
        if not prompt:
            prompt = [0]
        slot = self.kv.alloc_slot()
        try:
            self.kv.ensure(slot, len(prompt))
            dev = self.dev
            tokens = torch.tensor(prompt, dtype=torch.int32, device=dev)
            positions = torch.arange(len(prompt), dtype=torch.int32,
                                     device=dev)
            slot_t = torch.full((len(prompt),), slot, dtype=torch.int32,
                                device=dev)
            meta = AttnMeta(
                mode="prefill",
                slot_ids=torch.tensor([slot], dtype=torch.int32, device=dev),
                seq_lens=torch.tensor([len(prompt)], dtype=torch.int32,
                                      device=dev),
                cu_q=torch.tensor([0, len(prompt)], dtype=torch.int32,
                                  device=dev),
                logits_idx=torch.tensor([len(prompt) - 1], dtype=torch.long,
                                        device=dev),
                max_q=len(prompt), max_kv=len(prompt),
                window=getattr(self.model.cfg, "sliding_window", 0))
            h = self.model.forward(tokens, positions, self.kv, slot_t, meta,
                                   return_hidden=True)
            return h[0].float().tolist()
        finally:
            self.kv.free_slot(slot)

    # -- internals ---------------------------------------------------------
    def _reap_cancelled(self):
        if any(s.cancelled for s in self.running):
            self._drain_pipe()
        for lst in (self.waiting, self.running):
            for seq in [s for s in lst if s.cancelled]:
                lst.remove(seq)
                seq.finish_reason = "cancelled"
                # emit the done marker: stream consumers (the worker's
                # token queue) must unblock on cancellation too
                self._finish(seq, emit=True)

    def _admit(self) -> List[Sequence]:
        """Move waiting sequences into a prefill batch (chunked varlen)."""
        batch: List[Sequence] = []
        budget = self.prefill_chunk
        room = self.max_batch - len(self.running)
        for seq in list(self.waiting):
            if room <= 0 or budget <= 0:
                break
            todo = len(seq.prompt) - seq.prefill_done
            take = min(todo, budget)
            if take <= 0:
                break
            new_len = seq.prefill_done + take
            if not self.kv.can_fit(new_len - self.kv.seq_lens[seq.slot]
                                   if seq.slot >= 0 else new_len):
                break
            if seq.slot < 0:
                try:
                    seq.slot = self.kv.alloc_slot()
                except RuntimeError:
                    break
            seq.state = "prefill"
            seq._chunk = take  # tokens of prompt this step
            batch.append(seq)
            budget -= take
            if new_len == len(seq.prompt):
                room -= 1
        return batch

    def _mixed_step(self, batch: List[Sequence]):
        self._drain_pipe()   # decode rows need current host tokens
        # context-full sequences can't take another token; skip them this
        # step — _postprocess finishes them with reason "length"
        dec = [s for s in self.running
               if s.total_len < self.kv.max_ctx]
        token_list = [s.generated[-1] for s in dec]
        pos_list = [s.total_len - 1 for s in dec]
        q_lens = [1] * len(dec)
        for s in batch:
            take = s._chunk
            start = s.prefill_done
            token_list.extend(s.prompt[start:start + take])
            pos_list.extend(range(start, start + take))
            q_lens.append(take)
        logits = self._forward(dec + batch, token_list, pos_list, q_lens,
                               "mixed", n_decode=len(dec))
        # sample decode rows + finishing-prefill rows
        fin_rows = [len(dec) + i for i, s in enumerate(batch)
                    if s.prefill_done + s._chunk == len(s.prompt)]
        done_prefill = [batch[i - len(dec)] for i in fin_rows]
        sample_rows = list(range(len(dec))) + fin_rows
        toks = self._sample(dec + done_prefill, logits[sample_rows])
        now = time.monotonic()
        for s, tok in zip(dec, toks[:len(dec)]):
            s.generated.append(int(tok))
            self.tokens_out += 1
            if s.on_token:
                s.on_token(int(tok), False)
        ptoks = toks[len(dec):]
        for s in batch:
            s.prefill_done += s._chunk
            if s.prefill_done == len(s.prompt):
                s.first_token_at = now
                s.state = "running"
                self.waiting.remove(s)
                self.running.append(s)
                if s.params.max_tokens <= 0:
                    # zero-token request (Ollama num_predict: 0): admit
                    # without emitting; _postprocess finishes it this step
                    continue
                tok = ptoks[done_prefill.index(s)]
                s.generated.append(int(tok))
                self.tokens_out += 1
                if s.on_token:
                    s.on_token(int(tok), False)

    def _build_meta(self, seqs, q_lens, mode, n_decode=0) -> AttnMeta:
        dev = self.dev
        slot_ids = torch.tensor([s.slot for s in seqs], dtype=torch.int32,
                                device=dev)
        seq_lens = torch.tensor([self.kv.seq_lens[s.slot] for s in seqs],
                                dtype=torch.int32, device=dev)
        cu = [0]
        for ql in q_lens:
            cu.append(cu[-1] + ql)
        cu_q = torch.tensor(cu, dtype=torch.int32, device=dev)
        logits_idx = torch.tensor([c - 1 for c in cu[1:]], dtype=torch.long,
                                  device=dev)
        return AttnMeta(
            mode=mode, slot_ids=slot_ids, seq_lens=seq_lens, cu_q=cu_q,
            logits_idx=logits_idx, max_q=max(q_lens),
            max_kv=int(max(self.kv.seq_lens[s.slot] for s in seqs)),
            n_decode=n_decode,
            window=getattr(self.model.cfg, "sliding_window", 0),
        )

    def _forward(self, seqs, token_list, pos_list, q_lens, mode,
                 n_decode=0):
        dev = self.dev
        tokens = torch.tensor(token_list, dtype=torch.int32, device=dev)
        positions = torch.tensor(pos_list, dtype=torch.int32, device=dev)
        slot_per_tok = torch.tensor(
            [s.slot for s, ql in zip(seqs, q_lens) for _ in range(ql)],
            dtype=torch.int32, device=dev)
        for s, ql in zip(seqs, q_lens):
            self.kv.ensure(s.slot, self.kv.seq_lens[s.slot] + ql)
        meta = self._build_meta(seqs, q_lens, mode, n_decode)
        logits = self.model.forward(tokens, positions, self.kv, slot_per_tok,
                                    meta)
        return logits

    def _sample(self, seqs, logits):
        dev = logits.device
        temps = torch.tensor([s.params.temperature for s in seqs],
                             dtype=torch.float32, device=dev)
        # per-request top_k/top_p: each row is filtered by its own params
        # (vectorized in ops.sample — mixed batches never bleed)
        top_k = torch.tensor([s.params.top_k or 0 for s in seqs],
                             dtype=torch.long, device=dev)
        top_p = torch.tensor(
            [s.params.top_p if s.params.top_p else 1.0 for s in seqs],
            dtype=torch.float32, device=dev)
        toks = ops.sample(logits, temps, top_k, top_p, self._gen)
        out = toks.tolist()
        # per-request `seed` (Ollama options.seed): stochastic rows with a
        # seed re-sample alone under a generator keyed (seed, step-index)
        # so the same request reproduces regardless of batch composition
        for i, s in enumerate(seqs):
            p = s.params
            if p.seed is not None and p.temperature > 0:
                g = torch.Generator(device=logits.device)
                g.manual_seed((int(p.seed) << 20)
                              ^ (len(s.generated) + len(s.prompt)))
                t = ops.sample(logits[i:i + 1],
                               torch.tensor([p.temperature],
                                            dtype=torch.float32,
                                            device=logits.device),
                               p.top_k or 0, p.top_p or 1.0, g)
                out[i] = int(t[0])
        return out

    def _prefill_step(self, batch: List[Sequence]):
        self._drain_pipe()
        token_list, pos_list, q_lens = [], [], []
        for s in batch:
            take = s._chunk
            start = s.prefill_done
            token_list.extend(s.prompt[start:start + take])
            pos_list.extend(range(start, start + take))
            q_lens.append(take)
        logits = self._forward(batch, token_list, pos_list, q_lens, "prefill")
        # Only sequences whose whole prompt is now in KV emit a token.
        rows = [i for i, s in enumerate(batch)
                if s.prefill_done + s._chunk == len(s.prompt)]
        done_prefill = [batch[i] for i in rows]
        toks: List[int] = []
        if done_prefill:
            sub = logits if len(rows) == len(batch) else logits[rows]
            toks = self._sample(done_prefill, sub)
        for s in batch:
            s.prefill_done += s._chunk
            if s.prefill_done == len(s.prompt):
                s.first_token_at = time.monotonic()
                s.state = "running"
                self.waiting.remove(s)
                self.running.append(s)
                if s.params.max_tokens <= 0:
                    # zero-token request (Ollama num_predict: 0): admit
                    # without emitting; _postprocess finishes it this step
                    continue
                tok = toks[done_prefill.index(s)]
                s.generated.append(int(tok))
                self.tokens_out += 1
                if s.on_token:
                    s.on_token(int(tok), False)

    def _samp_class(self, seqs) -> int:
        """0 = all greedy (argmax tail), 1 = temperature-only mix (Gumbel
        tail — exact), 2 = top-k/top-p present.

        Class 2 is captured ONLY behind OLLAMAMQ_GRAPH_TOPK=1: the
        candidate tail (torch.topk over [B, 128256] + filters) is clean
        in isolation (tools/repro_tail.py: 3000 mutated replays) and on
        tiny models, but faults under real multi-graph 8B serving with
        or without the fused chain (tools/repro_sampled.py WORKLOAD=topk)
        — an unresolved interaction between the captured topk and the
        serving graph set.  Until root-caused, top-k/p rows sample on the
        host (exact, full vocabulary) with the forward still graphed."""
        k = 0
        for s in seqs:
            p = s.params
            if p.temperature > 0:
                if (p.top_k or 0) > 0 or (p.top_p if p.top_p else 1.0) \
                        < 1.0:
                    return 2 if self._GRAPH_TOPK else 3
                k = 1
        return k

    def _bucket(self, n: int) -> int:
        for b in self._buckets:
            if b >= n:
                return b
        return self._buckets[-1]

    def _ensure_dummy(self) -> int:
        if self._dummy_slot < 0:
            self._dummy_slot = self.kv.alloc_slot()
            self.kv.ensure(self._dummy_slot, 1)
        return self._dummy_slot

    def _graph_entry(self, B: int, klass: int = 0):
        """Capture (once per (batch size, sampling class)) a full decode
        forward as a hipGraph reading its inputs from static device
        buffers."""
        entry = self._graphs.get((B, klass))
        if entry is not None:
            return entry
        dev = self.dev
        bufs = {
            "tok": torch.zeros(B, dtype=torch.int32, device=dev),
            "pos": torch.zeros(B, dtype=torch.int32, device=dev),
            "slot": torch.zeros(B, dtype=torch.int32, device=dev),
            "lens": torch.zeros(B, dtype=torch.int32, device=dev),
            "temps": torch.zeros(B, dtype=torch.float32, device=dev),
            "seeds": torch.zeros(B, dtype=torch.int64, device=dev),
            "topk": torch.zeros(B, dtype=torch.int64, device=dev),
            "topp": torch.ones(B, dtype=torch.float32, device=dev),
            # 1 for live rows, 0 for bucket padding: the in-graph
            # advance applies it so pad rows never grow their dummy slot
            "adv": torch.ones(B, dtype=torch.int32, device=dev),
        }
        meta = AttnMeta(
            mode="decode", slot_ids=bufs["slot"], seq_lens=bufs["lens"],
            cu_q=torch.arange(B + 1, dtype=torch.int32, device=dev),
            logits_idx=torch.arange(B, dtype=torch.long, device=dev),
            max_q=1, max_kv=self.kv.max_ctx,
            window=getattr(self.model.cfg, "sliding_window", 0))
        entry = {"bufs": bufs, "meta": meta, "graph": None, "logits": None,
                 "out": None, "ring_i": 0, "klass": klass,
                 "pinned": [torch.empty(B, dtype=torch.int64,
                                        pin_memory=True) for _ in range(2)],
                 "events": [torch.cuda.Event() for _ in range(2)]}
        self._graphs[(B, klass)] = entry
        return entry

    def _fill_bufs(self, entry, seqs, token_list, pos_list):
        bufs = entry["bufs"]
        B = bufs["tok"].shape[0]
        pad = B - len(seqs)
        dummy = self._ensure_dummy() if pad else 0
        host = torch.tensor(
            [token_list + [0] * pad,
             pos_list + [0] * pad,
             [s.slot for s in seqs] + [dummy] * pad,
             [self.kv.seq_lens[s.slot] for s in seqs] + [1] * pad,
             [1] * len(seqs) + [0] * pad], dtype=torch.int32)
        staged = host.to(self.dev, non_blocking=True)
        bufs["tok"].copy_(staged[0])
        bufs["pos"].copy_(staged[1])
        bufs["slot"].copy_(staged[2])
        bufs["lens"].copy_(staged[3])
        bufs["adv"].copy_(staged[4])
        if entry["klass"] >= 1:
            bufs["temps"].copy_(torch.tensor(
                [s.params.temperature for s in seqs] + [0.0] * pad,
                dtype=torch.float32).to(self.dev, non_blocking=True))
            bufs["seeds"].copy_(torch.tensor(
                [getattr(s, "noise_seed", 1234) for s in seqs]
                + [0] * pad,
                dtype=torch.int64).to(self.dev, non_blocking=True))
        if entry["klass"] == 2:
            bufs["topk"].copy_(torch.tensor(
                [s.params.top_k or 0 for s in seqs] + [0] * pad,
                dtype=torch.int64).to(self.dev, non_blocking=True))
            bufs["topp"].copy_(torch.tensor(
                [s.params.top_p if s.params.top_p else 1.0
                 for s in seqs] + [1.0] * pad,
                dtype=torch.float32).to(self.dev, non_blocking=True))

    def _graph_tail(self, entry, logits):
        """The in-graph sampling tail: class 0 = greedy argmax; class 1 =
        exact Gumbel-max temperature sampling (kernels.hip sample_gumbel,
        greedy rows ride the same kernel with T<=0), noise keyed by
        (per-seq seed, kv length) so replays draw fresh randomness with
        no host RNG; class 2 = per-row top-k/top-p over the top-256
        candidates (capturable torch ops), then exact Gumbel-max on the
        filtered set.

        This MUST also run once EAGERLY in the pre-capture warmup: the
        samplers cache split-scratch buffers (ops/hip.py _scratch), and a
        first allocation INSIDE capture would live in one graph's private
        memory pool while every other graph and the eager path reuse the
        cached tensor — undefined, and it produced real memory faults
        under mixed sampled serving (r02 stress)."""
        bufs = entry["bufs"]
        if entry["klass"] == 1:
            from ..ops import hip as _hip
            return _hip.sample_gumbel(logits, bufs["temps"],
                                      bufs["seeds"], bufs["lens"])
        if entry["klass"] == 2:
            from ..ops import hip as _hip
            C = min(256, logits.shape[1])
            v, idx = torch.topk(logits.float(), C, dim=-1)
            t = bufs["temps"].clamp(min=1e-6).unsqueeze(1)
            p = torch.softmax(v / t, dim=-1)
            ar = torch.arange(C, device=logits.device)
            kk = torch.where(bufs["topk"] > 0,
                             bufs["topk"].clamp(max=C),
                             torch.full_like(bufs["topk"], C))
            keep = ar.unsqueeze(0) < kk.unsqueeze(1)
            cum = p.cumsum(dim=-1)
            keep &= (cum - p) < bufs["topp"].unsqueeze(1)
            keep[:, 0] = True
            # greedy rows bypass the (garbage) filter math
            keep |= (bufs["temps"] <= 0).unsqueeze(1)
            vm = v.masked_fill(~keep, float("-inf")).bfloat16()
            ci = _hip.sample_gumbel(vm.contiguous(), bufs["temps"],
                                    bufs["seeds"], bufs["lens"])
            return idx.gather(1, ci.long().unsqueeze(1)).squeeze(1).int()
        return ops.sample(logits, 0.0, 0, 1.0, None).int()

    def _graph_replay(self, entry):
        bufs, meta = entry["bufs"], entry["meta"]
        if entry["graph"] is None:
            # warmup twice on a side stream, then capture.  The Python GC
            # must not run DURING capture: collecting a dead CUDA tensor
            # issues a hipFree, which is illegal while a stream is
            # capturing and aborts the process (observed under the full
            # GPU suite) — flush pending frees first, then hold GC off.
            import gc
            gc.collect()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    warm_logits = self.model.forward(
                        bufs["tok"], bufs["pos"], self.kv, bufs["slot"],
                        meta)
                    # eager tail warmup — see _graph_tail docstring
                    self._graph_tail(entry, warm_logits)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            gc.disable()
            try:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    entry["logits"] = self.model.forward(
                        bufs["tok"], bufs["pos"], self.kv, bufs["slot"],
                        meta)
                    # self-advancing tail: sample in-graph and stage the
                    # NEXT step's inputs on-device, so a steady decode
                    # batch replays back-to-back with no host round-trip
                    toks32 = self._graph_tail(entry, entry["logits"])
                    toks = toks32.long()
                    bufs["tok"].copy_(toks32)
                    entry["out"] = toks
                    bufs["pos"] += bufs["adv"]
                    bufs["lens"] += bufs["adv"]
            finally:
                gc.enable()
            entry["graph"] = g
        entry["graph"].replay()
        return entry["logits"]

    def _decode_forward_graphed(self, seqs, token_list, pos_list):
        entry = self._graph_entry(self._bucket(len(seqs)))  # class-0:
        self._fill_bufs(entry, seqs, token_list, pos_list)  # host samples
        return self._graph_replay(entry)[:len(seqs)]

    def warm_graphs(self, sizes=None, classes=None):
        """Pre-capture the decode graphs for EVERY (bucket, sampling
        class) at load time — a capture costs ~150 ms and ~75 MB, and a
        LAZY capture mid-serving stalls every in-flight stream for that
        long (a 50%-sampled stress wave against lazy class-1/2 capture
        timed out 189 of 303 requests before this).  With bucketing the
        set is small and fixed; load time pays ~5-10 s once."""
        if not self.use_graphs:
            return
        if classes is None:
            classes = (0, 1, 2) if self._GRAPH_TOPK else (0, 1)
        sizes = [b for b in (sizes or self._buckets)
                 if 0 < b <= self.max_batch]
        for b in sorted(set(sizes)):
            slots = []
            try:
                for _ in range(b):
                    s = self.kv.alloc_slot()
                    self.kv.ensure(s, 1)
                    slots.append(s)
            except RuntimeError:      # not enough slots/pages: skip size
                for s in slots:
                    self.kv.free_slot(s)
                continue
            host = torch.tensor(
                [[0] * b, [0] * b, slots, [1] * b], dtype=torch.int32)
            staged = host.to(self.dev, non_blocking=True)
            for kl in classes:
                entry = self._graph_entry(b, kl)
                for i, k in enumerate(("tok", "pos", "slot", "lens")):
                    entry["bufs"][k].copy_(staged[i])
                self._graph_replay(entry)
                torch.cuda.synchronize()
            for s in slots:
                self.kv.free_slot(s)

    def _decode_step(self):
        seqs = self.running
        if (self.use_pipeline and self.use_graphs
                and self._samp_class(seqs) <= 2
                and all(self.kv.seq_lens[s.slot] + 1 <= self.kv.max_ctx
                        for s in seqs)):
            self._decode_step_pipelined(seqs)
            return
        self._drain_pipe()
        # after the drain, total_len is host truth: context-full sequences
        # can't take another token — skip them; _postprocess finishes them
        seqs = [s for s in seqs if s.total_len < self.kv.max_ctx]
        if not seqs:
            return
        token_list = [s.generated[-1] for s in seqs]
        pos_list = [s.total_len - 1 for s in seqs]
        q_lens = [1] * len(seqs)
        if self.use_graphs:
            for s in seqs:
                self.kv.ensure(s.slot, self.kv.seq_lens[s.slot] + 1)
            logits = self._decode_forward_graphed(seqs, token_list, pos_list)
        else:
            logits = self._forward(seqs, token_list, pos_list, q_lens,
                                   "decode")
        toks = self._sample(seqs, logits)
        now = time.monotonic()
        for s, tok in zip(seqs, toks):
            s.generated.append(int(tok))
            self.tokens_out += 1
            if s.first_token_at is None:
                s.first_token_at = now
            if s.on_token:
                s.on_token(int(tok), False)

    # -- pipelined decode --------------------------------------------------
    def _decode_step_pipelined(self, seqs):
        """One decode step with the host one token behind the device.

        The graph's tail already advanced tok/pos/lens on-device, so a
        steady batch needs no input upload; after replay the sampled
        tokens stream back asynchronously (pinned ring + event) and the
        PREVIOUS step's tokens are applied while this step runs.
        """
        key = (tuple(s.seq_id for s in seqs), self._samp_class(seqs))
        if self._pipe is not None and self._pipe["key"] != key:
            self._drain_pipe()
        # host/device page bookkeeping for the token this replay appends
        for s in seqs:
            self.kv.ensure(s.slot, self.kv.seq_lens[s.slot] + 1)
        entry = self._graph_entry(self._bucket(len(seqs)), key[1])
        if self._pipe is None:
            # (re)sync device input buffers from host truth
            self._fill_bufs(entry, seqs,
                            [s.generated[-1] for s in seqs],
                            [s.total_len - 1 for s in seqs])
            self._pipe = {"key": key, "inflight": None}
        self._graph_replay(entry)
        i = entry["ring_i"]
        entry["ring_i"] = i ^ 1
        pinned, ev = entry["pinned"][i], entry["events"][i]
        pinned.copy_(entry["out"], non_blocking=True)
        ev.record()
        prev = self._pipe["inflight"]
        self._pipe["inflight"] = (ev, pinned, list(seqs))
        if prev is not None:
            self._apply_tokens(*prev)

    def _drain_pipe(self):
        if self._pipe is None:
            return
        inflight = self._pipe["inflight"]
        self._pipe = None
        if inflight is not None:
            self._apply_tokens(*inflight)

    def _apply_tokens(self, ev, pinned, seqs):
        ev.synchronize()
        toks = pinned.tolist()
        now = time.monotonic()
        for s, tok in zip(seqs, toks):
            if s.state != "running":   # finished/cancelled after issue
                continue
            s.generated.append(int(tok))
            self.tokens_out += 1
            if s.first_token_at is None:
                s.first_token_at = now
            if s.on_token:
                s.on_token(int(tok), False)

    def _postprocess(self, seqs) -> List[Sequence]:
        finished = []
        for s in seqs:
            if s.state != "running":
                continue
            p = s.params
            if len(s.generated) >= p.max_tokens:
                s.finish_reason = "length"
                finished.append(s)
            elif p.stop_token is not None and s.generated \
                    and s.generated[-1] == p.stop_token:
                s.finish_reason = "stop"
                finished.append(s)
            elif s.total_len >= self.kv.max_ctx:
                # context window exhausted: finish gracefully (another
                # decode would have no KV slot to append into)
                s.finish_reason = "length"
                finished.append(s)
        return finished

    def _finish(self, seq: Sequence, emit: bool = True):
        if seq.state == "done":    # already finished: never double-emit
            return
        if seq in self.running:
            self.running.remove(seq)
        if seq in self.waiting:
            self.waiting.remove(seq)
        if seq.slot >= 0:
            self.kv.free_slot(seq.slot)
            seq.slot = -1
        seq.state = "done"
        seq.finished_at = time.monotonic()
        if emit and seq.on_token:
            seq.on_token(-1, True)  # done marker
        self.seqs.pop(seq.seq_id, None)
