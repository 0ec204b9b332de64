"""Llama-3 architecture, MI355X-first implementation.

The reference (Chleba/ollamaMQ) delegates all model math to external Ollama
servers over HTTP (reference src/dispatcher.rs:742-747); this module is the
in-process compute contract those backends implicitly satisfied
(SURVEY.md §2 "CUDA kernels ... There are none").

Design notes (MI355X):
* weights live resident in HBM3E as bf16 (8B = ~16 GB of 288 GB);
* projection GEMMs go through torch F.linear (hipBLASLt/rocBLAS) — plain
  library GEMMs; every fused hot op (fused residual+RMSNorm, RoPE,
  prefill/decode attention over the paged KV pool, SwiGLU, sampler) is a
  hand-written gfx950 HIP kernel behind ollamamq_amd.ops;
* tensor parallelism: column-parallel QKV/gate-up, row-parallel o/down with
  one RCCL all-reduce each (2/layer), vocab-parallel logits with all-gather.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Optional

import torch

from ..ops import interface as ops


@dataclass
class LlamaConfig:
    name: str = "llama3-8b"
    n_layers: int = 32
    hidden: int = 4096
    n_heads: int = 32
    n_kv_heads: int = 8
    head_dim: int = 128
    ffn: int = 14336
    vocab: int = 128256
    norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    max_ctx: int = 8192
    # family variants sharing the Llama block: Qwen2 adds a bias on the
    # QKV projection; Mistral restricts attention to a sliding window
    qkv_bias: bool = False
    sliding_window: int = 0       # 0 = full causal
    # Llama-3.1-style NTK rope scaling (factor, low_freq_factor,
    # high_freq_factor, original_max_ctx); None = plain RoPE
    rope_scaling: Optional[tuple] = None

    @property
    def q_dim(self) -> int:
        return self.n_heads * self.head_dim

    @property
    def kv_dim(self) -> int:
        return self.n_kv_heads * self.head_dim


PRESETS = {
    # Llama-3 8B: 32 layers, hidden 4096, 32Q/8KV, ffn 14336, vocab 128256
    "llama3-8b": LlamaConfig(),
    # Llama-3 70B: 80 layers, hidden 8192, 64Q/8KV, ffn 28672
    "llama3-70b": LlamaConfig(
        name="llama3-70b", n_layers=80, hidden=8192, n_heads=64, n_kv_heads=8,
        ffn=28672,
    ),
    # Llama-3.1 8B: same arch as 8B + NTK-scaled RoPE for 32k+ contexts
    "llama3.1-8b": LlamaConfig(
        name="llama3.1-8b", max_ctx=32768,
        rope_scaling=(8.0, 1.0, 4.0, 8192),
    ),
    # Llama-2 7B/13B: MHA (no GQA), rope theta 1e4, vocab 32000
    "llama2-7b": LlamaConfig(
        name="llama2-7b", n_layers=32, hidden=4096, n_heads=32,
        n_kv_heads=32, ffn=11008, vocab=32000, rope_theta=10000.0,
        max_ctx=4096,
    ),
    "llama2-13b": LlamaConfig(
        name="llama2-13b", n_layers=40, hidden=5120, n_heads=40,
        n_kv_heads=40, ffn=13824, vocab=32000, rope_theta=10000.0,
        max_ctx=4096,
    ),
    # Qwen2-7B: QKV bias, GQA 28Q/4KV, rope theta 1e6, vocab 152064
    "qwen2-7b": LlamaConfig(
        name="qwen2-7b", n_layers=28, hidden=3584, n_heads=28, n_kv_heads=4,
        ffn=18944, vocab=152064, rope_theta=1_000_000.0, qkv_bias=True,
    ),
    # Mistral-7B: sliding-window attention (W=4096), vocab 32000
    "mistral-7b": LlamaConfig(
        name="mistral-7b", n_layers=32, hidden=4096, n_heads=32,
        n_kv_heads=8, ffn=14336, vocab=32000, rope_theta=10000.0,
        sliding_window=4096,
    ),
    # Tiny config for CPU tests and fast GPU smoke: same head_dim=128 the
    # kernels are specialized for.
    "tiny": LlamaConfig(
        name="tiny", n_layers=2, hidden=512, n_heads=4, n_kv_heads=2,
        head_dim=128, ffn=1024, vocab=512, max_ctx=512, rope_theta=10000.0,
    ),
    # Extra-small for scheduler tests (CPU-fast).
    "tiny-cpu": LlamaConfig(
        name="tiny-cpu", n_layers=2, hidden=256, n_heads=2, n_kv_heads=1,
        head_dim=128, ffn=512, vocab=256, max_ctx=256, rope_theta=10000.0,
    ),
    # Tiny family-variant configs exercising the Qwen2 bias path and the
    # Mistral sliding-window path end to end (CPU tests + GPU smoke).
    "tiny-qwen": LlamaConfig(
        name="tiny-qwen", n_layers=2, hidden=512, n_heads=4, n_kv_heads=2,
        head_dim=128, ffn=1024, vocab=512, max_ctx=512, rope_theta=10000.0,
        qkv_bias=True,
    ),
    "tiny-swa": LlamaConfig(
        name="tiny-swa", n_layers=2, hidden=512, n_heads=4, n_kv_heads=2,
        head_dim=128, ffn=1024, vocab=512, max_ctx=512, rope_theta=10000.0,
        sliding_window=96,
    ),
    # Disjoint NAME from every other preset: multi-model routing tests
    # need a model that neither smart- nor fuzzy-matches "tiny*"
    # (fuzzy matching is substring-both-ways, reference
    # src/dispatcher.rs:381-393, so "tiny" routes to "tiny-qwen" too).
    "nano": LlamaConfig(
        name="nano", n_layers=2, hidden=512, n_heads=4, n_kv_heads=2,
        head_dim=128, ffn=1024, vocab=512, max_ctx=512, rope_theta=10000.0,
    ),
}


def _randn(shape, dev, dtype, gen, scale):
    # generate on the generator's own device (CPU gen => bit-identical
    # weights across CPU/GPU test runs; CUDA gen => fast 16 GB init for the
    # big models straight in HBM)
    w = torch.empty(shape, device=gen.device, dtype=torch.float32)
    w.normal_(0.0, scale, generator=gen)
    return w.to(device=dev, dtype=dtype)


class LlamaLayer:
    """One transformer block's resident weights (possibly TP-sharded).

    Every rank generates the SAME seeded global matrix and keeps its own
    shard, so a TP=N model computes the same function as TP=1 (the gloo
    CPU tests assert logits parity across degrees)."""

    __slots__ = (
        "wqkv", "bqkv", "wo", "wgate_up", "wdown", "attn_norm", "mlp_norm",
        "wqkv_pk", "wo_pk", "wgu_pk", "wdown_pk", "bqkv_rp",
    )

    def __init__(self, cfg: LlamaConfig, dev, dtype, gen, tp: int, rank: int):
        h = cfg.hidden
        scale = 1.0 / math.sqrt(h)
        nh, nkv = cfg.n_heads // tp, max(1, cfg.n_kv_heads // tp)
        d = cfg.head_dim
        Hq, KVH = cfg.n_heads, cfg.n_kv_heads
        # global fused QKV [Hq*d + 2*KVH*d, h]; column-parallel: this rank
        # keeps its q-head rows + its kv-head rows
        wqkv_g = _randn(((Hq + 2 * KVH) * d, h), dev, dtype, gen, scale)
        q = wqkv_g[rank * nh * d:(rank + 1) * nh * d]
        k = wqkv_g[Hq * d + rank * nkv * d: Hq * d + (rank + 1) * nkv * d]
        v = wqkv_g[(Hq + KVH) * d + rank * nkv * d:
                   (Hq + KVH) * d + (rank + 1) * nkv * d]
        self.wqkv = torch.cat([q, k, v], dim=0).contiguous()
        del wqkv_g
        # Qwen2-style QKV bias, sharded with the same row split
        self.bqkv = None
        if cfg.qkv_bias:
            b_g = _randn(((Hq + 2 * KVH) * d,), dev, dtype, gen, scale)
            bq = b_g[rank * nh * d:(rank + 1) * nh * d]
            bk = b_g[Hq * d + rank * nkv * d: Hq * d + (rank + 1) * nkv * d]
            bv = b_g[(Hq + KVH) * d + rank * nkv * d:
                     (Hq + KVH) * d + (rank + 1) * nkv * d]
            self.bqkv = torch.cat([bq, bk, bv], dim=0).contiguous()
            del b_g
        # row-parallel output projection: columns of the global [h, Hq*d]
        wo_g = _randn((h, Hq * d), dev, dtype, gen, scale)
        self.wo = wo_g[:, rank * nh * d:(rank + 1) * nh * d].contiguous()
        del wo_g
        # column-parallel gate+up: global [2F, h], shard gate + up rows
        F, f = cfg.ffn, cfg.ffn // tp
        wgu_g = _randn((2 * F, h), dev, dtype, gen, scale)
        self.wgate_up = torch.cat(
            [wgu_g[rank * f:(rank + 1) * f],
             wgu_g[F + rank * f:F + (rank + 1) * f]], dim=0).contiguous()
        del wgu_g
        # row-parallel down: columns of the global [h, F]
        wd_g = _randn((h, F), dev, dtype, gen, 1.0 / math.sqrt(cfg.ffn))
        self.wdown = wd_g[:, rank * f:(rank + 1) * f].contiguous()
        del wd_g
        # rmsnorm weights are FOLDED into the projections at build
        # (rmsnorm(x)*w @ W^T == rmsnorm_unit(x) @ (W diag w)^T — a pure
        # reparameterization), so every execution path (lib GEMM, packed
        # stream, fused chain) shares one consistent weight set and the
        # runtime norm weights are identically ones.  A checkpoint
        # loader must apply the same fold.
        attn_w = torch.ones(h, device=dev, dtype=dtype)
        mlp_w = torch.ones(h, device=dev, dtype=dtype)
        self.wqkv = (self.wqkv * attn_w).contiguous()
        self.wgate_up = (self.wgate_up * mlp_w).contiguous()
        self.attn_norm = torch.ones(h, device=dev, dtype=dtype)
        self.mlp_norm = torch.ones(h, device=dev, dtype=dtype)
        # packed decode copies filled by LlamaModel._pack_weights()
        self.wqkv_pk = self.wo_pk = self.wgu_pk = self.wdown_pk = None
        self.bqkv_rp = None


class LlamaModel:
    """Random-init resident Llama model, forward built on ollamamq_amd.ops.

    Holds no KV state: attention reads/writes the engine's paged KV pool.
    """

    def __init__(
        self,
        cfg: LlamaConfig,
        device: str = "cpu",
        dtype: torch.dtype = torch.float32,
        seed: int = 1234,
        tp_rank: int = 0,
        tp_size: int = 1,
        process_group=None,
        fast_init: bool = False,
    ):
        assert cfg.n_heads % tp_size == 0, "n_heads must divide TP"
        self.cfg = cfg
        self.device = torch.device(device)
        self.dtype = dtype
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.group = process_group
        # KV-head replication when tp > n_kv_heads is not supported; shard.
        self.n_local_heads = cfg.n_heads // tp_size
        self.n_local_kv_heads = max(1, cfg.n_kv_heads // tp_size)

        gen_dev = "cuda" if (fast_init and self.device.type == "cuda") else "cpu"
        gen = torch.Generator(device=gen_dev)
        gen.manual_seed(seed)  # same seed on every rank: shards slice one
                               # consistent global weight set
        dev = self.device
        scale = 1.0 / math.sqrt(cfg.hidden)
        self.embed = _randn((cfg.vocab, cfg.hidden), dev, dtype, gen, scale)
        self.layers = [
            LlamaLayer(cfg, dev, dtype, gen, tp_size, tp_rank)
            for _ in range(cfg.n_layers)
        ]
        self.final_norm = torch.ones(cfg.hidden, device=dev, dtype=dtype)
        # Vocab-parallel LM head shard: rows of the global [vocab, hidden]
        # (final_norm folds in like the per-layer norms — see LlamaLayer)
        vshard = cfg.vocab // tp_size
        lm_g = _randn((cfg.vocab, cfg.hidden), dev, dtype, gen, scale)
        self.lm_head = (lm_g[tp_rank * vshard:(tp_rank + 1) * vshard]
                        * self.final_norm).contiguous()
        del lm_g
        # RoPE cos/sin tables precomputed on host (guide: trig on device
        # turns memory-bound RoPE into VALU-bound).
        pos = torch.arange(cfg.max_ctx, dtype=torch.float32)
        inv = 1.0 / (
            cfg.rope_theta
            ** (torch.arange(0, cfg.head_dim, 2, dtype=torch.float32) / cfg.head_dim)
        )
        if cfg.rope_scaling is not None:
            # Llama-3.1 NTK scaling: long wavelengths divided by `factor`,
            # short kept, smooth ramp between (host-side table only — the
            # RoPE kernel is scaling-agnostic)
            factor, lo_f, hi_f, orig = cfg.rope_scaling
            wavelen = 2 * math.pi / inv
            lo_wl, hi_wl = orig / lo_f, orig / hi_f
            smooth = ((orig / wavelen - lo_f) / (hi_f - lo_f)).clamp(0, 1)
            scaled = (1 - smooth) * inv / factor + smooth * inv
            inv = torch.where(wavelen > lo_wl, inv / factor,
                              torch.where(wavelen < hi_wl, inv, scaled))
        ang = torch.outer(pos, inv)  # [max_ctx, head_dim/2]
        self.rope_cos = ang.cos().to(dev)
        self.rope_sin = ang.sin().to(dev)
        self.lm_head_pk = None
        self.fused_chain = False
        self._pack_weights()

    def _pack_weights(self):
        """Build MFMA-fragment-order packed copies of every projection for
        the weight-streaming decode GEMM.  Doubles weight HBM (8B: 16->32
        of 288 GB — prefill keeps the standard layout for hipBLASLt); on
        models where the copy would not fit (70B on one GPU) packing is
        skipped wholesale and decode stays on the library GEMMs."""
        import os
        if self.device.type != "cuda" or self.dtype != torch.bfloat16 \
                or os.environ.get("OLLAMAMQ_NO_PACK") == "1":
            return
        free, _ = torch.cuda.mem_get_info(self.device)
        if self.weight_bytes() > free * 0.45:
            return
        # The rmsnorm WEIGHT is folded into the qkv / gate_up / lm_head
        # packs (rmsnorm(x)*w @ W^T == rmsnorm_unit(x) @ (W*diag(w))^T),
        # so the fused decode chain needs no rmsnorm kernel at all — the
        # GEMM epilogues carry rstd scaling + residual + stats.  Folded
        # packs are ONLY read by the fused chain; the generic path keeps
        # explicit rmsnorm + unfolded weights.
        pk = ops.pack_weight
        try:
            for l in self.layers:
                # qkv pack is PAIR-ORDERED for the fused RoPE/KV-append
                # epilogue (head_dim==128 models; the chain requires it).
                # Norm weights are already folded at build, so the packs
                # are pure layout transforms of the live weights.
                if self.cfg.head_dim == 128:
                    l.wqkv_pk = ops.pack_weight_qkv_rope(
                        l.wqkv, self.n_local_heads,
                        self.n_local_kv_heads)
                    if l.bqkv is not None:
                        l.bqkv_rp = ops.qkv_rope_bias_order(
                            l.bqkv, self.n_local_heads,
                            self.n_local_kv_heads)
                else:
                    l.wqkv_pk = None
                l.wo_pk = pk(l.wo)
                l.wgu_pk = ops.pack_weight_gu(l.wgate_up)
                l.wdown_pk = pk(l.wdown)
            self.lm_head_pk = pk(self.lm_head)
        except torch.cuda.OutOfMemoryError:
            for l in self.layers:
                l.wqkv_pk = l.wo_pk = l.wgu_pk = l.wdown_pk = None
            self.lm_head_pk = None
            torch.cuda.empty_cache()
            return
        self.fused_chain = (
            self.tp_size == 1 and self.group is None
            and os.environ.get("OLLAMAMQ_NO_CHAIN") != "1"
            and self.lm_head_pk is not None
            and all(l.wqkv_pk is not None and l.wo_pk is not None
                    and l.wgu_pk is not None and l.wdown_pk is not None
                    for l in self.layers))
        if self.fused_chain:
            th = self.cfg.hidden // 32
            self._sq_a = torch.zeros(max(th, 1) * 64, dtype=torch.float32,
                                     device=self.device)
            self._sq_b = torch.zeros_like(self._sq_a)
            # frag-layout residual stream: two 32-row halves (the MT2
            # chain serves batches up to 64; rows >= batch hold garbage
            # that only dead accumulator rows ever see)
            self._res_frag = torch.zeros(64 * self.cfg.hidden,
                                         dtype=self.dtype,
                                         device=self.device)

    # -- helpers -----------------------------------------------------------
    def _allreduce(self, x: torch.Tensor) -> torch.Tensor:
        # tp_size > 1: the 2-per-layer TP all-reduces.  An explicitly
        # passed group at tp_size == 1 ALSO reduces (numerically a no-op):
        # that is how the graph+RCCL capture mechanics are validated on a
        # single GPU (tests/test_tp_graph_gpu.py) before an 8-GPU node
        # ever runs them.
        if self.tp_size > 1 or self.group is not None:
            torch.distributed.all_reduce(x, group=self.group)
        return x

    def weight_bytes(self) -> int:
        n = self.embed.numel() + self.lm_head.numel() + self.final_norm.numel()
        for l in self.layers:
            n += (
                l.wqkv.numel() + l.wo.numel() + l.wgate_up.numel()
                + l.wdown.numel() + l.attn_norm.numel() + l.mlp_norm.numel()
                + (l.bqkv.numel() if l.bqkv is not None else 0)
            )
        return n * self.embed.element_size()

    # -- forward -----------------------------------------------------------
    def forward(
        self,
        tokens: torch.Tensor,      # [T] int32/int64 flat token ids
        positions: torch.Tensor,   # [T] int32 position of each token in its seq
        kv_cache,                  # engine.kvcache.PagedKVCache
        slot_ids: torch.Tensor,    # [T] int32 kv slot (sequence) of each token
        attn_meta,                 # ops.AttnMeta (prefill/decode metadata)
        return_hidden: bool = False,
    ) -> torch.Tensor:
        """Returns logits [T_last, vocab_full] for the tokens attn_meta
        selects as "last" (decode: all; prefill: final token per seq)."""
        cfg = self.cfg
        if (self.fused_chain and attn_meta.mode == "decode"
                and tokens.shape[0] <= 64 and not return_hidden):
            return self._forward_decode_fused(tokens, positions, kv_cache,
                                              slot_ids, attn_meta)
        x = ops.embedding(tokens, self.embed)
        residual = None
        for li, layer in enumerate(self.layers):
            normed, residual = ops.rmsnorm_residual(
                x, residual, layer.attn_norm, cfg.norm_eps
            )
            # NB: no packed= here — wqkv_pk has the norm weight folded in
            # and is readable only by the fused chain
            qkv = ops.linear(normed, layer.wqkv, layer.bqkv)
            nl, nkl, d = self.n_local_heads, self.n_local_kv_heads, cfg.head_dim
            q, k, v = qkv.split([nl * d, nkl * d, nkl * d], dim=-1)
            q = q.view(-1, nl, d)
            k = k.view(-1, nkl, d)
            v = v.view(-1, nkl, d)
            ops.rope_append(kv_cache, li, q, k, v, positions, slot_ids,
                            self.rope_cos, self.rope_sin)
            attn = ops.attention(q, kv_cache, li, attn_meta)
            x = ops.linear(attn.view(-1, nl * d), layer.wo,
                           packed=layer.wo_pk)
            self._allreduce(x)  # RCCL all-reduce #1 (TP)
            normed, residual = ops.rmsnorm_residual(
                x, residual, layer.mlp_norm, cfg.norm_eps
            )
            act = ops.gateup_swiglu(normed, layer.wgate_up,
                                    layer.wgu_pk)
            x = ops.linear(act, layer.wdown, packed=layer.wdown_pk)
            self._allreduce(x)  # RCCL all-reduce #2 (TP)

        # Only the last token of each sequence needs logits.
        idx = attn_meta.logits_idx
        if idx is not None:
            h = x[idx] + residual[idx]
        else:
            h = x + residual
        h, _ = ops.rmsnorm_residual(h, None, self.final_norm, cfg.norm_eps)
        if return_hidden:
            return h
        # NB: no packed= — the MT2 streaming GEMM loses to the library
        # above batch 32 (doubled x scatter); the fused chain covers <=32
        logits = ops.linear(h, self.lm_head)
        if self.tp_size > 1:
            # vocab-parallel logits: all-gather shards on the last dim
            shards = [torch.empty_like(logits) for _ in range(self.tp_size)]
            torch.distributed.all_gather(shards, logits, group=self.group)
            logits = torch.cat(shards, dim=-1)
        return logits

    def _forward_decode_fused(self, tokens, positions, kv_cache, slot_ids,
                              attn_meta):
        """Decode step as a fused weight-streaming chain (tp=1, B<=32):
        4 GEMM-class kernels per layer, zero rmsnorm/swiglu kernels.
        rmsnorm weights are folded into the packs; the rstd scale rides
        each consumer GEMM's epilogue from sum-of-squares partials the
        producer GEMM emitted; residual adds happen in place inside the
        o/down epilogues (SURVEY.md §2 kernel-table GEMM rows — this is
        the MFMA-native decode path)."""
        cfg = self.cfg
        B = tokens.shape[0]
        H = cfg.hidden
        nl, nkl, d = self.n_local_heads, self.n_local_kv_heads, cfg.head_dim
        emb = ops.embedding(tokens, self.embed)
        if not emb.is_contiguous():
            emb = emb.contiguous()
        sq_a, sq_b = self._sq_a, self._sq_b
        # res lives in the 32-row frag layout for the WHOLE chain: every
        # consumer GEMM streams its activation input linearly exactly
        # like packed weights (the scattered x machinery measured 15-30%
        # on top of the pure stream rate — profiles/r02_gemm_sweep.md)
        res = ops.fragify_sumsq(emb, xf=self._res_frag, sq=sq_a[:B])[0]
        nt = 1                      # embed stats = one partial tile
        inv_h = 1.0 / H
        eps = cfg.norm_eps
        tiles_h = H // 32
        for li, layer in enumerate(self.layers):
            # fused rmsnorm -> qkv -> RoPE -> paged KV append: q comes
            # back rotated in standard layout, k/v land in the pool
            qkv = ops.qkv_rope_fused(
                res, layer.wqkv_pk, layer.wqkv.shape[0], layer.bqkv_rp,
                kv_cache, li, positions, slot_ids,
                self.rope_cos, self.rope_sin, nl, nkl,
                rstd=sq_a, rstd_nt=nt, inv_h=inv_h, eps=eps,
                K=H, M_real=B)
            q = qkv[:, :nl * d].view(-1, nl, d)
            attn = ops.attention_decode_frag(q, kv_cache, li, attn_meta)
            ops.linear_fused(attn, layer.wo_pk, H, res=res, sq_out=sq_b,
                             y=res, yfrag=1, K=nl * d, xlds=2, M_frag=B)
            act = ops.gu_fused(res, layer.wgu_pk,
                               layer.wgate_up.shape[0], rstd=sq_b,
                               rstd_nt=tiles_h, inv_h=inv_h, eps=eps,
                               K=H, yfrag=1, M_frag=B)
            ops.linear_fused(act, layer.wdown_pk, H, res=res,
                             sq_out=sq_a, y=res, yfrag=1,
                             K=layer.wdown.shape[1], xlds=2, M_frag=B)
            nt = tiles_h
        logits = ops.linear_fused(res, self.lm_head_pk,
                                  self.lm_head.shape[0], rstd=sq_a,
                                  rstd_nt=nt, inv_h=inv_h, eps=eps,
                                  K=H, xlds=2, M_frag=B)
        return logits[:B]
