"""Paged KV cache sized for 288 GB HBM3E per MI355X.

Layout [L, n_pages, KVH, page_size, D]: one (page, kv-head) is a contiguous
page_size*D*2B run (4 KiB at page=16, D=128), the unit the decode-attention
kernel streams with wide vector loads.  Page tables are int32 per slot.

The reference has no KV concept (its context handling is only plumbing the
`num_ctx` option to Ollama, reference src/control.rs:804-806); the paged pool
is this framework's realization of "model loaded with context N" — "load"
allocates weights + this pool in HBM, "unload" frees them.
"""
from __future__ import annotations

from typing import List

import torch


class PagedKVCache:
    def __init__(
        self,
        n_layers: int,
        n_kv_heads: int,
        head_dim: int,
        page_size: int = 16,
        n_pages: int = 1024,
        max_slots: int = 64,
        max_ctx: int = 8192,
        device: str = "cpu",
        dtype: torch.dtype = torch.float32,
    ):
        self.page_size = page_size
        self.head_dim = head_dim
        self.n_kv_heads = n_kv_heads
        self.max_ctx = max_ctx
        dev = torch.device(device)
        shape = (n_layers, n_pages, n_kv_heads, page_size, head_dim)
        self.k_pool = torch.zeros(shape, device=dev, dtype=dtype)
        self.v_pool = torch.zeros(shape, device=dev, dtype=dtype)
        max_pages = (max_ctx + page_size - 1) // page_size
        self.page_table = torch.zeros(
            (max_slots, max_pages), device=dev, dtype=torch.int32
        )
        self._free_pages: List[int] = list(range(n_pages - 1, -1, -1))
        self._free_slots: List[int] = list(range(max_slots - 1, -1, -1))
        self._slot_pages: List[List[int]] = [[] for _ in range(max_slots)]
        self.seq_lens = [0] * max_slots  # host-side truth
        self.max_slots = max_slots
        self.n_pages = n_pages

    @classmethod
    def for_model(cls, cfg, tp_size=1, **kw):
        n_kv = max(1, cfg.n_kv_heads // tp_size)
        return cls(cfg.n_layers, n_kv, cfg.head_dim, **kw)

    def free_page_count(self) -> int:
        return len(self._free_pages)

    def alloc_slot(self) -> int:
        if not self._free_slots:
            raise RuntimeError("KV cache: no free sequence slots")
        slot = self._free_slots.pop()
        self.seq_lens[slot] = 0
        return slot

    def free_slot(self, slot: int) -> None:
        self._free_pages.extend(self._slot_pages[slot])
        self._slot_pages[slot] = []
        self.seq_lens[slot] = 0
        self._free_slots.append(slot)

    def ensure(self, slot: int, new_len: int) -> None:
        """Grow slot's page list to cover new_len tokens."""
        if new_len > self.max_ctx:
            raise RuntimeError(f"sequence exceeds max_ctx {self.max_ctx}")
        need = (new_len + self.page_size - 1) // self.page_size
        pages = self._slot_pages[slot]
        while len(pages) < need:
            if not self._free_pages:
                raise RuntimeError("KV cache: out of pages")
            p = self._free_pages.pop()
            self.page_table[slot, len(pages)] = p
            pages.append(p)
        self.seq_lens[slot] = new_len

    def can_fit(self, extra_tokens: int) -> bool:
        return len(self._free_pages) * self.page_size >= extra_tokens
