"""Paged KV cache for the continuous-batching decoder.

Layout: per layer, K and V caches of shape (num_pages, page_size, Hk, D) bf16
— the layout the HIP kernels (kv_cache.hip, attention_decode.hip) index.
A simple free-list page allocator; per-sequence page lists grown on demand.
Sized against a byte budget (288 GB HBM per MI355X; the scheduler decides the
split between weights / KV / activations — SURVEY.md §5.9).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch


class PagedKVCache:
    def __init__(self, num_layers: int, num_kv_heads: int, head_dim: int,
                 num_pages: int, page_size: int = 16,
                 dtype: torch.dtype = torch.bfloat16, device: str = "cuda"):
        self.num_layers = num_layers
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        self.num_pages = num_pages
        self.page_size = page_size
        self.device = device
        self.k_cache = [
            torch.zeros(num_pages, page_size, num_kv_heads, head_dim,
                        dtype=dtype, device=device)
            for _ in range(num_layers)
        ]
        self.v_cache = [
            torch.zeros(num_pages, page_size, num_kv_heads, head_dim,
                        dtype=dtype, device=device)
            for _ in range(num_layers)
        ]
        self._free: List[int] = list(range(num_pages - 1, -1, -1))
        self._seq_pages: Dict[int, List[int]] = {}
        self._seq_len: Dict[int, int] = {}
        # page refcounts for prefix sharing (radix-cache capability,
        # SURVEY.md §2.4.3): n samples of one prompt share its full pages
        self._ref: Dict[int, int] = {}

    # ------------------------------------------------------------- accounting
    @classmethod
    def bytes_per_token(cls, num_layers: int, num_kv_heads: int,
                        head_dim: int, dtype_bytes: int = 2) -> int:
        return 2 * num_layers * num_kv_heads * head_dim * dtype_bytes

    @property
    def free_pages(self) -> int:
        return len(self._free)

    def free_token_capacity(self) -> int:
        return len(self._free) * self.page_size

    # -------------------------------------------------------------- lifecycle
    def can_allocate(self, num_tokens: int) -> bool:
        return (num_tokens + self.page_size - 1) // self.page_size <= len(self._free)

    def allocate(self, seq_id: int, num_tokens: int) -> bool:
        """Grow seq's allocation to hold num_tokens MORE tokens.  Returns
        False (allocating nothing) if pages are exhausted."""
        cur_len = self._seq_len.get(seq_id, 0)
        pages = self._seq_pages.setdefault(seq_id, [])
        need_pages = (cur_len + num_tokens + self.page_size - 1) // self.page_size
        grow = need_pages - len(pages)
        if grow > len(self._free):
            return False
        for _ in range(grow):
            p = self._free.pop()
            self._ref[p] = 1
            pages.append(p)
        self._seq_len[seq_id] = cur_len + num_tokens
        return True

    def free_seq(self, seq_id: int):
        for p in self._seq_pages.pop(seq_id, []):
            self._ref[p] = self._ref.get(p, 1) - 1
            if self._ref[p] <= 0:
                self._ref.pop(p, None)
                self._free.append(p)
        self._seq_len.pop(seq_id, None)

    def fork_seq(self, parent_id: int, child_ids: List[int],
                 prefix_len: int) -> Optional[List[Tuple[int, int]]]:
        """Prefix sharing: every child references the parent's FULL pages
        of the first ``prefix_len`` tokens; a partial trailing page is given
        to each child as a fresh page (the caller copies its KV content —
        decode writes must never touch shared pages).  The parent keeps its
        own pages (it becomes sample 0).

        Returns [(src_page, dst_page), ...] partial-page copies the caller
        must perform (empty when prefix_len %% page_size == 0), or None if
        pages ran out (nothing allocated)."""
        parent_pages = self._seq_pages[parent_id]
        full = prefix_len // self.page_size
        rem = prefix_len % self.page_size
        need_fresh = len(child_ids) * (1 if rem else 0)
        if need_fresh > len(self._free):
            return None
        copies: List[Tuple[int, int]] = []
        for cid in child_ids:
            pages = list(parent_pages[:full])
            for p in pages:
                self._ref[p] = self._ref.get(p, 1) + 1
            if rem:
                fresh = self._free.pop()
                self._ref[fresh] = 1
                copies.append((parent_pages[full], fresh))
                pages.append(fresh)
            self._seq_pages[cid] = pages
            self._seq_len[cid] = prefix_len
        return copies

    def seq_len(self, seq_id: int) -> int:
        return self._seq_len.get(seq_id, 0)

    # --------------------------------------------------------------- indexing
    def slots_for(self, seq_id: int, start: int, count: int) -> torch.Tensor:
        """slot ids (page*page_size + off) for token positions [start, start+count)."""
        pages = self._seq_pages[seq_id]
        out = torch.empty(count, dtype=torch.int32)
        for i in range(count):
            pos = start + i
            out[i] = pages[pos // self.page_size] * self.page_size + pos % self.page_size
        return out

    def page_table(self, seq_ids: List[int], max_pages: Optional[int] = None
                   ) -> torch.Tensor:
        if max_pages is None:
            max_pages = max((len(self._seq_pages[s]) for s in seq_ids), default=1)
        max_pages = max(max_pages, 1)
        pt = torch.zeros(len(seq_ids), max_pages, dtype=torch.int32)
        for i, s in enumerate(seq_ids):
            pages = self._seq_pages[s]
            pt[i, :len(pages)] = torch.tensor(pages, dtype=torch.int32)
        return pt
