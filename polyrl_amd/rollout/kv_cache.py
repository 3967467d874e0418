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

    def seed_seq(self, seq_id: int, pages: List[int], length: int):
        """Start a sequence whose first ``length`` (page-aligned) tokens
        live in already-filled shared pages (radix-cache hit).  Takes
        OWNERSHIP of one pre-taken ref per page (RadixCache.match)."""
        assert seq_id not in self._seq_pages and length % self.page_size == 0
        self._seq_pages[seq_id] = list(pages)
        self._seq_len[seq_id] = length

    def unref_pages(self, pages: List[int]):
        """Drop refs taken by RadixCache.match on an admission bail-out."""
        for p in pages:
            r = self._ref.get(p, 1) - 1
            if r <= 0:
                self._ref.pop(p, None)
                self._free.append(p)
            else:
                self._ref[p] = r

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

    def shrink_seq(self, seq_id: int, num_tokens: int):
        """Roll back the last ``num_tokens`` of an allocation (chunked-decode
        growth that was abandoned): trailing pages no longer needed are
        released (decref — shared pages survive).  The seq-len bookkeeping
        must stay exact: RadixCache.insert donates pages for exactly the
        WRITTEN tokens, so an inflated length would cache garbage KV."""
        cur = self._seq_len[seq_id] - num_tokens
        assert cur >= 0
        pages = self._seq_pages[seq_id]
        need = (cur + self.page_size - 1) // self.page_size
        while len(pages) > max(need, 0):
            p = pages.pop()
            r = self._ref.get(p, 1) - 1
            if r <= 0:
                self._ref.pop(p, None)
                self._free.append(p)
            else:
                self._ref[p] = r
        self._seq_len[seq_id] = cur

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


class _RadixNode:
    __slots__ = ("chunk", "page", "children", "parent", "t")

    def __init__(self, chunk, page, parent):
        self.chunk = chunk          # tuple of page_size token ids
        self.page = page            # page id this node pins
        self.children: Dict[tuple, "_RadixNode"] = {}
        self.parent = parent
        self.t = 0                  # LRU clock


class RadixCache:
    """Cross-request KV prefix cache (the SGLang radix-cache capability the
    reference relies on for shared system prompts / multi-turn reuse —
    SURVEY.md §2.2.2 kernel-suite row, §2.4.3).

    Page-granular trie: each node is exactly ``page_size`` tokens pinning one
    KV page (+1 on the allocator refcount).  A new request walks the trie
    over its prompt in page chunks; matched pages seed its page table
    (skipping their prefill entirely) and only the tail is computed.
    Finished sequences donate their full pages back into the trie.  Under
    page pressure the engine evicts LRU leaves (pages still referenced by
    running sequences survive the node's removal via the refcount).

    MI355X sizing note: at 288 GB HBM the KV pool is large enough that
    eviction is rare — the tree mostly just grows, which is the cheap path.
    """

    def __init__(self, kv: PagedKVCache):
        self.kv = kv
        self.ps = kv.page_size
        self.root: Dict[tuple, _RadixNode] = {}
        self._clock = 0
        self._nodes = 0
        self.hit_tokens = 0
        self.query_tokens = 0

    # ---------------------------------------------------------------- match
    def match(self, tokens: List[int]) -> Tuple[List[int], int]:
        """Longest page-aligned cached prefix of ``tokens`` that still
        leaves >= 1 token to prefill.  Takes ONE allocator ref per returned
        page (the caller hands them to ``seed_seq`` — which takes ownership
        of those refs — or must ``unref_pages`` on bail-out)."""
        self._clock += 1
        self.query_tokens += len(tokens)
        max_match = ((len(tokens) - 1) // self.ps) * self.ps
        pages: List[int] = []
        level = self.root
        off = 0
        while off < max_match:
            chunk = tuple(tokens[off:off + self.ps])
            node = level.get(chunk)
            if node is None:
                break
            node.t = self._clock
            pages.append(node.page)
            level = node.children
            off += self.ps
        for p in pages:
            self.kv._ref[p] = self.kv._ref.get(p, 0) + 1
        self.hit_tokens += off
        return pages, off

    # --------------------------------------------------------------- insert
    def insert(self, tokens: List[int], pages: List[int]) -> int:
        """Absorb a finished sequence's FULL pages into the trie.  ``pages``
        must align 1:1 with page-sized chunks of ``tokens``.  Existing nodes
        are just touched (the donor's duplicate page stays with the donor);
        new nodes take +1 ref on the donated page.  Returns nodes added."""
        self._clock += 1
        added = 0
        level = self.root
        for i in range(len(tokens) // self.ps):
            chunk = tuple(tokens[i * self.ps:(i + 1) * self.ps])
            node = level.get(chunk)
            if node is None:
                node = _RadixNode(chunk, pages[i], None)
                node.t = self._clock
                level[chunk] = node
                self.kv._ref[pages[i]] = self.kv._ref.get(pages[i], 0) + 1
                self._nodes += 1
                added += 1
            else:
                node.t = self._clock
            level = node.children
        return added

    # --------------------------------------------------------------- evict
    def _leaves(self):
        out = []
        stack = [(self.root, None)]
        while stack:
            level, parent = stack.pop()
            for chunk, node in level.items():
                if node.children:
                    stack.append((node.children, node))
                else:
                    out.append((node, level, chunk))
        return out

    def evict(self, need_pages: int) -> int:
        """Drop LRU leaves until >= need_pages returned to the free list (or
        the tree is empty).  Only pages whose last reference was the cache's
        actually free; others (still used by running seqs) just unpin."""
        freed = 0
        while freed < need_pages:
            leaves = self._leaves()
            if not leaves:
                break
            leaves.sort(key=lambda x: x[0].t)
            progressed = False
            for node, level, chunk in leaves:
                if freed >= need_pages:
                    break
                del level[chunk]
                self._nodes -= 1
                progressed = True
                r = self.kv._ref.get(node.page, 1) - 1
                if r <= 0:
                    self.kv._ref.pop(node.page, None)
                    self.kv._free.append(node.page)
                    freed += 1
                else:
                    self.kv._ref[node.page] = r
            if not progressed:
                break
        return freed

    def reset(self):
        """Drop the whole tree (weight update => cached KV is stale)."""
        self.evict(1 << 62)

    @property
    def num_nodes(self) -> int:
        return self._nodes
