"""Sequence-length balancing + dynamic token-budget micro-batching.

Capability parity with verl's seqlen_balancing utilities consumed by the
reference (prepare_dynamic_batch / restore_dynamic_batch /
get_seqlen_balanced_partitions, _balance_batch at stream_ray_trainer.py:406-410).
Greedy Karmarkar-Karp-style partitioning, implemented fresh.
"""
from __future__ import annotations

import heapq
from typing import List, Tuple

import torch

from ..protocol import TensorBatch


def get_seqlen_balanced_partitions(seqlens: List[int], k_partitions: int,
                                   equal_size: bool) -> List[List[int]]:
    """Partition indices into k groups with near-equal total seqlen.

    equal_size=True additionally requires equal item counts per group
    (needed for DP dispatch where each rank must get the same batch size).
    """
    n = len(seqlens)
    assert n >= k_partitions, f"{n} items < {k_partitions} partitions"
    if equal_size:
        assert n % k_partitions == 0
        per = n // k_partitions
        # sort desc, round-robin into the currently lightest non-full bucket
        order = sorted(range(n), key=lambda i: -seqlens[i])
        heap: List[Tuple[int, int]] = [(0, g) for g in range(k_partitions)]
        heapq.heapify(heap)
        groups: List[List[int]] = [[] for _ in range(k_partitions)]
        for idx in order:
            # pop until we find a non-full bucket
            popped = []
            while True:
                tot, g = heapq.heappop(heap)
                if len(groups[g]) < per:
                    break
                popped.append((tot, g))
            groups[g].append(idx)
            heapq.heappush(heap, (tot + seqlens[idx], g))
            for item in popped:
                heapq.heappush(heap, item)
        return [sorted(g) for g in groups]
    # unequal sizes: plain greedy into lightest bucket
    order = sorted(range(n), key=lambda i: -seqlens[i])
    heap = [(0, g) for g in range(k_partitions)]
    heapq.heapify(heap)
    groups = [[] for _ in range(k_partitions)]
    for idx in order:
        tot, g = heapq.heappop(heap)
        groups[g].append(idx)
        heapq.heappush(heap, (tot + seqlens[idx], g))
    return [sorted(g) for g in groups]


def balance_batch_indices(attention_mask: torch.Tensor, world_size: int) -> torch.Tensor:
    """Reorder a global batch so each DP rank gets a near-equal token count.

    Returns the permutation (global indices).  The caller slices the reordered
    batch contiguously per rank.  (reference semantics: _balance_batch)."""
    seqlens = attention_mask.sum(dim=-1).tolist()
    parts = get_seqlen_balanced_partitions(seqlens, world_size, equal_size=True)
    idx = [i for part in parts for i in part]
    return torch.tensor(idx, dtype=torch.long)


def _dense_partitions(seqlens: List[int], budget: int) -> List[List[int]]:
    """First-fit-decreasing bin packing: fill micros to ~the budget, leave
    ONE small tail instead of k equally-underfull micros.  Measured on
    MI355X (profiles/PROFILES.md r2): balanced partitions at budget 8192
    produce 4 x ~6960-token micros that fixed-M padding rounds up to 8192
    each — ~15% padded GEMM work; dense packing makes that pad ~free."""
    order = sorted(range(len(seqlens)), key=lambda i: -seqlens[i])
    bins: List[List[int]] = []
    loads: List[int] = []
    for idx in order:
        s = seqlens[idx]
        placed = False
        for b in range(len(bins)):
            if loads[b] + s <= budget:
                bins[b].append(idx)
                loads[b] += s
                placed = True
                break
        if not placed:
            bins.append([idx])
            loads.append(s)
    return [sorted(b) for b in bins]


def prepare_dynamic_batch(batch: TensorBatch, max_token_len: int,
                          packing: str = "dense"
                          ) -> Tuple[List[TensorBatch], List[List[int]]]:
    """Split a batch into micro-batches bounded by a per-micro-batch token
    budget.  ``packing``: "dense" (default — first-fit-decreasing, micros
    ~full + one tail) or "balanced" (near-equal micro sizes, the reference
    seqlen_balancing behavior).

    Returns (micro_batches, index_lists); restore_dynamic_batch inverts it.
    """
    attention_mask = batch["attention_mask"]
    seqlens = [int(s) for s in attention_mask.sum(dim=-1).tolist()]
    max_seq = max(seqlens) if seqlens else 0
    assert max_seq <= max_token_len, \
        f"one sample has {max_seq} tokens > budget {max_token_len}"
    if packing == "dense":
        parts = _dense_partitions(seqlens, max_token_len)
    else:
        total = sum(seqlens)
        k = max(1, -(-total // max_token_len))  # ceil
        # grow k until every partition fits the budget
        while k <= len(seqlens):
            parts = get_seqlen_balanced_partitions(
                seqlens, min(k, len(seqlens)), equal_size=False)
            if all(sum(seqlens[i] for i in p) <= max_token_len
                   for p in parts):
                break
            k += 1
        parts = get_seqlen_balanced_partitions(seqlens, min(k, len(seqlens)),
                                               equal_size=False)
    micro = [batch.slice(torch.tensor(p, dtype=torch.long)) for p in parts]
    return micro, parts


def restore_dynamic_batch(tensor: torch.Tensor, index_lists: List[List[int]]
                          ) -> torch.Tensor:
    """Undo prepare_dynamic_batch ordering on a result tensor concatenated
    in micro-batch order."""
    flat = [i for part in index_lists for i in part]
    inv = torch.empty(len(flat), dtype=torch.long)
    inv[torch.tensor(flat)] = torch.arange(len(flat))
    return tensor[inv]


def fixed_micro_batches(batch: TensorBatch, micro_batch_size: int
                        ) -> Tuple[List[TensorBatch], List[List[int]]]:
    n = len(batch)
    parts = [list(range(i, min(i + micro_batch_size, n)))
             for i in range(0, n, micro_batch_size)]
    return [batch.slice(slice(p[0], p[-1] + 1)) for p in parts], parts
