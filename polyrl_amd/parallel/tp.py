"""Tensor-parallel rollout decoder support.

Reference capability (SURVEY.md §2.3 TP row): the rollout engine's TP mode —
row-parallel q/k/v + gate/up producers, col-parallel o/down consumers with an
all-reduce after each (RCCL over xGMI; ring all-reduce is per-link bound, so
the two all-reduces per layer move 2*N*H bytes per step), vocab-parallel
lm_head with an all-gather before sampling.  Weight delivery re-uses
transfer.collective.tp_slice (receiver-side resharding, the
patches.py:196-241 contract).

Determinism note: all TP ranks run the same scheduling decisions and the
same fused sampling kernel on identical (all-reduced / gathered) logits with
the same seed, so sampled tokens agree bitwise across ranks with no token
broadcast.
"""
from __future__ import annotations

import torch
import torch.distributed as dist


class TPContext:
    def __init__(self, group=None):
        self.group = group

    @property
    def enabled(self) -> bool:
        return self.group is not None and dist.get_world_size(self.group) > 1

    @property
    def size(self) -> int:
        return dist.get_world_size(self.group) if self.group is not None else 1

    @property
    def rank(self) -> int:
        return dist.get_rank(self.group) if self.group is not None else 0

    def all_reduce_(self, x: torch.Tensor) -> torch.Tensor:
        if self.enabled:
            dist.all_reduce(x, group=self.group)
        return x

    def all_gather_cat(self, x: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if not self.enabled:
            return x
        xs = [torch.empty_like(x) for _ in range(self.size)]
        dist.all_gather(xs, x.contiguous(), group=self.group)
        return torch.cat(xs, dim=dim)
