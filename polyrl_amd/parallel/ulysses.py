"""Ulysses sequence parallelism on RCCL over xGMI.

Reference capability: verl's `FSDPUlyssesShardingManager` /
`ulysses_pad_and_slice_inputs` / `gather_outputs_and_unpad`, entered around
actor/critic updates (stream_fsdp_workers.py:395,566).  Mechanism: each SP
rank holds a 1/sp sequence shard of the same batch rows; inside attention an
all-to-all exchanges the sequence shard for a head shard so every rank sees
the FULL sequence for Hq/sp heads — a perfect fit for the 8x MI355X xGMI
mesh, where all-to-all puts 1/sp of the bytes on each of the 7 p2p links
simultaneously.

Autograd: the backward of an all-to-all is the inverse all-to-all; the
backward of a sequence gather is a slice.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def _all_to_all_list(out, inp, group):
    """dist.all_to_all with a gloo fallback (sp all_gather rounds — CPU test
    tier only; RCCL has native alltoall on xGMI)."""
    if dist.get_backend(group) == "gloo":
        sp = dist.get_world_size(group)
        rank = dist.get_rank(group)
        for t in range(sp):
            lst = [torch.empty_like(inp[t]) for _ in range(sp)]
            dist.all_gather(lst, inp[t], group=group)
            if rank == t:
                for p in range(sp):
                    out[p].copy_(lst[p])
        return
    dist.all_to_all(out, inp, group=group)


class _AllToAll4D(torch.autograd.Function):
    """all-to-all on a 4-D tensor: scatter ``scatter_dim`` into sp chunks,
    gather ``gather_dim``."""

    @staticmethod
    def forward(ctx, x, scatter_dim: int, gather_dim: int, group):
        ctx.scatter_dim = scatter_dim
        ctx.gather_dim = gather_dim
        ctx.group = group
        sp = dist.get_world_size(group)
        inp = [c.contiguous() for c in x.chunk(sp, dim=scatter_dim)]
        out = [torch.empty_like(c) for c in inp]
        _all_to_all_list(out, inp, group)
        return torch.cat(out, dim=gather_dim)

    @staticmethod
    def backward(ctx, grad):
        sp = dist.get_world_size(ctx.group)
        inp = [c.contiguous() for c in grad.chunk(sp, dim=ctx.gather_dim)]
        out = [torch.empty_like(c) for c in inp]
        _all_to_all_list(out, inp, ctx.group)
        return torch.cat(out, dim=ctx.scatter_dim), None, None, None


def all_to_all_4d(x: torch.Tensor, scatter_dim: int, gather_dim: int,
                  group) -> torch.Tensor:
    return _AllToAll4D.apply(x, scatter_dim, gather_dim, group)


class _GatherSeq(torch.autograd.Function):
    """Autograd-aware all-gather along a sequence dim (equal shards)."""

    @staticmethod
    def forward(ctx, x, dim: int, group):
        ctx.dim = dim
        ctx.group = group
        sp = dist.get_world_size(group)
        ctx.rank = dist.get_rank(group)
        xs = [torch.empty_like(x) for _ in range(sp)]
        dist.all_gather(xs, x.contiguous(), group=group)
        xs[ctx.rank] = x  # keep the autograd-connected local shard
        return torch.cat(xs, dim=dim)

    @staticmethod
    def backward(ctx, grad):
        sp = dist.get_world_size(ctx.group)
        shard = grad.chunk(sp, dim=ctx.dim)[ctx.rank]
        return shard.contiguous(), None, None


def gather_seq(x: torch.Tensor, dim: int, group) -> torch.Tensor:
    return _GatherSeq.apply(x, dim, group)


def pad_to_multiple(x: torch.Tensor, multiple: int, dim: int,
                    value=0) -> torch.Tensor:
    L = x.shape[dim]
    pad = (-L) % multiple
    if pad == 0:
        return x
    shape = list(x.shape)
    shape[dim] = pad
    fill = torch.full(shape, value, dtype=x.dtype, device=x.device)
    return torch.cat([x, fill], dim=dim)


def slice_for_rank(x: torch.Tensor, dim: int, group) -> torch.Tensor:
    sp = dist.get_world_size(group)
    r = dist.get_rank(group)
    return x.chunk(sp, dim=dim)[r]


class UlyssesContext:
    """Holds the SP process group; installed on a model via
    ``model.model.ulysses = ctx`` (DecoderModel reads it)."""

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


def build_sp_groups(sp_size: int, ranks: Optional[List[int]] = None):
    """Partition ``ranks`` (default: the whole world) into contiguous SP
    groups of ``sp_size``; returns (sp_group, dp_group) for this rank —
    (None, None) if this rank is not in ``ranks``.  dp groups connect the
    same-sp-position ranks across SP groups (the FSDP data dimension).

    ``dist.new_group`` is a WORLD collective: EVERY rank of the default
    group must call this with the same arguments, even ranks outside
    ``ranks`` (e.g. disaggregated rollout ranks)."""
    if ranks is None:
        ranks = list(range(dist.get_world_size()))
    assert len(ranks) % sp_size == 0, \
        f"len(ranks) {len(ranks)} % sp {sp_size} != 0"
    rank = dist.get_rank()
    sp_group = dp_group = None
    for g0 in range(0, len(ranks), sp_size):
        rs = ranks[g0:g0 + sp_size]
        g = dist.new_group(rs)
        if rank in rs:
            sp_group = g
    for pos in range(sp_size):
        rs = ranks[pos::sp_size]
        g = dist.new_group(rs)
        if rank in rs:
            dp_group = g
    return sp_group, dp_group
