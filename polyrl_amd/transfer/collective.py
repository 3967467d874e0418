"""Cross-process weight plane: bucketed broadcast over RCCL/xGMI with
receiver-side TP resharding.

Reference capability being replaced: sender agent packs the
full state dict into a /dev/shm buffer, ships it over N TCP streams, and the
receiving instance H2D-uploads ≤2 GiB chunks and NCCL-broadcasts them across
its TP group (patches.py:167,196-241).  On one MI355X node all of that
becomes: pack params into ≤``bucket_bytes`` device buckets and
``dist.broadcast`` them from the trainer's source rank straight over xGMI;
every receiver slices its TP shard out of the bucket on device.  No host
staging, no serialization, no extra copies.

Works on gloo/CPU for the test tier (multi-process
weight-transfer round-trip, trainer buffer -> instance tensors bitwise).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

DEFAULT_BUCKET_BYTES = 1 << 30  # ≤ the reference's 2 GiB chunk cap


# ------------------------------------------------------------- TP sharding


def tp_shard_dim(name: str, arch: str = "llama") -> Optional[int]:
    """Which dim of the FULL parameter a TP rank slices (None = replicate).

    Row/col split mirrors the standard Megatron partitioning the reference's
    engines use: q/k/v and gate/up are row-parallel producers (dim 0),
    o_proj/down_proj are col-parallel consumers (dim 1); norms and (for now)
    embeddings replicate."""
    if arch == "gpt2":
        return None  # gpt2 serves the CPU plumbing tier only: replicate
    if name.endswith(("q_proj.weight", "k_proj.weight", "v_proj.weight",
                      "gate_proj.weight", "up_proj.weight")):
        return 0
    if name.endswith(("q_proj.bias", "k_proj.bias", "v_proj.bias")):
        return 0
    if name.endswith(("o_proj.weight", "down_proj.weight")):
        return 1
    return None


def tp_slice(name: str, full: torch.Tensor, tp_rank: int, tp_size: int,
             arch: str = "llama") -> torch.Tensor:
    if tp_size <= 1:
        return full
    dim = tp_shard_dim(name, arch)
    if dim is None:
        return full
    n = full.shape[dim]
    assert n % tp_size == 0, f"{name}: dim {dim} size {n} % tp {tp_size} != 0"
    step = n // tp_size
    idx = slice(tp_rank * step, (tp_rank + 1) * step)
    return full[idx] if dim == 0 else full[:, idx]


# -------------------------------------------------------------- meta plan


@dataclass
class BucketPlan:
    """Deterministic packing plan shared by sender and receivers."""
    names: List[str]
    shapes: List[Tuple[int, ...]]
    dtypes: List[str]
    buckets: List[List[int]]        # bucket -> param indices

    @staticmethod
    def build(named: List[Tuple[str, Tuple[int, ...], torch.dtype]],
              bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> "BucketPlan":
        names, shapes, dtypes, buckets = [], [], [], []
        cur, cur_bytes = [], 0
        for i, (n, shape, dt) in enumerate(named):
            nbytes = int(torch.tensor([], dtype=dt).element_size()
                         * int(torch.prod(torch.tensor(shape)).item() or 1))
            if cur and cur_bytes + nbytes > bucket_bytes:
                buckets.append(cur)
                cur, cur_bytes = [], 0
            names.append(n)
            shapes.append(tuple(shape))
            dtypes.append(str(dt).replace("torch.", ""))
            cur.append(i)
            cur_bytes += nbytes
        if cur:
            buckets.append(cur)
        return BucketPlan(names, shapes, dtypes, buckets)

    def dtype_of(self, i: int) -> torch.dtype:
        return getattr(torch, self.dtypes[i])

    def numel_of(self, i: int) -> int:
        n = 1
        for s in self.shapes[i]:
            n *= s
        return n


class CollectiveWeightPlane:
    """Versioned bucketed broadcast of full weights from ``src`` to every
    rank of ``group``; receivers apply (optionally TP-sliced) views.

    The plan is exchanged once (broadcast_object_list); each ``publish`` /
    ``receive`` moves only tensor bytes."""

    def __init__(self, group=None, src: int = 0, device="cpu",
                 bucket_bytes: int = DEFAULT_BUCKET_BYTES):
        self.group = group
        self.src = src              # GLOBAL rank of the sender
        self.device = device
        self.bucket_bytes = bucket_bytes
        self.plan: Optional[BucketPlan] = None
        self.version = 0

    # ------------------------------------------------------------- sender
    def publish(self, named_params: Dict[str, torch.Tensor],
                version: Optional[int] = None) -> int:
        """Sender side.  ``named_params`` values may be DTensors (FSDP
        shards) — they are all-gathered per bucket entry via full_tensor()."""
        rank = dist.get_rank()
        if self.plan is None:
            meta = [(n, tuple(p.shape), p.dtype)
                    for n, p in named_params.items()]
            self.plan = BucketPlan.build(meta, self.bucket_bytes)
            dist.broadcast_object_list([self.plan], src=self.src,
                                       group=self.group)
        self.version = version if version is not None else self.version + 1
        dist.broadcast_object_list([self.version], src=self.src,
                                   group=self.group)
        params = list(named_params.values())
        for bucket in self.plan.buckets:
            for dt, flat in self._pack(bucket, params):
                dist.broadcast(flat, src=self.src, group=self.group)
        return self.version

    def _pack(self, bucket: List[int], params: List[torch.Tensor]):
        """Pack one bucket grouped by dtype into contiguous flats."""
        by_dtype: Dict[torch.dtype, List[torch.Tensor]] = {}
        for i in bucket:
            p = params[i]
            full = p.full_tensor() if hasattr(p, "full_tensor") else p
            by_dtype.setdefault(full.dtype, []).append(
                full.detach().reshape(-1))
        for dt in sorted(by_dtype, key=str):
            flat = torch.cat(by_dtype[dt]).to(self.device)
            yield dt, flat

    # ----------------------------------------------------------- receiver
    def receive(self, apply_fn: Callable[[str, torch.Tensor], None],
                tp_rank: int = 0, tp_size: int = 1, arch: str = "llama"
                ) -> int:
        """Receiver side.  ``apply_fn(name, tensor)`` gets the (TP-sliced)
        full-precision view for every parameter, on ``self.device``."""
        if self.plan is None:
            box = [None]
            dist.broadcast_object_list(box, src=self.src, group=self.group)
            self.plan = box[0]
        vbox = [None]
        dist.broadcast_object_list(vbox, src=self.src, group=self.group)
        self.version = int(vbox[0])
        plan = self.plan
        for bucket in plan.buckets:
            by_dtype: Dict[str, List[int]] = {}
            for i in bucket:
                by_dtype.setdefault(plan.dtypes[i], []).append(i)
            for dts in sorted(by_dtype):
                idxs = by_dtype[dts]
                dt = getattr(torch, dts)
                total = sum(plan.numel_of(i) for i in idxs)
                flat = torch.empty(total, dtype=dt, device=self.device)
                dist.broadcast(flat, src=self.src, group=self.group)
                off = 0
                for i in idxs:
                    n = plan.numel_of(i)
                    full = flat[off:off + n].view(plan.shapes[i])
                    off += n
                    apply_fn(plan.names[i],
                             tp_slice(plan.names[i], full, tp_rank, tp_size,
                                      arch))
        return self.version
