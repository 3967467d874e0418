"""TensorBatch — the batch protocol that moves data between every layer.

Equivalent in capability to verl's ``DataProto`` (reference:
rlboost/verl_stream/trainer/ppo/stream_ray_trainer.py:41-43 consumes union /
select / split / concat / slicing / .to(device) / pad_unpad), rebuilt without
tensordict: a plain dict of same-batch-dim tensors + a dict of non-tensor
numpy arrays + a meta_info dict.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

import numpy as np
import torch


def _batch_size_of(tensors: Dict[str, torch.Tensor],
                   non_tensors: Dict[str, np.ndarray]) -> Optional[int]:
    for v in tensors.values():
        return v.shape[0]
    for v in non_tensors.values():
        return v.shape[0]
    return None


@dataclass
class TensorBatch:
    """A batch: tensors sharing dim-0, per-sample numpy arrays, and metadata."""

    tensors: Dict[str, torch.Tensor] = field(default_factory=dict)
    non_tensors: Dict[str, np.ndarray] = field(default_factory=dict)
    meta_info: Dict[str, Any] = field(default_factory=dict)

    def __post_init__(self):
        self.check_consistency()

    # ------------------------------------------------------------------ util
    def check_consistency(self):
        bs = _batch_size_of(self.tensors, self.non_tensors)
        if bs is None:
            return
        for k, v in self.tensors.items():
            assert v.shape[0] == bs, f"tensor {k!r} batch {v.shape[0]} != {bs}"
        for k, v in self.non_tensors.items():
            assert isinstance(v, np.ndarray), f"non_tensor {k!r} must be ndarray"
            assert v.shape[0] == bs, f"non_tensor {k!r} batch {v.shape[0]} != {bs}"

    def __len__(self) -> int:
        bs = _batch_size_of(self.tensors, self.non_tensors)
        return 0 if bs is None else bs

    @property
    def batch_size(self) -> int:
        return len(self)

    def keys(self):
        return list(self.tensors.keys()) + list(self.non_tensors.keys())

    def __contains__(self, key: str) -> bool:
        return key in self.tensors or key in self.non_tensors

    def __getitem__(self, item):
        if isinstance(item, str):
            if item in self.tensors:
                return self.tensors[item]
            return self.non_tensors[item]
        # slice / index / bool mask / index tensor -> sub-batch
        return self.slice(item)

    def __setitem__(self, key: str, value):
        if isinstance(value, torch.Tensor):
            self.tensors[key] = value
        elif isinstance(value, np.ndarray):
            self.non_tensors[key] = value
        else:
            raise TypeError(f"unsupported value type {type(value)} for {key!r}")

    # ----------------------------------------------------------- constructors
    @classmethod
    def from_dict(cls, tensors: Optional[Dict[str, torch.Tensor]] = None,
                  non_tensors: Optional[Dict[str, Any]] = None,
                  meta_info: Optional[Dict[str, Any]] = None) -> "TensorBatch":
        nt = {}
        for k, v in (non_tensors or {}).items():
            if not isinstance(v, np.ndarray):
                v = np.array(v, dtype=object)
            nt[k] = v
        return cls(tensors=dict(tensors or {}), non_tensors=nt,
                   meta_info=dict(meta_info or {}))

    # ------------------------------------------------------------- transforms
    def to(self, device, non_blocking: bool = False) -> "TensorBatch":
        self.tensors = {k: v.to(device, non_blocking=non_blocking)
                        for k, v in self.tensors.items()}
        return self

    def select(self, tensor_keys: Optional[Sequence[str]] = None,
               non_tensor_keys: Optional[Sequence[str]] = None,
               deepcopy_meta: bool = False) -> "TensorBatch":
        t = ({k: self.tensors[k] for k in tensor_keys}
             if tensor_keys is not None else dict(self.tensors))
        nt = ({k: self.non_tensors[k] for k in non_tensor_keys}
              if non_tensor_keys is not None else dict(self.non_tensors))
        meta = copy.deepcopy(self.meta_info) if deepcopy_meta else dict(self.meta_info)
        return TensorBatch(tensors=t, non_tensors=nt, meta_info=meta)

    def pop(self, tensor_keys: Sequence[str] = (),
            non_tensor_keys: Sequence[str] = ()) -> "TensorBatch":
        t = {k: self.tensors.pop(k) for k in tensor_keys}
        nt = {k: self.non_tensors.pop(k) for k in non_tensor_keys}
        return TensorBatch(tensors=t, non_tensors=nt, meta_info=dict(self.meta_info))

    def union(self, other: "TensorBatch") -> "TensorBatch":
        """Merge ``other`` into self (key collision requires identical shape)."""
        for k, v in other.tensors.items():
            if k in self.tensors and self.tensors[k].shape != v.shape:
                raise ValueError(f"union collision on tensor {k!r}")
            self.tensors[k] = v
        for k, v in other.non_tensors.items():
            self.non_tensors[k] = v
        self.meta_info.update(other.meta_info)
        self.check_consistency()
        return self

    def slice(self, idx) -> "TensorBatch":
        if isinstance(idx, int):
            idx = slice(idx, idx + 1)
        np_idx = idx
        if isinstance(idx, torch.Tensor):
            np_idx = idx.cpu().numpy()
        t = {k: v[idx] for k, v in self.tensors.items()}
        nt = {k: v[np_idx] for k, v in self.non_tensors.items()}
        return TensorBatch(tensors=t, non_tensors=nt, meta_info=dict(self.meta_info))

    def split(self, split_size: int) -> List["TensorBatch"]:
        n = len(self)
        return [self.slice(slice(i, min(i + split_size, n)))
                for i in range(0, n, split_size)]

    def chunk(self, chunks: int) -> List["TensorBatch"]:
        n = len(self)
        assert n % chunks == 0, f"batch {n} not divisible into {chunks} chunks"
        return self.split(n // chunks)

    @classmethod
    def concat(cls, batches: Sequence["TensorBatch"]) -> "TensorBatch":
        assert len(batches) > 0
        t: Dict[str, torch.Tensor] = {}
        nt: Dict[str, np.ndarray] = {}
        for k in batches[0].tensors:
            t[k] = torch.cat([b.tensors[k] for b in batches], dim=0)
        for k in batches[0].non_tensors:
            nt[k] = np.concatenate([b.non_tensors[k] for b in batches], axis=0)
        meta = dict(batches[0].meta_info)
        return cls(tensors=t, non_tensors=nt, meta_info=meta)

    def repeat(self, repeat_times: int, interleave: bool = True) -> "TensorBatch":
        """Unroll n samples per prompt (reference: rollout payload builder,
        sglang_rollout_remote.py:198-225 unrolls n samples/prompt)."""
        if interleave:
            t = {k: v.repeat_interleave(repeat_times, dim=0)
                 for k, v in self.tensors.items()}
            nt = {k: np.repeat(v, repeat_times, axis=0)
                  for k, v in self.non_tensors.items()}
        else:
            t = {k: v.repeat(repeat_times, *([1] * (v.dim() - 1)))
                 for k, v in self.tensors.items()}
            nt = {k: np.tile(v, (repeat_times,) + (1,) * (v.ndim - 1))
                  for k, v in self.non_tensors.items()}
        return TensorBatch(tensors=t, non_tensors=nt, meta_info=dict(self.meta_info))

    def reorder(self, indices: torch.Tensor) -> "TensorBatch":
        return self.slice(indices)

    # -------------------------------------------------- pad/unpad for dispatch
    def pad_to_divisor(self, divisor: int) -> "TensorBatch":
        """Pad by repeating leading rows so len % divisor == 0; records pad size."""
        n = len(self)
        pad = (-n) % divisor
        if pad == 0:
            out = self.select()
            out.meta_info["_pad_size"] = 0
            return out
        idx = torch.arange(n + pad) % n
        out = self.slice(idx)
        out.meta_info["_pad_size"] = pad
        return out

    def unpad(self) -> "TensorBatch":
        pad = self.meta_info.pop("_pad_size", 0)
        if pad == 0:
            return self
        return self.slice(slice(0, len(self) - pad))

    def __repr__(self):
        tk = {k: tuple(v.shape) for k, v in self.tensors.items()}
        nk = {k: tuple(v.shape) for k, v in self.non_tensors.items()}
        return f"TensorBatch(n={len(self)}, tensors={tk}, non_tensors={nk}, meta={list(self.meta_info)})"
