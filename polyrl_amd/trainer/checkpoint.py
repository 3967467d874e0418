"""Sharded checkpointing, verl-compatible directory layout.

Layout:
    {root}/global_step_{N}/actor/model_world_size_{W}_rank_{r}.pt
                                /optim_world_size_{W}_rank_{r}.pt
                                /extra_state_world_size_{W}_rank_{r}.pt
                          /critic/...
    {root}/latest_checkpointed_iteration.txt

DTensor (FSDP2) params are stored as local shards + a resume contract of the
same world size — the same contract verl's sharded FSDP checkpoints have.
"""
from __future__ import annotations

import os
import re
from typing import Any, Dict, Optional

import torch
import torch.distributed as dist


def _world_rank(pg=None):
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(pg), dist.get_rank(pg)
    return 1, 0


def _localize(sd: Dict[str, Any]) -> Dict[str, Any]:
    out = {}
    for k, v in sd.items():
        if hasattr(v, "to_local"):           # DTensor -> local shard
            out[k] = v.to_local()
        elif isinstance(v, torch.Tensor):
            out[k] = v
        else:
            out[k] = v
    return out


def _copy_into(dst_sd: Dict[str, Any], src: Dict[str, Any]):
    for k, v in dst_sd.items():
        if k not in src:
            raise KeyError(f"checkpoint missing key {k}")
        s = src[k]
        if hasattr(v, "to_local"):
            v.to_local().copy_(s)
        elif isinstance(v, torch.Tensor):
            v.copy_(s)


def _load_optim_state(optimizer, src: Dict[str, Any]):
    """Structural load, then re-wrap plain local shards as DTensors for
    DTensor (FSDP2) params — Adam kernels refuse mixed Tensor/DTensor
    state (aten.lerp_ "mixed torch.Tensor and DTensor")."""
    optimizer.load_state_dict(src)
    try:
        from torch.distributed.tensor import DTensor
    except ImportError:
        return
    for group in optimizer.param_groups:
        for prm in group["params"]:
            st = optimizer.state.get(prm)
            if not st or not isinstance(prm, DTensor):
                continue
            for k, v in st.items():
                if torch.is_tensor(v) and v.dim() > 0 \
                        and not isinstance(v, DTensor):
                    st[k] = DTensor.from_local(
                        v.to(prm.device), prm.device_mesh, prm.placements)


class CheckpointManager:
    def __init__(self, root: str, role: str = "actor", pg=None):
        self.root = root
        self.role = role
        # collective scope: the trainer subgroup in disaggregated mode —
        # a WORLD barrier would deadlock (rollout ranks never save)
        self.pg = pg

    def _dir(self, step: int) -> str:
        return os.path.join(self.root, f"global_step_{step}", self.role)

    def save(self, step: int, model, optimizer=None, lr_scheduler=None,
             extra: Optional[Dict[str, Any]] = None):
        world, rank = _world_rank(self.pg)
        d = self._dir(step)
        os.makedirs(d, exist_ok=True)
        torch.save(_localize(model.state_dict()),
                   os.path.join(d, f"model_world_size_{world}_rank_{rank}.pt"))
        if optimizer is not None:
            osd = optimizer.state_dict()
            osd["state"] = {i: _localize(s) for i, s in osd["state"].items()}
            torch.save(osd,
                       os.path.join(d, f"optim_world_size_{world}_rank_{rank}.pt"))
        ex = dict(extra or {})
        if lr_scheduler is not None:
            ex["lr_scheduler"] = lr_scheduler.state_dict()
        ex["global_step"] = step
        torch.save(ex, os.path.join(
            d, f"extra_state_world_size_{world}_rank_{rank}.pt"))
        if rank == 0:
            with open(os.path.join(self.root,
                                   "latest_checkpointed_iteration.txt"), "w") as f:
                f.write(str(step))
        if dist.is_available() and dist.is_initialized():
            dist.barrier(group=self.pg)

    def load(self, model, optimizer=None, lr_scheduler=None,
             step: Optional[int] = None) -> Optional[Dict[str, Any]]:
        if step is None:
            step = self.latest_step()
            if step is None:
                return None
        world, rank = _world_rank(self.pg)
        d = self._dir(step)
        msd = torch.load(os.path.join(
            d, f"model_world_size_{world}_rank_{rank}.pt"),
            map_location="cpu", weights_only=False)
        _copy_into(model.state_dict(), msd)
        if optimizer is not None:
            p = os.path.join(d, f"optim_world_size_{world}_rank_{rank}.pt")
            if os.path.exists(p):
                _load_optim_state(optimizer,
                                  torch.load(p, map_location="cpu",
                                             weights_only=False))
        ex = torch.load(os.path.join(
            d, f"extra_state_world_size_{world}_rank_{rank}.pt"),
            map_location="cpu", weights_only=False)
        if lr_scheduler is not None and "lr_scheduler" in ex:
            lr_scheduler.load_state_dict(ex["lr_scheduler"])
        return ex

    def latest_step(self) -> Optional[int]:
        return find_latest_ckpt_step(self.root)


def find_latest_ckpt_step(root: str) -> Optional[int]:
    marker = os.path.join(root, "latest_checkpointed_iteration.txt")
    if os.path.exists(marker):
        with open(marker) as f:
            return int(f.read().strip())
    if not os.path.isdir(root):
        return None
    steps = [int(m.group(1)) for name in os.listdir(root)
             if (m := re.match(r"global_step_(\d+)$", name))]
    return max(steps) if steps else None


def find_latest_ckpt_path(root: str, role: str = "actor") -> Optional[str]:
    step = find_latest_ckpt_step(root)
    if step is None:
        return None
    return os.path.join(root, f"global_step_{step}", role)
