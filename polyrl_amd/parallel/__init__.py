"""Parallelism strategies beyond FSDP data-parallel.

  * ulysses — Ulysses sequence parallelism (all-to-all seq<->heads inside
    attention), the reference stack's only long-context strategy
    (verl's FSDPUlyssesShardingManager capability).
  * tp — tensor-parallel rollout decoder helpers (SURVEY.md §2.3 TP row).
"""
from .ulysses import (UlyssesContext, all_to_all_4d, gather_seq,
                      pad_to_multiple, slice_for_rank)

__all__ = ["UlyssesContext", "all_to_all_4d", "gather_seq",
           "pad_to_multiple", "slice_for_rank"]
