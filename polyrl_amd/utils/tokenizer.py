"""Tokenizer loading (the reference's hf_tokenizer capability).

Offline-friendly: only loads when ``path`` is a local directory with
tokenizer files; otherwise returns None — parquet datasets that carry a
precomputed ``input_ids`` column (examples/data_preprocess/) need no
tokenizer at train time."""
from __future__ import annotations

import os
from typing import Callable, Optional


def get_tokenizer(path: str) -> Optional[Callable]:
    """Returns a callable text -> list[int], or None if no local tokenizer
    files exist at ``path`` (registry model names have none)."""
    if not os.path.isdir(path):
        return None
    has_tok = any(os.path.exists(os.path.join(path, f))
                  for f in ("tokenizer.json", "tokenizer.model",
                            "tokenizer_config.json"))
    if not has_tok:
        return None
    from transformers import AutoTokenizer
    tok = AutoTokenizer.from_pretrained(path)
    return lambda text: tok.encode(text)
