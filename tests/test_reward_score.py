"""Rule-based reward scorers + naive manager + parquet ground-truth flow
(reference: verl_stream/utils/reward_score dispatch)."""
import numpy as np
import pytest
import torch

from polyrl_amd.protocol import TensorBatch
from polyrl_amd.reward import load_reward_manager
from polyrl_amd.reward_score import default_compute_score
from polyrl_amd.reward_score import gsm8k, math_score


def test_gsm8k_extraction():
    assert gsm8k.compute_score("blah blah #### 42", "42") == 1.0
    assert gsm8k.compute_score("#### 1,234", "1234") == 1.0
    assert gsm8k.compute_score("#### $18.00", "18") == 1.0
    assert gsm8k.compute_score("#### 41", "42") == 0.0
    # flexible fallback: last number
    assert gsm8k.compute_score("the answer is 7.", "7") == 1.0
    assert gsm8k.compute_score("no numbers here", "7") == 0.0


def test_math_boxed():
    assert math_score.compute_score(r"so \boxed{\frac{1}{2}}", r"\frac{1}{2}") == 1.0
    assert math_score.compute_score(r"\boxed{ \frac{1}{2} }", r"\tfrac{1}{2}") == 1.0
    assert math_score.compute_score(r"\boxed{3.0}", "3") == 1.0
    assert math_score.compute_score(r"nested \boxed{\sqrt{x^{2}}}", r"\sqrt{x^{2}}") == 1.0
    assert math_score.compute_score(r"\boxed{5}", "6") == 0.0


def test_dispatch():
    assert default_compute_score("openai/gsm8k", "#### 3", "3") == 1.0
    assert default_compute_score("lighteval/MATH", r"\boxed{3}", "3") == 1.0
    with pytest.raises(KeyError):
        default_compute_score("unknown_source", "x", "y")


class ToyTok:
    """Maps token id -> char; decode joins (deterministic test text)."""
    def decode(self, ids):
        return "".join(chr(ord('0') + (i % 75)) for i in ids)


def test_naive_manager_places_scores():
    # craft responses whose decoded text ends in '#### <digit>'
    tok = ToyTok()
    # token ids for '#', '#', '#', '#', ' ', '3' under ToyTok: chr(48+i%75)
    def enc(s):
        return [ord(c) - 48 for c in s]
    good = enc("#### 3")
    bad = enc("#### 4")
    Lr = 8
    resp = torch.zeros(2, Lr, dtype=torch.long)
    mask = torch.zeros(2, Lr, dtype=torch.long)
    resp[0, :len(good)] = torch.tensor(good)
    mask[0, :len(good)] = 1
    resp[1, :len(bad)] = torch.tensor(bad)
    mask[1, :len(bad)] = 1
    batch = TensorBatch(
        tensors={"responses": resp, "response_mask": mask},
        non_tensors={
            "data_source": np.array(["gsm8k", "gsm8k"], dtype=object),
            "ground_truth": np.array(["3", "3"], dtype=object),
        })
    rm = load_reward_manager("naive", tokenizer=tok)
    scores = rm(batch)
    assert scores[0].sum() == 1.0
    assert scores[1].sum() == 0.0
    # score sits on the LAST valid response token
    assert scores[0, len(good) - 1] == 1.0


def test_parquet_ground_truth_roundtrip(tmp_path):
    import pandas as pd
    from polyrl_amd.data import ParquetRLHFDataset
    df = pd.DataFrame({
        "prompt": ["1+2=", "2+2="],
        "input_ids": [[1, 2, 3], [4, 5]],
        "data_source": ["gsm8k", "gsm8k"],
        "reward_model": [{"ground_truth": "3"}, {"ground_truth": "4"}],
    })
    f = str(tmp_path / "d.parquet")
    df.to_parquet(f)
    ds = ParquetRLHFDataset([f], max_prompt_length=8, input_ids_key="input_ids")
    b = ds.batch([0, 1])
    assert list(b.non_tensors["ground_truth"]) == ["3", "4"]
    assert list(b.non_tensors["data_source"]) == ["gsm8k", "gsm8k"]
    assert b["input_ids"].shape == (2, 8)


def test_gsm8k_preprocess_to_dataset_roundtrip(tmp_path):
    """examples/data_preprocess/gsm8k.py writes the schema
    ParquetRLHFDataset + the gsm8k scorer consume (reference:
    examples/data_preprocess/openr1.py capability)."""
    import subprocess
    import sys

    out = tmp_path / "g.parquet"
    r = subprocess.run(
        [sys.executable, "examples/data_preprocess/gsm8k.py",
         "--synthetic", "6", "--out", str(out)],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-2000:]
    from polyrl_amd.data import ParquetRLHFDataset
    from polyrl_amd.reward_score import default_compute_score
    ds = ParquetRLHFDataset([str(out)], max_prompt_length=64,
                            tokenizer=lambda s: [ord(c) % 512 for c in s][:64])
    b = ds.batch([0, 1, 2])
    gts = b.non_tensors["ground_truth"]
    assert default_compute_score("openai/gsm8k", f"x #### {gts[0]}",
                                 str(gts[0])) == 1.0
    assert default_compute_score("openai/gsm8k", "x #### 999999",
                                 str(gts[0])) == 0.0


def test_custom_reward_function_from_file(tmp_path):
    """reward manager 'custom' loads a scoring fn from a python file
    (the reference's custom_reward_function config)."""
    import torch

    from polyrl_amd.protocol import TensorBatch
    from polyrl_amd.reward import load_reward_manager

    f = tmp_path / "score.py"
    f.write_text(
        "def compute_score(sample, bonus=0.0):\n"
        "    return float(sample['responses'].sum()) + bonus\n")
    rm = load_reward_manager("custom", path=str(f), bonus=1.0)
    b = TensorBatch.from_dict(tensors={
        "responses": torch.tensor([[1, 2], [3, 4]]),
        "response_mask": torch.ones(2, 2, dtype=torch.long)})
    scores = rm(b)
    # sequence score lands on the last response token
    assert scores.shape == (2, 2)
    assert float(scores[0].sum()) == 4.0 and float(scores[1].sum()) == 8.0
