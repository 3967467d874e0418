"""GSM8K → parquet in the schema ParquetRLHFDataset / NaiveRewardManager
consume (reference capability: examples/data_preprocess/openr1.py — a
dataset-to-parquet preprocessor with data_source + ground_truth columns).

    python examples/data_preprocess/gsm8k.py \
        --input /path/to/gsm8k/main/train.jsonl \
        --tokenizer /path/to/hf_tokenizer_dir \
        --out /data/gsm8k_train.parquet

Offline-friendly: --input is a local jsonl with {"question", "answer"} rows
(the HF datasets layout); with --synthetic N it instead writes N synthetic
rows so the full pipeline (dataset -> trainer -> NaiveRewardManager gsm8k
scorer) can be exercised without any downloads.

Output columns:
    prompt        str  — question with the reference's instruction suffix
    input_ids     list[int] (only when a tokenizer is given)
    data_source   str  — "openai/gsm8k" (drives reward_score dispatch)
    ground_truth  str  — the #### answer, extracted like the scorer expects
"""
from __future__ import annotations

import argparse
import json
import re

INSTR = " Let's think step by step and output the final answer after \"####\"."


def extract_gt(answer: str) -> str:
    m = re.search(r"####\s*([\-0-9\.,]+)", answer)
    assert m, f"no #### answer in {answer[:80]!r}"
    return m.group(1).replace(",", "").strip()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--input", default=None, help="local gsm8k jsonl")
    p.add_argument("--synthetic", type=int, default=0,
                   help="write N synthetic rows instead (offline smoke)")
    p.add_argument("--tokenizer", default=None,
                   help="HF tokenizer dir (optional; adds input_ids)")
    p.add_argument("--out", required=True)
    args = p.parse_args()

    rows = []
    if args.synthetic:
        for i in range(args.synthetic):
            a, b = 3 + i % 7, 4 + i % 5
            rows.append({"question": f"Tom has {a} apples and buys {b} more. "
                                     f"How many apples does he have?",
                         "answer": f"{a} + {b} = {a + b}\n#### {a + b}"})
    else:
        assert args.input, "--input or --synthetic required"
        with open(args.input) as f:
            rows = [json.loads(line) for line in f if line.strip()]

    tok = None
    if args.tokenizer:
        from transformers import AutoTokenizer
        tok = AutoTokenizer.from_pretrained(args.tokenizer)

    out = []
    for r in rows:
        rec = {"prompt": r["question"] + INSTR,
               "data_source": "openai/gsm8k",
               "ground_truth": extract_gt(r["answer"])}
        if tok is not None:
            rec["input_ids"] = tok.encode(rec["prompt"])
        out.append(rec)

    import pandas as pd
    pd.DataFrame(out).to_parquet(args.out)
    print(f"wrote {len(out)} rows -> {args.out}")


if __name__ == "__main__":
    main()
