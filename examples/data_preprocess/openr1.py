"""OpenR1-Math-220k → parquet in the schema ParquetRLHFDataset consumes
(reference capability: examples/data_preprocess/openr1.py — take the first
10k rows, 90/10 train/test split, append the boxed-answer instruction, tag
data_source for the math scorer).

    # from a local snapshot (HF datasets layout or a jsonl export)
    python examples/data_preprocess/openr1.py \
        --input /path/to/openr1.jsonl --out-dir /data/openr1

    # offline smoke: synthetic rows in the same schema
    python examples/data_preprocess/openr1.py --synthetic 64 --out-dir /tmp/o

Input rows need {"problem", "answer"} (the OpenR1 'extended' config
columns); --input accepts a .jsonl file or a directory of parquet shards.

Output (train.parquet / test.parquet):
    prompt        str  — problem + the reference's instruction suffix
    input_ids     list[int] (only when a tokenizer is given)
    data_source   str  — "open-r1/OpenR1-Math-220k" (math scorer dispatch)
    ground_truth  str  — the final answer
    extra_info    dict — {"split", "index"}
"""
from __future__ import annotations

import argparse
import json
import os

DATA_SOURCE = "open-r1/OpenR1-Math-220k"
INSTR = (" Please reason step by step, and put your final answer within "
         "\\boxed{}.")


def load_rows(path: str):
    if os.path.isdir(path):
        import pandas as pd
        rows = []
        for f in sorted(os.listdir(path)):
            if f.endswith(".parquet"):
                rows.extend(pd.read_parquet(os.path.join(path, f))
                            .to_dict("records"))
        return rows
    with open(path) as f:
        return [json.loads(line) for line in f if line.strip()]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--input", default=None,
                   help="local OpenR1 jsonl or parquet-shard dir")
    p.add_argument("--synthetic", type=int, default=0,
                   help="write N synthetic rows instead (offline smoke)")
    p.add_argument("--tokenizer", default=None)
    p.add_argument("--out-dir", required=True)
    p.add_argument("--max-rows", type=int, default=10000,
                   help="reference takes the first 10k rows")
    p.add_argument("--test-frac", type=float, default=0.1)
    args = p.parse_args()

    if args.synthetic:
        rows = []
        for i in range(args.synthetic):
            a, b = 2 + i % 9, 3 + i % 6
            rows.append({"problem": f"Compute {a} * {b}.",
                         "answer": str(a * b)})
    else:
        assert args.input, "--input or --synthetic required"
        rows = load_rows(args.input)
    rows = rows[: args.max_rows]

    tok = None
    if args.tokenizer:
        from transformers import AutoTokenizer
        tok = AutoTokenizer.from_pretrained(args.tokenizer)

    n_test = max(1, int(len(rows) * args.test_frac)) if len(rows) > 1 else 0
    splits = {"train": rows[: len(rows) - n_test],
              "test": rows[len(rows) - n_test:]}
    os.makedirs(args.out_dir, exist_ok=True)
    import pandas as pd
    for split, rs in splits.items():
        out = []
        for idx, r in enumerate(rs):
            prompt = str(r["problem"]) + INSTR
            rec = {"prompt": prompt,
                   "data_source": DATA_SOURCE,
                   "ground_truth": str(r["answer"]),
                   "extra_info": {"split": split, "index": idx}}
            if tok is not None:
                rec["input_ids"] = tok.encode(prompt)
            out.append(rec)
        path = os.path.join(args.out_dir, f"{split}.parquet")
        pd.DataFrame(out).to_parquet(path)
        print(f"wrote {len(out)} rows -> {path}")
    # one example for eyeballing, like the reference
    if splits["train"]:
        with open(os.path.join(args.out_dir, "train_example.json"), "w") as f:
            ex = dict(splits["train"][0])
            json.dump({"problem": str(ex.get("problem"))[:500],
                       "answer": str(ex.get("answer"))[:200]}, f, indent=2)


if __name__ == "__main__":
    main()
