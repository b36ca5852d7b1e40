"""Offline dataset tokenizer (reference dl_dataset.py:8-34 capability):
tokenize a HF dataset to const-length packed sequences and save to disk.

    python dl_dataset.py data=openwebtext model=gptneo \
        train.max_length=1024 out_dir=./tokenized/openwebtext
"""

from __future__ import annotations

import sys

from acco_amd.config import load_config
from acco_amd.data.packing import make_tokenize_const_len_fn


def main(argv=None):
    cfg = load_config(argv if argv is not None else sys.argv[1:])
    out_dir = cfg.get("out_dir", "./tokenized")

    import datasets
    from transformers import AutoTokenizer

    from acco_amd.data import load_raw_dataset, tokenizer_path

    ds = load_raw_dataset(cfg.data)
    ds = ds["train"].train_test_split(0.05, seed=42)   # reference main.py:50
    tokenizer = AutoTokenizer.from_pretrained(tokenizer_path(cfg))
    tokenizer.pad_token_id = tokenizer.eos_token_id

    fn = make_tokenize_const_len_fn(tokenizer, "text", cfg.train.max_length)
    out = {}
    for split in ("train", "test"):
        out[split] = ds[split].map(
            fn, batched=True, remove_columns=ds[split].column_names,
            num_proc=cfg.train.dataloader_num_workers or 1)
    packed = datasets.DatasetDict(out)
    # out_shards=N splits each split into N arrow shards (parallel loaders /
    # per-rank file mapping on big corpora); 0/absent keeps one file
    shards = int(cfg.get("out_shards", 0) or 0)
    if shards > 1:
        packed.save_to_disk(out_dir,
                            num_shards={s: shards for s in packed})
    else:
        packed.save_to_disk(out_dir)
    print(f"saved packed dataset to {out_dir}"
          + (f" ({shards} shards/split)" if shards > 1 else ""))


if __name__ == "__main__":
    main()
