"""Const-length sequence packing and dataset preparation.

The packing algorithm mirrors the reference's
``tokenize_data_const_len`` (trainer_base.py:84-97 and dl_dataset.py:10-23):
tokenize without truncation, append EOS per document, concatenate all ids,
chop into max_length blocks, drop the remainder.
"""

from __future__ import annotations

from typing import Dict, List

import torch


def pack_const_len(token_id_lists: List[List[int]], eos_token_id: int,
                   max_length: int) -> torch.Tensor:
    """Pack a batch of variable-length token-id lists into [n, max_length]."""
    concat: List[int] = []
    for ids in token_id_lists:
        concat.extend(ids)
        concat.append(eos_token_id)
    n = len(concat) // max_length
    if n == 0:
        return torch.empty(0, max_length, dtype=torch.long)
    return torch.tensor(concat[:n * max_length],
                        dtype=torch.long).reshape(n, max_length)


def make_tokenize_const_len_fn(tokenizer, text_column: str, max_length: int):
    """datasets.map(batched=True) function producing packed input_ids."""
    def fn(batch: Dict) -> Dict:
        out = tokenizer(batch[text_column], truncation=False)
        packed = pack_const_len(out["input_ids"], tokenizer.eos_token_id,
                                max_length)
        return {"input_ids": packed.tolist()}
    return fn


def make_tokenize_truncate_fn(tokenizer, text_column: str, max_length: int):
    """Plain truncating tokenization (reference tokenize_data,
    trainer_base.py:77-82) — used with const_len_batch=false (finetuning)."""
    def fn(batch: Dict) -> Dict:
        return tokenizer(batch[text_column], truncation=True,
                         max_length=max_length)
    return fn
