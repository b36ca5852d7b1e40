"""Synthetic OpenWebText/Alpaca-shaped data for benchmarking.

This environment has no network, so the benchmark path trains
on random token ids of the reference workload's shape (const-length packed
sequences, reference config/train/acco.yaml:4,16). Deterministic per index.
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset


class SyntheticCausalLMDataset(Dataset):
    def __init__(self, n_sequences: int, seq_len: int, vocab_size: int,
                 seed: int = 1234):
        self.n = int(n_sequences)
        self.seq_len = int(seq_len)
        self.vocab = int(vocab_size)
        self.seed = int(seed)

    def __len__(self) -> int:
        return self.n

    def __getitem__(self, idx: int):
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + idx)
        ids = torch.randint(0, self.vocab, (self.seq_len,), generator=g)
        return {"input_ids": ids}


def collate_input_ids(batch):
    """Stack const-length sequences (reference trainer_base.py:131-132)."""
    return {"input_ids": torch.stack(
        [torch.as_tensor(b["input_ids"], dtype=torch.long) for b in batch])}
