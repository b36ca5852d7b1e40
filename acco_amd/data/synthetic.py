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


def make_padded_collator(pad_token_id: int, pad_to_multiple: int = 64):
    """Ragged-batch collator for the finetune path (const_len_batch=false):
    right-pads input_ids and masks padded positions with -100 in labels —
    the behaviour of the reference's DataCollatorForLanguageModeling
    (trainer_base.py:209). Padding to a multiple of 64 keeps the flash
    kernels' S%64 contract."""
    def collate(batch):
        seqs = [torch.as_tensor(b["input_ids"], dtype=torch.long)
                for b in batch]
        longest = max(s.numel() for s in seqs)
        target = ((longest + pad_to_multiple - 1) // pad_to_multiple
                  * pad_to_multiple)
        ids = torch.full((len(seqs), target), pad_token_id, dtype=torch.long)
        labels = torch.full((len(seqs), target), -100, dtype=torch.long)
        for i, s in enumerate(seqs):
            ids[i, :s.numel()] = s
            labels[i, :s.numel()] = s
        return {"input_ids": ids, "labels": labels}
    return collate
