"""Data pipeline: packing, synthetic generators, raw-dataset loading."""

from __future__ import annotations


def load_raw_dataset(data_cfg):
    """HF datasets loader for both hub paths (reference main.py:45-50) and
    local HF-format corpora (kind: hf_local — a committed jsonl with a
    "text" column, used where the hub is unreachable)."""
    import datasets
    path = data_cfg.path
    if data_cfg.get("kind") == "hf_local" or path.endswith(".jsonl"):
        return datasets.load_dataset("json", data_files=path)
    return datasets.load_dataset(path)


def tokenizer_path(cfg):
    """Tokenizer source: data-group override (local corpora ship their own
    tokenizer) falling back to the model preset's hub name."""
    return cfg.data.get("tokenizer") or cfg.model.tokenizer
