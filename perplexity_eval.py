"""Standalone perplexity evaluator (reference perplexity_eval.py:13-111
capability): per-token CE of a CausalLM over a dataset → exp(mean NLL).

Usage:
    python perplexity_eval.py model=llama-1b data=synthetic \
        checkpoint=checkpoints/run_model.pt train.batch_size=16

Works with our native models (HF-compatible state_dict checkpoints) on
synthetic or HF datasets (lambada/openwebtext when the hub is reachable).
"""

from __future__ import annotations

import math
import sys

import torch
from torch.utils.data import DataLoader

from acco_amd.config import load_config
from acco_amd.data.synthetic import SyntheticCausalLMDataset, collate_input_ids
from acco_amd.models import build_model


@torch.no_grad()
def compute_perplexity(model, dataloader, device, dtype=torch.float32,
                       max_batches: int | None = None) -> float:
    """Mean per-token NLL → perplexity (reference compute() :13-90)."""
    model.eval()
    total_nll = 0.0
    total_tokens = 0
    for i, batch in enumerate(dataloader):
        if max_batches is not None and i >= max_batches:
            break
        ids = batch["input_ids"].to(device)
        loss, _ = model(ids, labels=ids)
        n_tok = ids.shape[0] * (ids.shape[1] - 1)
        total_nll += float(loss) * n_tok
        total_tokens += n_tok
    return math.exp(total_nll / max(total_tokens, 1))


def main(argv=None):
    cfg = load_config(argv if argv is not None else sys.argv[1:])
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if (device.type == "cuda"
                               and cfg.train.use_mixed_precision) else torch.float32

    model = build_model(cfg.model)
    ckpt = cfg.get("checkpoint")
    if ckpt:
        sd = torch.load(ckpt, map_location="cpu", weights_only=True)
        model.load_state_dict(sd)
    model = model.to(device, dtype=dtype)

    if cfg.data.get("kind") == "synthetic":
        ds = SyntheticCausalLMDataset(cfg.data.n_eval_sequences or 64,
                                      cfg.train.max_length,
                                      model.cfg.vocab_size, seed=cfg.seed)
    else:
        from transformers import AutoTokenizer
        from acco_amd.data import load_raw_dataset, tokenizer_path
        from acco_amd.data.packing import make_tokenize_const_len_fn
        raw = load_raw_dataset(cfg.data)["train"]
        tok = AutoTokenizer.from_pretrained(tokenizer_path(cfg))
        tok.pad_token_id = tok.eos_token_id
        ds = raw.map(make_tokenize_const_len_fn(tok, "text",
                                                cfg.train.max_length),
                     batched=True, remove_columns=raw.column_names)
    dl = DataLoader(ds, batch_size=cfg.train.batch_size,
                    collate_fn=collate_input_ids, drop_last=True)
    ppl = compute_perplexity(model, dl, device)
    print(f"perplexity: {ppl:.4f}")
    return ppl


if __name__ == "__main__":
    main()
