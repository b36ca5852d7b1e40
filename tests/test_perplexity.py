"""perplexity_eval capability (reference perplexity_eval.py): the compute
helper gives exp(mean NLL) on a tiny model/dataset."""

import math

import torch
from torch.utils.data import DataLoader

from acco_amd.data.synthetic import SyntheticCausalLMDataset, collate_input_ids
from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM
from perplexity_eval import compute_perplexity


def test_compute_perplexity_near_uniform():
    torch.manual_seed(0)
    V = 64
    cfg = GPTNeoConfig(hidden_size=32, num_layers=1, num_heads=2,
                       vocab_size=V, max_position_embeddings=32)
    model = GPTNeoForCausalLM(cfg)
    ds = SyntheticCausalLMDataset(8, 16, V, seed=1)
    dl = DataLoader(ds, batch_size=4, collate_fn=collate_input_ids)
    ppl = compute_perplexity(model, dl, torch.device("cpu"))
    # random-init model on random tokens ≈ uniform ≈ V
    assert 0.3 * V < ppl < 3 * V, ppl
