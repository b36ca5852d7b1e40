"""Label-smoothed loss parity vs HF's LabelSmoother semantics."""

import pytest
import torch

from acco_amd.ops import torch_ref

transformers = pytest.importorskip("transformers")


def test_matches_hf_labelsmoother():
    from transformers.trainer_pt_utils import LabelSmoother
    torch.manual_seed(0)
    B, S, V = 2, 9, 17
    logits = torch.randn(B, S, V)
    labels = torch.randint(0, V, (B, S))
    labels[0, 3] = -100

    ours = torch_ref.label_smoothed_causal_lm_loss(logits, labels, 0.1)

    hf = LabelSmoother(epsilon=0.1)((logits,), labels, shift_labels=True)
    assert torch.allclose(ours, hf, atol=1e-6), (ours, hf)


def test_zero_epsilon_equals_ce():
    torch.manual_seed(1)
    logits = torch.randn(2, 8, 16)
    labels = torch.randint(0, 16, (2, 8))
    a = torch_ref.label_smoothed_causal_lm_loss(logits, labels, 0.0)
    b = torch_ref.causal_lm_loss(logits, labels)
    assert torch.allclose(a, b, atol=1e-6)
