import torch

from acco_amd.data.packing import pack_const_len
from acco_amd.data.synthetic import SyntheticCausalLMDataset, collate_input_ids


def test_pack_const_len_semantics():
    """Matches reference tokenize_data_const_len (trainer_base.py:84-97):
    concat + per-doc EOS + chop, drop remainder."""
    docs = [[1, 2, 3], [4, 5], [6, 7, 8, 9]]
    out = pack_const_len(docs, eos_token_id=0, max_length=4)
    concat = [1, 2, 3, 0, 4, 5, 0, 6, 7, 8, 9, 0]
    assert out.shape == (3, 4)
    assert out.flatten().tolist() == concat[:12]


def test_pack_drops_remainder():
    out = pack_const_len([[1, 2, 3, 4, 5]], eos_token_id=9, max_length=4)
    assert out.shape == (1, 4)
    assert out.flatten().tolist() == [1, 2, 3, 4]


def test_pack_empty():
    out = pack_const_len([[1]], eos_token_id=9, max_length=4)
    assert out.shape == (0, 4)


def test_synthetic_dataset_deterministic():
    ds = SyntheticCausalLMDataset(10, 16, 100, seed=3)
    a = ds[4]["input_ids"]
    b = ds[4]["input_ids"]
    assert torch.equal(a, b)
    assert ds[3]["input_ids"].shape == (16,)
    assert not torch.equal(ds[3]["input_ids"], ds[4]["input_ids"])
    batch = collate_input_ids([ds[0], ds[1]])
    assert batch["input_ids"].shape == (2, 16)
    assert batch["input_ids"].dtype == torch.long


def test_padded_collator():
    from acco_amd.data.synthetic import make_padded_collator
    c = make_padded_collator(pad_token_id=9, pad_to_multiple=8)
    out = c([{"input_ids": [1, 2, 3]}, {"input_ids": [4, 5, 6, 7, 8]}])
    assert out["input_ids"].shape == (2, 8)
    assert out["input_ids"][0].tolist() == [1, 2, 3, 9, 9, 9, 9, 9]
    assert out["labels"][0].tolist() == [1, 2, 3, -100, -100, -100, -100, -100]
    assert out["labels"][1].tolist() == [4, 5, 6, 7, 8, -100, -100, -100]
