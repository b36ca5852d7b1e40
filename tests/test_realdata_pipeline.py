"""Real-data pipeline end-to-end on CPU: committed jsonl corpus + committed
BPE tokenizer → dl_dataset offline packing → main.py training (HF-datasets
branch) → perplexity_eval. Mirrors the reference's openwebtext flow
(reference main.py:45-50, dl_dataset.py:8-34, perplexity_eval.py:95-111)
without hub access."""

import itertools
import json
import os

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CORPUS = os.path.join(REPO, "corpus", "openwebtext_local", "data.jsonl")
TOKENIZER = os.path.join(REPO, "corpus", "tokenizer")

TINY = [
    "model.hidden_size=64", "model.num_layers=2", "model.num_heads=2",
    "model.max_position_embeddings=128", "model.vocab_size=8192",
    "train.max_length=128", "train.batch_size=2",
    "train.n_grad_accumulation=1", "train.dataloader_num_workers=0",
]


@pytest.fixture(scope="module")
def small_corpus(tmp_path_factory):
    """First 120 docs of the committed corpus (full-corpus tokenization is
    a GPU-box job; the machinery is identical)."""
    d = tmp_path_factory.mktemp("corpus")
    path = d / "data.jsonl"
    with open(CORPUS) as src, open(path, "w") as dst:
        for line in itertools.islice(src, 120):
            dst.write(line)
    return str(path)


def test_corpus_and_tokenizer_committed():
    assert os.path.exists(CORPUS)
    assert os.path.exists(os.path.join(TOKENIZER, "tokenizer.json"))
    doc = json.loads(open(CORPUS).readline())
    assert isinstance(doc["text"], str) and len(doc["text"]) > 20
    from transformers import AutoTokenizer
    tok = AutoTokenizer.from_pretrained(TOKENIZER)
    ids = tok(doc["text"])["input_ids"]
    assert len(ids) > 10
    assert tok.eos_token_id is not None


def test_dl_dataset_offline_pack(small_corpus, tmp_path, monkeypatch):
    monkeypatch.chdir(REPO)
    import dl_dataset
    out = str(tmp_path / "packed")
    dl_dataset.main(["data=localtext", "model=gptneo",
                     f"data.path={small_corpus}",
                     "train.max_length=128",
                     "out_shards=2",
                     f"out_dir={out}"])
    import glob as _glob
    import datasets
    packed = datasets.load_from_disk(out)
    assert "train" in packed and "test" in packed
    row = packed["train"][0]["input_ids"]
    assert len(row) == 128
    assert all(0 <= t < 8192 for t in row)
    assert len(_glob.glob(out + "/train/data-*.arrow")) == 2


def test_train_and_perplexity_on_local_corpus(small_corpus, tmp_path,
                                              monkeypatch):
    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    import main as train_main
    import perplexity_eval
    corpus_args = [f"data.path={small_corpus}",
                   f"data.tokenizer={TOKENIZER}"]
    train_main.main(["train=dpu", "data=localtext", "model=gptneo",
                     "run_dir=.",
                     "train.nb_steps_tot=4", "train.n_warmup_steps=0",
                     "train.eval=false", "train.save=true",
                     "train.use_mixed_precision=false",
                     "run_name=realdata_cpu"] + TINY + corpus_args)
    ckpts = [f for f in os.listdir(tmp_path / "checkpoints")
             if f.endswith("_model.pt")]
    assert ckpts, "final checkpoint written"
    ppl = perplexity_eval.main(
        ["data=localtext", "model=gptneo",
         f"checkpoint=checkpoints/{ckpts[0]}",
         "train.use_mixed_precision=false"] + TINY + corpus_args)
    # 4 tiny stale-gradient steps don't reach the uniform ceiling yet —
    # finite/sane says the whole path is wired correctly (the committed GPU
    # run in profiles/ shows the real falling loss curve + perplexity)
    import math
    assert math.isfinite(ppl) and 0 < ppl < 1e6
