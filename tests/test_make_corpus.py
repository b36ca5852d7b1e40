"""tools/make_corpus.py generates a valid corpus + BPE tokenizer offline
(the hub-free real-data path's input artifacts)."""

import json
import os
import subprocess
import sys


def test_make_corpus_tiny(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = tmp_path / "corpus"
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "tools", "make_corpus.py"),
         "--docs", "30", "--vocab-size", "300", "--out", str(out)],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr

    docs = [json.loads(l)
            for l in open(out / "openwebtext_local" / "data.jsonl")]
    assert len(docs) == 30 and all(d["text"].strip() for d in docs)

    # the generated tokenizer loads through the real tokenizers lib and
    # round-trips text
    from tokenizers import Tokenizer
    tok = Tokenizer.from_file(str(out / "tokenizer" / "tokenizer.json"))
    assert tok.get_vocab_size() <= 300
    ids = tok.encode(docs[0]["text"]).ids
    assert ids and tok.decode(ids) == docs[0]["text"]
