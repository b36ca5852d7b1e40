"""Generate the committed local text corpus + tokenizer for the real-data
end-to-end path (no network in this environment, so the HF-datasets-format
corpus is synthesized once and committed; reference main.py:45-50 /
dl_dataset.py:8-34 load openwebtext from the hub instead).

The corpus is pseudo-English with learnable structure (Zipfian unigrams +
a deterministic bigram preference), so a short training run shows a real
falling loss curve and a perplexity well under the uniform ceiling.

    python tools/make_corpus.py [--docs 8000] [--out corpus]

Writes:
    corpus/openwebtext_local/data.jsonl   {"text": ...} per document
    corpus/tokenizer/                     ByteLevel-BPE (vocab 8192),
                                          AutoTokenizer.from_pretrained-able
"""

from __future__ import annotations

import argparse
import json
import os
import random


def make_vocab(rng: random.Random, n_words: int = 2400):
    onset = ["b", "br", "c", "ch", "d", "dr", "f", "fl", "g", "gr", "h", "j",
             "k", "l", "m", "n", "p", "pl", "pr", "qu", "r", "s", "sh", "sl",
             "st", "t", "th", "tr", "v", "w"]
    nucleus = ["a", "ai", "e", "ea", "ee", "i", "o", "oa", "oo", "ou", "u"]
    coda = ["", "b", "ck", "d", "g", "l", "ll", "m", "n", "nd", "ng", "nt",
            "p", "r", "rd", "s", "st", "t", "th"]
    words = set()
    while len(words) < n_words:
        syllables = rng.choices((1, 2, 3), weights=(5, 4, 1))[0]
        w = "".join(rng.choice(onset) + rng.choice(nucleus) +
                    (rng.choice(coda) if s == syllables - 1 else "")
                    for s in range(syllables))
        words.add(w)
    return sorted(words)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=8000)
    ap.add_argument("--out", default="corpus")
    ap.add_argument("--vocab-size", type=int, default=8192)
    args = ap.parse_args()

    rng = random.Random(20260914)
    words = make_vocab(rng)
    n = len(words)
    # Zipfian unigram weights
    weights = [1.0 / (i + 3) for i in range(n)]
    # deterministic bigram preference: after word i, a small set of
    # "successor" words is 8x more likely — gives the model something
    # beyond unigram frequency to learn
    succ = [[(i * 7 + k * 911) % n for k in range(6)] for i in range(n)]

    data_dir = os.path.join(args.out, "openwebtext_local")
    os.makedirs(data_dir, exist_ok=True)
    corpus_path = os.path.join(data_dir, "data.jsonl")
    with open(corpus_path, "w") as f:
        for _ in range(args.docs):
            n_sent = rng.randint(3, 18)
            sents = []
            prev = rng.choices(range(n), weights=weights)[0]
            for _ in range(n_sent):
                n_w = rng.randint(4, 22)
                ws = []
                for _ in range(n_w):
                    if rng.random() < 0.55:
                        prev = rng.choice(succ[prev])
                    else:
                        prev = rng.choices(range(n), weights=weights)[0]
                    ws.append(words[prev])
                s = " ".join(ws)
                s = s[0].upper() + s[1:] + rng.choices(
                    [".", ".", ".", "?", "!"], k=1)[0]
                sents.append(s)
            f.write(json.dumps({"text": " ".join(sents)}) + "\n")
    size_mb = os.path.getsize(corpus_path) / 1e6
    print(f"wrote {corpus_path} ({args.docs} docs, {size_mb:.1f} MB)")

    # ---- ByteLevel BPE tokenizer trained on the corpus (offline)
    from tokenizers import ByteLevelBPETokenizer

    texts = [json.loads(l)["text"] for l in open(corpus_path)]
    tok = ByteLevelBPETokenizer()
    tok.train_from_iterator(texts, vocab_size=args.vocab_size,
                            min_frequency=2,
                            special_tokens=["<|endoftext|>"])
    tok_dir = os.path.join(args.out, "tokenizer")
    os.makedirs(tok_dir, exist_ok=True)
    tok.save(os.path.join(tok_dir, "tokenizer.json"))
    with open(os.path.join(tok_dir, "tokenizer_config.json"), "w") as f:
        json.dump({"tokenizer_class": "PreTrainedTokenizerFast",
                   "eos_token": "<|endoftext|>",
                   "bos_token": "<|endoftext|>",
                   "unk_token": "<|endoftext|>",
                   "model_max_length": 1_000_000}, f, indent=1)
    with open(os.path.join(tok_dir, "special_tokens_map.json"), "w") as f:
        json.dump({"eos_token": "<|endoftext|>",
                   "bos_token": "<|endoftext|>",
                   "unk_token": "<|endoftext|>"}, f, indent=1)
    print(f"wrote tokenizer to {tok_dir}")

    from transformers import AutoTokenizer
    t = AutoTokenizer.from_pretrained(tok_dir)
    ids = t("Hello brainth stoom.")["input_ids"]
    assert t.eos_token_id is not None and len(ids) > 0
    print(f"tokenizer round-trip OK (eos id {t.eos_token_id}, "
          f"vocab {t.vocab_size})")


if __name__ == "__main__":
    main()
