"""Sample text from a trained checkpoint (sanity/demo tool).

Greedy or temperature sampling with a full-context re-forward per token
(the models are training-shaped — no KV cache; fine for short demos).
Beyond the reference's surface (it has no generation path); useful for
eyeballing that a checkpoint learned anything.

Usage (config overrides use the same syntax as main.py):
  python tools/generate.py --ckpt checkpoints/<id>_model.pt \\
      --tokenizer corpus/tokenizer --prompt "the" --max-new 64 \\
      --temperature 0.8 model=gptneo [model.hidden_size=... ...]
"""

from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


@torch.no_grad()
def generate(model, ids: torch.Tensor, max_new: int, temperature: float,
             max_len: int, seed: int = 0) -> torch.Tensor:
    gen = torch.Generator(device="cpu").manual_seed(seed)
    model.eval()
    for _ in range(max_new):
        out = model(ids[:, -max_len:])
        logits = out[0] if isinstance(out, tuple) else out
        step = logits[:, -1, :].float().cpu()
        if temperature <= 0:
            nxt = step.argmax(-1, keepdim=True)
        else:
            probs = torch.softmax(step / temperature, dim=-1)
            nxt = torch.multinomial(probs, 1, generator=gen)
        ids = torch.cat([ids, nxt.to(ids.device)], dim=1)
    return ids


def main(argv=None) -> str:
    ap = argparse.ArgumentParser()
    ap.add_argument("--ckpt", required=True,
                    help=".pt state dict or HF-style model dir")
    ap.add_argument("--tokenizer", required=True,
                    help="dir with tokenizer.json")
    ap.add_argument("--prompt", default="the")
    ap.add_argument("--max-new", type=int, default=64)
    ap.add_argument("--temperature", type=float, default=0.8)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("overrides", nargs="*",
                    help="config overrides, e.g. model=gptneo")
    args = ap.parse_args(argv)

    from acco_amd.config import load_config
    from acco_amd.models import build_model, load_pretrained
    from tokenizers import Tokenizer

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    tok = Tokenizer.from_file(os.path.join(args.tokenizer, "tokenizer.json"))

    cfg = load_config(args.overrides)
    model = build_model(cfg.model,
                        vocab_size_override=tok.get_vocab_size())
    load_pretrained(model, args.ckpt)
    model = model.to(device)

    ids = torch.tensor([tok.encode(args.prompt).ids], device=device)
    max_len = int(model.cfg.max_position_embeddings)
    out = generate(model, ids, args.max_new, args.temperature, max_len,
                   seed=args.seed)
    text = tok.decode(out[0].tolist())
    print(text)
    return text


if __name__ == "__main__":
    main()
