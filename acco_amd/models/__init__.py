"""Model factory: build a model from the `model` config group
(the capability of reference main.py:33-41)."""

from __future__ import annotations

import os

from acco_amd.models.config import GPTNeoConfig, LlamaConfig
from acco_amd.models.gptneo import GPTNeoForCausalLM
from acco_amd.models.llama import LlamaForCausalLM

__all__ = ["build_model", "load_pretrained", "GPTNeoConfig", "LlamaConfig",
           "GPTNeoForCausalLM", "LlamaForCausalLM"]


def build_model(model_cfg, vocab_size_override=None):
    """Instantiate a random-init model from a `model` group config (Cfg)."""
    family = model_cfg.get("family", "gptneo")
    if family == "llama":
        cfg = LlamaConfig.from_cfg(model_cfg)
        if vocab_size_override:
            cfg.vocab_size = vocab_size_override
        return LlamaForCausalLM(cfg)
    if family == "gptneo":
        cp = model_cfg.get("config_path")
        if cp and cp.endswith(".json"):
            here = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", "config", "model", cp)
            if os.path.exists(here):
                cfg = GPTNeoConfig.from_hf_json(here)
            else:
                cfg = GPTNeoConfig.from_cfg(model_cfg)
        else:
            cfg = GPTNeoConfig.from_cfg(model_cfg)
        if vocab_size_override:
            cfg.vocab_size = vocab_size_override
        return GPTNeoForCausalLM(cfg)
    raise ValueError(f"unknown model family {family!r}")


def load_pretrained(model, path):
    """Load pretrained weights for the finetune path (reference
    main.py:33-41 capability, network-free): `path` is either a
    state-dict file (our checkpoint format / torch.save of an HF state
    dict) or an HF model directory (pytorch_model.bin or
    model.safetensors). Key layout is HF-compatible for both families."""
    import torch

    if os.path.isdir(path):
        for name in ("model.safetensors", "pytorch_model.bin"):
            f = os.path.join(path, name)
            if os.path.exists(f):
                if name.endswith(".safetensors"):
                    from safetensors.torch import load_file
                    sd = load_file(f)
                else:
                    sd = torch.load(f, map_location="cpu", weights_only=True)
                break
        else:
            raise FileNotFoundError(
                f"no pytorch_model.bin / model.safetensors under {path}")
    else:
        sd = torch.load(path, map_location="cpu", weights_only=True)
    if "model_state_dict" in sd:      # full-resume checkpoint envelope
        sd = sd["model_state_dict"]
    missing, unexpected = model.load_state_dict(sd, strict=False)
    # tied lm_head may be absent from HF dicts; anything else is an error
    bad = [k for k in missing if "lm_head" not in k]
    if bad or unexpected:
        raise RuntimeError(
            f"pretrained load mismatch: missing={bad} unexpected={unexpected}")
    return model
