"""Model factory: build a model from the `model` config group
(the capability of reference main.py:33-41)."""

from __future__ import annotations

import os

from acco_amd.models.config import GPTNeoConfig, LlamaConfig
from acco_amd.models.gptneo import GPTNeoForCausalLM
from acco_amd.models.llama import LlamaForCausalLM

__all__ = ["build_model", "GPTNeoConfig", "LlamaConfig",
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
