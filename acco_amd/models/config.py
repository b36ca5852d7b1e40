"""Model configuration dataclasses (HF-config-compatible field names)."""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class LlamaConfig:
    hidden_size: int = 2048
    num_layers: int = 16
    num_heads: int = 32
    num_kv_heads: int = 8
    intermediate_size: int = 8192
    vocab_size: int = 50304
    max_position_embeddings: int = 4096
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @classmethod
    def from_cfg(cls, cfg) -> "LlamaConfig":
        keys = cls.__dataclass_fields__.keys()
        return cls(**{k: cfg[k] for k in keys if k in cfg})

    @classmethod
    def from_hf(cls, hf) -> "LlamaConfig":
        """From a transformers LlamaConfig."""
        return cls(
            hidden_size=hf.hidden_size,
            num_layers=hf.num_hidden_layers,
            num_heads=hf.num_attention_heads,
            num_kv_heads=getattr(hf, "num_key_value_heads", hf.num_attention_heads),
            intermediate_size=hf.intermediate_size,
            vocab_size=hf.vocab_size,
            max_position_embeddings=hf.max_position_embeddings,
            rms_norm_eps=hf.rms_norm_eps,
            rope_theta=getattr(hf, "rope_theta", 10000.0),
            tie_word_embeddings=getattr(hf, "tie_word_embeddings", False),
        )


@dataclass
class GPTNeoConfig:
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    vocab_size: int = 50257
    max_position_embeddings: int = 1024
    window_size: int = 256
    layer_norm_epsilon: float = 1e-5
    activation: str = "gelu_new"
    attention_pattern: List[str] = field(default_factory=lambda: ["global", "local"])
    intermediate_size: Optional[int] = None  # None → 4*hidden (HF default)
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @property
    def inner_size(self) -> int:
        return self.intermediate_size or 4 * self.hidden_size

    def layer_attention_type(self, i: int) -> str:
        return self.attention_pattern[i % len(self.attention_pattern)]

    @classmethod
    def from_cfg(cls, cfg) -> "GPTNeoConfig":
        keys = cls.__dataclass_fields__.keys()
        d = {k: cfg[k] for k in keys if k in cfg}
        return cls(**d)

    @classmethod
    def from_hf_json(cls, path: str) -> "GPTNeoConfig":
        """From an HF GPTNeoConfig json (reference config/model/gpt-neo-125M.json)."""
        with open(path) as f:
            j = json.load(f)
        return cls(
            hidden_size=j["hidden_size"],
            num_layers=j["num_layers"],
            num_heads=j["num_heads"],
            vocab_size=j["vocab_size"],
            max_position_embeddings=j["max_position_embeddings"],
            window_size=j.get("window_size", 256),
            layer_norm_epsilon=j.get("layer_norm_epsilon", 1e-5),
            activation=j.get("activation_function", "gelu_new"),
            attention_pattern=list(j.get("attention_layers", ["global", "local"]))[:2]
            or ["global", "local"],
            intermediate_size=j.get("intermediate_size"),
        )
