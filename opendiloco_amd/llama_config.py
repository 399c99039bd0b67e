"""Llama model configuration.

Mirrors the fields the reference consumes from its config JSONs
(reference: open_diloco/configs/config_150m.json, config_1b.json and the
checked-in tests/models/llama-2m-fresh/config.json, loaded through
``transformers.LlamaConfig`` at open_diloco/train_fsdp.py:171-174).
Defaults equal transformers' LlamaConfig defaults so that a partial config
JSON (the reference configs specify only a few fields) resolves to the same
model shapes.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, asdict


@dataclass
class LlamaModelConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int | None = None  # None -> == num_attention_heads (MHA)
    max_position_embeddings: int = 2048
    rms_norm_eps: float = 1e-6
    rope_theta: float = 10000.0
    initializer_range: float = 0.02
    tie_word_embeddings: bool = False
    attention_bias: bool = False
    mlp_bias: bool = False

    def __post_init__(self):
        if self.num_key_value_heads is None:
            self.num_key_value_heads = self.num_attention_heads
        assert self.hidden_size % self.num_attention_heads == 0
        assert self.num_attention_heads % self.num_key_value_heads == 0

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_json(cls, path: str) -> "LlamaModelConfig":
        """Load from a HF-style config.json (or a directory containing one)."""
        if os.path.isdir(path):
            path = os.path.join(path, "config.json")
        with open(path) as f:
            raw = json.load(f)
        keys = {f.name for f in cls.__dataclass_fields__.values()}  # type: ignore[attr-defined]
        kwargs = {k: v for k, v in raw.items() if k in keys}
        # transformers >=4.41 nests rope_theta under rope_parameters in some dumps
        if "rope_theta" not in kwargs and isinstance(raw.get("rope_parameters"), dict):
            if "rope_theta" in raw["rope_parameters"]:
                kwargs["rope_theta"] = raw["rope_parameters"]["rope_theta"]
        return cls(**kwargs)

    def to_dict(self) -> dict:
        return asdict(self)

    def num_params(self) -> int:
        """Total parameter count (untied embeddings counted twice like the ref)."""
        h, v, L = self.hidden_size, self.vocab_size, self.num_hidden_layers
        kvh = self.num_key_value_heads * self.head_dim
        per_layer = h * h * 2 + h * kvh * 2 + 3 * h * self.intermediate_size + 2 * h
        n = v * h + L * per_layer + h  # embed + layers + final norm
        n += v * h if not self.tie_word_embeddings else 0
        return n
