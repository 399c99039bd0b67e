"""Deterministic per-parameter init shared by the oracle golden generator
(CPU, this container) and the GPU parity tests (MI355X box).

The llama-1b fixture would be 4.4 GB — too large to commit or snapshot — so
full-depth 1b parity instead pins the WEIGHTS by
construction: every parameter is filled from its own torch.Generator seeded
by crc32 of its state-dict name (norm weights = 1.0, everything else
N(0, initializer_range), the reference init's distribution — HF
_init_weights via init_weights.py:10-29).  The product model and
``transformers.LlamaForCausalLM`` share state-dict names (verified by
tests/test_model_cpu.py), so applying this to either yields bit-identical
fp32 tensors without any weight file.

TEST INFRASTRUCTURE (see oracle/__init__.py).
"""

from __future__ import annotations

import zlib

import torch

INIT_STD = 0.02  # HF LlamaConfig.initializer_range default (config_1b.json omits it)


def apply_deterministic_init(model: torch.nn.Module, base_seed: int = 42) -> None:
    """Fill every parameter in state-dict-name order from a per-name seed."""
    with torch.no_grad():
        for name, p in sorted(model.named_parameters()):
            if "norm" in name.split(".")[-2]:
                p.fill_(1.0)
                continue
            gen = torch.Generator().manual_seed(base_seed + zlib.crc32(name.encode()))
            vals = torch.empty(p.shape, dtype=torch.float32)
            vals.normal_(0.0, INIT_STD, generator=gen)
            p.copy_(vals)


CONFIG_1B = {
    # reference open_diloco/configs/config_1b.json (vocab/max_pos = HF defaults)
    "architectures": ["LlamaForCausalLM"],
    "model_type": "llama",
    "hidden_size": 2048,
    "intermediate_size": 5632,
    "num_attention_heads": 32,
    "num_hidden_layers": 22,
    "num_key_value_heads": 4,
    "rms_norm_eps": 1e-05,
    "use_cache": False,
    "vocab_size": 32000,
    "max_position_embeddings": 2048,
    "tie_word_embeddings": False,
    "torch_dtype": "float32",
}
