"""Generate the checked-in 2M-parameter Llama test fixture.

Analogue of the reference's fresh-init util (open_diloco/init_weights.py:10-29)
and its checked-in tests/models/llama-2m-fresh fixture: a seeded fresh
``LlamaForCausalLM`` built from the 2m shape (hidden 64, 2 layers, 2 heads,
vocab 1024 — reference open_diloco/configs/config_2m.json), saved as
safetensors under tests/models/llama-2m.  Run once; the output is committed.

Usage: python -m oracle.gen_fixture
"""

import json
import os

import torch

OUT = os.path.join(os.path.dirname(__file__), "..", "tests", "models", "llama-2m")

CONFIG_2M = {
    # shape fields from the reference's config_2m.json
    "architectures": ["LlamaForCausalLM"],
    "model_type": "llama",
    "hidden_size": 64,
    "intermediate_size": 256,
    "num_attention_heads": 2,
    "num_key_value_heads": 2,
    "num_hidden_layers": 2,
    "rms_norm_eps": 1e-05,
    "use_cache": False,
    "vocab_size": 1024,
    "max_position_embeddings": 2048,
    "tie_word_embeddings": False,
    "torch_dtype": "float32",
}


def main():
    from transformers import LlamaConfig, LlamaForCausalLM

    os.makedirs(OUT, exist_ok=True)
    cfg_path = os.path.join(OUT, "config.json")
    with open(cfg_path, "w") as f:
        json.dump(CONFIG_2M, f, indent=2)

    torch.manual_seed(1234)
    lcfg = LlamaConfig.from_pretrained(cfg_path)
    lcfg.use_cache = False
    model = LlamaForCausalLM(lcfg).float()
    model.save_pretrained(OUT, safe_serialization=True)
    n = sum(p.numel() for p in model.parameters())
    print(f"saved {n} params to {OUT}")


if __name__ == "__main__":
    main()
