"""Generate the FULL-DEPTH llama-1b parity golden.

Runs the CPU oracle (restated train_diloco_torch.py loop over the real
``transformers.LlamaForCausalLM``) on the reference's 22-layer 1b config
(configs/config_1b.json: hidden 2048, 32 q heads / 4 KV heads — the GQA
case) with deterministically-seeded weights (oracle/det_init.py, no weight
file needed), 3 real steps at H=2 (one outer crossing), batch 8 x seq 256.
The committed trace is what tests/test_gpu_model.py::
test_1b_full_depth_matches_golden compares the GPU bf16 path against.

Takes a few minutes of CPU (1.2B-param fp32 fwd+bwd x 3 + AdamW + outer).

Usage: python -m oracle.gen_golden_1b
"""

import json
import os
import tempfile

import torch

from oracle.det_init import CONFIG_1B, apply_deterministic_init
from oracle.diloco_oracle import OracleConfig, run_diloco_oracle

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = os.path.join(HERE, "..", "tests", "golden")

CFG = OracleConfig(model_path="", n_workers=1, local_steps=2, batch_size=8,
                   per_device_train_batch_size=8, seq_length=256,
                   vocab_size=32000, max_steps=3, seed=42)


def model_factory(cfg):
    from transformers import LlamaConfig, LlamaForCausalLM

    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "config.json")
        with open(p, "w") as f:
            json.dump(CONFIG_1B, f)
        lcfg = LlamaConfig.from_pretrained(p)
    lcfg.use_cache = False
    model = LlamaForCausalLM(lcfg).float()
    apply_deterministic_init(model)
    return model


def main():
    torch.use_deterministic_algorithms(True)
    result = run_diloco_oracle(CFG, model_factory=model_factory)
    payload = {"config": {k: getattr(CFG, k) for k in CFG.__dataclass_fields__},
               "model": "llama-1b (config_1b.json shape, det_init seed 42)",
               **result}
    out = os.path.join(GOLDEN, "llama1b_w1_h2_full_depth.json")
    with open(out, "w") as f:
        json.dump(payload, f, indent=1)
    print("losses:", [r["losses"] for r in result["records"]], "->", out)


if __name__ == "__main__":
    main()
