"""Generate golden parity vectors from the CPU oracle.

Runs the oracle (the restated train_diloco_torch.py loop over the real
``transformers.LlamaForCausalLM``) on the committed llama-2m fixture at a few
small configurations and writes per-step (Loss, lr) traces + final parameter
digests to tests/golden/*.json.  These are the vectors the GPU parity tests
and the oracle self-tests compare against (reference test scheme:
tests/test_training/test_train.py:76-83 — loss atol 1e-3, lr exact).

Usage: python -m oracle.gen_golden
"""

import json
import os

import torch

from oracle.diloco_oracle import OracleConfig, run_diloco_oracle

HERE = os.path.dirname(__file__)
GOLDEN = os.path.join(HERE, "..", "tests", "golden")
FIXTURE = os.path.join(HERE, "..", "tests", "models", "llama-2m")

CASES = {
    # single worker, H=1 (pure local AdamW; outer step fires every real step)
    "llama2m_w1_h1": OracleConfig(model_path=FIXTURE, n_workers=1, local_steps=1,
                                  batch_size=16, per_device_train_batch_size=8,
                                  seq_length=128, max_steps=6, seed=42,
                                  record_param_hash_every=3),
    # two workers, H=3, two outer rounds
    "llama2m_w2_h3": OracleConfig(model_path=FIXTURE, n_workers=2, local_steps=3,
                                  batch_size=16, per_device_train_batch_size=8,
                                  seq_length=128, max_steps=6, seed=42,
                                  record_param_hash_every=3),
    # the reference e2e test shape (test_train.py:24-39): lr 1e-2, batch 16/8, seq 1024
    "llama2m_w2_h5_seq1024": OracleConfig(model_path=FIXTURE, n_workers=2, local_steps=5,
                                          batch_size=16, per_device_train_batch_size=8,
                                          seq_length=1024, lr=1e-2, max_steps=10, seed=42,
                                          record_param_hash_every=5),
    # longer horizon: 30 steps with 10 outer rounds (bf16-drift parity case)
    "llama2m_w1_h3_long": OracleConfig(model_path=FIXTURE, n_workers=1, local_steps=3,
                                       batch_size=16, per_device_train_batch_size=8,
                                       seq_length=256, lr=1e-3, max_steps=30, seed=42,
                                       record_param_hash_every=10),
}


def main():
    torch.use_deterministic_algorithms(True)
    os.makedirs(GOLDEN, exist_ok=True)
    for name, cfg in CASES.items():
        result = run_diloco_oracle(cfg)
        payload = {"config": {k: getattr(cfg, k) for k in cfg.__dataclass_fields__}, **result}
        out = os.path.join(GOLDEN, f"{name}.json")
        with open(out, "w") as f:
            json.dump(payload, f, indent=1)
        print(name, "losses:", [r["losses"] for r in result["records"][:3]], "->", out)


if __name__ == "__main__":
    main()
