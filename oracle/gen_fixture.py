"""Re-save the reference's checked-in 2M-parameter Llama test fixture.

The reference pins its e2e parity tests on the checked-in fresh-init model
``tests/models/llama-2m-fresh`` (reference tests/test_training/test_train.py:24,
shape from open_diloco/configs/config_2m.json: hidden 64, 2 layers, 2 heads,
vocab 1024).  This script loads THOSE exact weights from
``/root/reference/tests/models/llama-2m-fresh`` via ``transformers`` and
re-saves them under ``tests/models/llama-2m`` so every golden vector in this
repo derives from the reference's own fixture tensors, not a regenerated
fresh init.  Run once in the container that has ``/root/reference``; the
output is committed (the GPU box never reads ``/root/reference``).

A bit-exactness check (max |saved - reference| == 0 per tensor) runs before
writing.

Usage: python -m oracle.gen_fixture
"""

import os

import torch

REF = "/root/reference/tests/models/llama-2m-fresh"
OUT = os.path.join(os.path.dirname(__file__), "..", "tests", "models", "llama-2m")


def main():
    from transformers import LlamaForCausalLM

    if not os.path.isdir(REF):
        raise SystemExit(
            f"{REF} not found - this script only runs in the build container "
            "that holds the read-only reference checkout."
        )

    model = LlamaForCausalLM.from_pretrained(REF)
    model.config.use_cache = False
    os.makedirs(OUT, exist_ok=True)
    model.save_pretrained(OUT, safe_serialization=True)

    # verify the round trip is bit-exact against the reference tensors
    reloaded = LlamaForCausalLM.from_pretrained(OUT)
    ref_sd = model.state_dict()
    new_sd = reloaded.state_dict()
    assert set(ref_sd) == set(new_sd)
    for k in ref_sd:
        assert torch.equal(ref_sd[k], new_sd[k]), f"tensor {k} changed in round trip"

    n = sum(p.numel() for p in model.parameters())
    print(f"re-saved {n} params (reference llama-2m-fresh weights) to {OUT}")


if __name__ == "__main__":
    main()
