"""torch.profiler breakdown of one microbatch fwd+bwd (aten-op level, with
shapes) — identifies which aten glue ops remain around the HIP kernels.

Usage (GPU box): python tools/torch_profile.py [--rows 40]
"""

import argparse
import os
import sys

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=40)
    ap.add_argument("--micro", type=int, default=32)
    args = ap.parse_args()

    from opendiloco_amd.gemm_tuning import enable_tuned_gemms
    from opendiloco_amd.llama_config import LlamaModelConfig
    from opendiloco_amd.model import LlamaForCausalLM

    enable_tuned_gemms()
    cfg = LlamaModelConfig(vocab_size=32000, hidden_size=1024, intermediate_size=2688,
                           num_hidden_layers=12, num_attention_heads=16)
    model = LlamaForCausalLM(cfg).init_weights(seed=42).to("cuda")
    model.compute_dtype = torch.bfloat16
    ids = torch.randint(3, 32000, (args.micro, 1024), device="cuda")
    batch = dict(input_ids=ids, attention_mask=torch.ones_like(ids), labels=ids.clone())

    for _ in range(2):  # warmup
        model(**batch).loss.backward()
        model.zero_grad(set_to_none=False)
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        for _ in range(2):
            model(**batch).loss.backward()
            model.zero_grad(set_to_none=False)
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="cuda_time_total" if torch.version.hip is None else "device_time_total",
        row_limit=args.rows, max_src_column_width=60))


if __name__ == "__main__":
    main()
