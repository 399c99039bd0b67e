"""Sweep the split-K chunk count per dW shape (R = tokens per micro-batch)
and print achieved TF, to pick the per-shape table hardcoded in ops.py.

Usage (GPU box): python tools/dw_sweep.py [--r 65536]
"""

import argparse
import os
import sys

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)

import torch  # noqa: E402

SHAPES_150M = [("qkv", 3072, 1024), ("o", 1024, 1024), ("gateup", 5376, 1024),
               ("down", 1024, 2688), ("lm_head", 32000, 1024)]
SHAPES_1B = [("qkv", 2560, 2048), ("o", 2048, 2048), ("gateup", 11264, 2048),
             ("down", 2048, 5632), ("lm_head", 32000, 2048)]


def _timeit(fn, reps=10):
    e0, e1 = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    e0.record()
    for _ in range(reps):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / 1000.0 / reps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--r", type=int, default=65536)
    ap.add_argument("--model", default="150m", choices=["150m", "1b"])
    args = ap.parse_args()
    SHAPES = SHAPES_150M if args.model == "150m" else SHAPES_1B
    from opendiloco_amd import ops

    ext = ops._ext()
    torch.manual_seed(0)
    R = args.r
    for name, N, K in SHAPES:
        dy = torch.randn(R, N, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(R, K, device="cuda", dtype=torch.bfloat16)
        wg = torch.zeros(N, K, device="cuda", dtype=torch.float32)
        flops = 2.0 * R * N * K
        line = f"{name:8s} R={R}:"
        best = (0, 0.0)
        for nc in [1, 2, 4, 8, 16, 32]:
            if R % nc or R // nc < 2048:
                continue
            partials = torch.empty(nc, N, K, dtype=torch.float32, device="cuda")

            def run():
                ext.dw_gemm_batched(dy, x, partials)
                ext.accum_chunks_(wg.reshape(-1), partials, 0)

            dt = _timeit(run)
            tf = flops / dt / 1e12
            line += f"  nc{nc}:{tf:5.0f}TF"
            if tf > best[1]:
                best = (nc, tf)
            del partials
        print(line + f"   -> best nc={best[0]} ({best[1]:.0f} TF)", flush=True)
        del dy, x, wg


if __name__ == "__main__":
    main()
