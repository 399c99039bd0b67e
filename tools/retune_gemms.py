"""Re-tune the training GEMMs (torch TunableOp) with an enlarged search
budget, and report per-shape timings.

Round-1 tuning left the dW class (nt_*: K=32768 token-reduction GEMMs, the
weight-gradient GEMMs of every projection) at 525-1011 TF while the fwd/dx
classes reach 1400-1733 TF — these shapes want split-K solutions that a
short tuning budget may never reach.  This script runs every GEMM the
llama-150m step issues (fwd y=xW^T, dx=dy W, dW=dy^T x for each projection
shape + lm_head) under PYTORCH_TUNABLEOP_TUNING=1 with a large per-shape
budget, writes the result csv, then re-times each shape with HIP events.

Usage (GPU box):
    PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop_retuned.csv \
        python tools/retune_gemms.py
The produced csv is committed as opendiloco_amd/tunableop_gfx950.csv when it
beats the old one.
"""

import os
import sys

os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_VERBOSE", "0")
# large budget: default is 30 ms / 100 iterations per solution set
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "2000")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "1000")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                                   "gpurun_out", "tunableop_retuned.csv"))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

R = 32768  # tokens per micro-batch (32 seqs x 1024)
# (out_features, in_features) of every projection in llama-150m
SHAPES = [
    ("qkv", 3072, 1024),
    ("o", 1024, 1024),
    ("gateup", 5376, 1024),
    ("down", 1024, 2688),
    ("lm_head", 32000, 1024),
]


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
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    total_t = {"fwd": 0.0, "dx": 0.0, "dw": 0.0}
    total_f = {"fwd": 0.0, "dx": 0.0, "dw": 0.0}
    for name, out_f, in_f in SHAPES:
        x = torch.randn(R, in_f, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(out_f, in_f, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(R, out_f, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * R * out_f * in_f
        # first call of each tunes (slow); then timed
        for kind, fn in (("fwd", lambda: F.linear(x, w)),
                         ("dx", lambda: dy @ w),
                         ("dw", lambda: dy.t() @ x)):
            fn()
            torch.cuda.synchronize()
            dt = _timeit(fn)
            n_layer = 1 if name == "lm_head" else 12
            total_t[kind] += dt * n_layer
            total_f[kind] += flops * n_layer
            print(f"{name:8s} {kind:3s} [{out_f}x{in_f} R={R}]: "
                  f"{dt*1e6:8.1f} us  {flops/dt/1e12:7.0f} TF", flush=True)
        del x, w, dy
    print("---- per-class totals (x12 layers + lm_head, one micro-batch) ----")
    for kind in ("fwd", "dx", "dw"):
        print(f"{kind}: {total_t[kind]*1e3:7.2f} ms  {total_f[kind]/total_t[kind]/1e12:7.0f} TF",
              flush=True)
    if hasattr(torch.cuda.tunable, "write_file"):
        torch.cuda.tunable.write_file()
    print("csv written to", os.environ["PYTORCH_TUNABLEOP_FILENAME"])


if __name__ == "__main__":
    main()
