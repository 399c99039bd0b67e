"""bf16-vs-f16 per-class timing diff on one box.

The two precisions move identical bytes through identical kernels; any gap
is instruction-level (conversions, MFMA variants) or library solution
quality — the round-2 native-cvt win was found this way.  Prints per-class
times for both dtypes at the bench shapes (per-device 64).
"""

import os
import sys

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from opendiloco_amd import ops  # noqa: E402
from opendiloco_amd.gemm_tuning import enable_tuned_gemms  # noqa: E402


def timeit(fn, reps=10):
    e0, e1 = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    e0.record()
    for _ in range(reps):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / reps  # ms


def main():
    enable_tuned_gemms()
    ext = ops._ext()
    torch.manual_seed(0)
    B, H, S, D, h, inter, V = 64, 16, 1024, 64, 1024, 2688, 32000
    R = B * S
    rows = {}
    for dt in (torch.bfloat16, torch.float16):
        n = dt == torch.bfloat16 and "bf16" or "f16"
        q = torch.randn(B, H, S, D, device="cuda", dtype=dt)
        k, v = torch.randn_like(q), torch.randn_like(q)
        o, lse = ext.attn_fwd(q, k, v, 0.125)
        do = torch.randn_like(o)
        rows.setdefault("attn_fwd", {})[n] = timeit(lambda: ext.attn_fwd(q, k, v, 0.125))
        rows.setdefault("attn_bwd", {})[n] = timeit(lambda: ext.attn_bwd(do, q, k, v, o, lse, 0.125))
        del q, k, v, o, lse, do
        x = torch.randn(R, h, device="cuda", dtype=dt)
        gu = torch.randn(R, 2 * inter, device="cuda", dtype=dt)
        dyi = torch.randn(R, inter, device="cuda", dtype=dt)
        w = (torch.randn(h, device="cuda", dtype=dt).abs() + 0.5)
        rows.setdefault("swiglu2_fwd", {})[n] = timeit(lambda: ext.swiglu2_fwd(gu, inter))
        rows.setdefault("swiglu2_bwd", {})[n] = timeit(lambda: ext.swiglu2_bwd(dyi, gu, inter))
        y, invrms = ext.rmsnorm_fwd(x, w, 1e-5)
        dy = torch.randn_like(x)
        rows.setdefault("rmsnorm_fwd", {})[n] = timeit(lambda: ext.rmsnorm_fwd(x, w, 1e-5))
        rows.setdefault("rmsnorm_bwd", {})[n] = timeit(lambda: ext.rmsnorm_bwd(dy, None, x, w, invrms))
        del gu, dyi, y, invrms, dy
        logits = torch.randn(B, S, V, device="cuda", dtype=dt)
        labels = torch.randint(3, V, (B, S), device="cuda")
        lr, lse2 = ext.ce_fwd(logits, labels)
        dl = torch.ones_like(lr)
        rows.setdefault("ce_fwd", {})[n] = timeit(lambda: ext.ce_fwd(logits, labels), 5)
        rows.setdefault("ce_bwd", {})[n] = timeit(lambda: ext.ce_bwd(logits, lse2, labels, dl, 1.0 / (B * (S - 1))), 5)
        del logits, labels, lr, lse2, dl
        for name, out_f, in_f in [("qkv", 3072, h), ("o", h, h), ("gateup", 5376, h),
                                  ("down", h, inter), ("lm_head", V, h)]:
            xx = torch.randn(R, in_f, device="cuda", dtype=dt)
            ww = torch.randn(out_f, in_f, device="cuda", dtype=dt)
            dyy = torch.randn(R, out_f, device="cuda", dtype=dt)
            wg = torch.zeros(out_f, in_f, device="cuda", dtype=torch.float32)
            rows.setdefault(f"gemm_{name}_fwd", {})[n] = timeit(lambda: F.linear(xx, ww), 5)
            rows.setdefault(f"gemm_{name}_dx", {})[n] = timeit(lambda: dyy @ ww, 5)
            rows.setdefault(f"gemm_{name}_dw", {})[n] = timeit(
                lambda: ops.dw_splitk_accum(dyy, xx, [(wg, 0)]), 5)
            del xx, ww, dyy, wg
    print(f"{'class':18s} {'bf16 ms':>9s} {'f16 ms':>9s} {'bf16/f16':>9s}")
    for k2, d in rows.items():
        print(f"{k2:18s} {d['bf16']:9.3f} {d['f16']:9.3f} {d['bf16']/d['f16']:9.2f}")


if __name__ == "__main__":
    main()
