"""Isolated kernel microbenches for rocprofv3 counter runs (MI355X).

Usage (on the GPU box; counters in their OWN pass, guide §7):
  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof \
      -- python $GRAFT_REPO_ROOT/tools/kernel_microbench.py --op all
  rocprofv3 --pmc FETCH_SIZE -d ... -- python .../kernel_microbench.py --op adamw
  rocprofv3 --pmc WRITE_SIZE -d ... -- python .../kernel_microbench.py --op adamw

Each op runs `--reps` launches at the llama-150m hot-path shapes and prints
HIP-event timings + algorithmic-bytes bandwidth.
"""

import argparse
import os
import sys

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)

import torch  # noqa: E402

from opendiloco_amd.ops import _ext  # noqa: E402

N_150M = 214_983_680


def _timeit(fn, reps):
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


def bench_adamw(ext, reps):
    n = N_150M
    p, g = torch.randn(n, device="cuda"), torch.randn(n, device="cuda")
    m, v = torch.zeros(n, device="cuda"), torch.zeros(n, device="cuda")
    dt = _timeit(lambda: ext.fused_adamw(p, g, m, v, 1e-4, 0.9, 0.95, 1e-8, 0.1, 1), reps)
    algo = 28.0 * n
    print(f"adamw: {dt*1e3:.3f} ms/launch, algo 28 B/param -> {algo/dt/1e9:.0f} GB/s "
          f"({algo/dt/8e12*100:.1f}% of 8 TB/s spec)")


def bench_outer(ext, reps):
    n = N_150M
    to, tl = torch.randn(n, device="cuda"), torch.randn(n, device="cuda")
    buf, g = torch.zeros(n, device="cuda"), torch.randn(n, device="cuda")
    dt = _timeit(lambda: ext.outer_nesterov(to, tl, buf, g, 0.7, 0.9, False), reps)
    algo = 20.0 * n  # r: to,buf,g; w: to,tl,buf  (5 x 4B)... r3+w3 = 24? see DESIGN
    print(f"outer_nesterov: {dt*1e3:.3f} ms/launch, 24 B/param -> {24.0*n/dt/1e9:.0f} GB/s")
    dt = _timeit(lambda: ext.pseudo_grad(g, to, tl), reps)
    print(f"pseudo_grad: {dt*1e3:.3f} ms/launch, 12 B/param -> {12.0*n/dt/1e9:.0f} GB/s")


def bench_rmsnorm(ext, reps):
    from opendiloco_amd import ops

    rows, cols = 32 * 1024, 1024
    x = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(cols, device="cuda", dtype=torch.bfloat16)
    dt = _timeit(lambda: ext.rmsnorm_fwd(x, w, 1e-5), reps)
    algo = rows * cols * 2 * 2  # read x + write y (bf16)
    print(f"rmsnorm_fwd [{rows}x{cols}]: {dt*1e3:.3f} ms, {algo/dt/1e9:.0f} GB/s")
    y = torch.randn_like(x)
    ir = torch.rand(rows, device="cuda") + 0.5
    dt = _timeit(lambda: ext.rmsnorm_bwd(y, None, x, w, ir), reps)
    print(f"rmsnorm_bwd: {dt*1e3:.3f} ms, {3*rows*cols*2/dt/1e9:.0f} GB/s (3x bf16 tensors)")


def bench_swiglu(ext, reps):
    rows, cols = 32 * 1024, 2688
    g = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16)
    u = torch.randn_like(g)
    y = torch.empty_like(g)
    dt = _timeit(lambda: ext.swiglu_fwd(g, u), reps)
    print(f"swiglu_fwd [{rows}x{cols}]: {dt*1e3:.3f} ms, {3*rows*cols*2/dt/1e9:.0f} GB/s")


def bench_rope(ext, reps):
    B, H, S, D = 32, 16, 1024, 64
    x = torch.randn(B * H * S, D, device="cuda", dtype=torch.bfloat16)
    inv = 1.0 / (10000 ** (torch.arange(0, D, 2, device="cuda", dtype=torch.float32) / D))
    f = torch.outer(torch.arange(S, device="cuda", dtype=torch.float32), inv)
    cos, sin = f.cos(), f.sin()
    dt = _timeit(lambda: ext.rope(x, cos, sin, S, False), reps)
    print(f"rope [{B*H*S}x{D}]: {dt*1e3:.3f} ms, {2*x.numel()*2/dt/1e9:.0f} GB/s")


def bench_ce(ext, reps):
    B, S, V = 8, 1024, 32000
    T = B * (S - 1)
    logits = torch.randn(B, S, V, device="cuda", dtype=torch.bfloat16)
    labels = torch.randint(0, V, (B, S), device="cuda")
    dt = _timeit(lambda: ext.ce_fwd(logits, labels), reps)
    print(f"ce_fwd [{B}x{S}x{V}]: {dt*1e3:.3f} ms, {T*V*2/dt/1e9:.0f} GB/s (1 read)")
    lse = torch.randn(T, device="cuda")
    dl = torch.ones((), device="cuda")
    dt = _timeit(lambda: ext.ce_bwd(logits, lse, labels, dl, 1.0 / T), reps)
    print(f"ce_bwd: {dt*1e3:.3f} ms, {2*B*S*V*2/dt/1e9:.0f} GB/s (r+w)")


def bench_attn(ext, reps):
    B, Hq, S, D = 32, 16, 1024, 64
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    k, v = torch.randn_like(q), torch.randn_like(q)
    dt = _timeit(lambda: ext.attn_fwd(q, k, v, D**-0.5), reps)
    flops = 4.0 * B * Hq * S * S * D / 2  # causal
    print(f"attn_fwd [B{B} H{Hq} S{S} D{D}]: {dt*1e3:.3f} ms, {flops/dt/1e12:.0f} TFLOP/s")
    o, lse = ext.attn_fwd(q, k, v, D**-0.5)
    do = torch.randn_like(o)
    dt = _timeit(lambda: ext.attn_bwd(do, q, k, v, o, lse, D**-0.5), reps)
    print(f"attn_bwd: {dt*1e3:.3f} ms, {2.5*flops/dt/1e12:.0f} TFLOP/s (2.5x fwd flops)")


def bench_clip(ext, reps):
    g = torch.randn(N_150M, device="cuda")
    dt = _timeit(lambda: ext.clip_grad_(g, 1e9), reps)
    print(f"clip [{N_150M}]: {dt*1e3:.3f} ms, {12.0*N_150M/dt/1e9:.0f} GB/s (2r+1w)")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--op", default="all",
                    choices=["all", "adamw", "outer", "rmsnorm", "swiglu", "rope", "ce",
                             "attn", "clip"])
    ap.add_argument("--reps", type=int, default=20)
    args = ap.parse_args()
    ext = _ext()
    torch.manual_seed(0)
    table = dict(adamw=bench_adamw, outer=bench_outer, rmsnorm=bench_rmsnorm,
                 swiglu=bench_swiglu, rope=bench_rope, ce=bench_ce, attn=bench_attn,
                 clip=bench_clip)
    for name, fn in table.items():
        if args.op in ("all", name):
            fn(ext, args.reps)


if __name__ == "__main__":
    main()
