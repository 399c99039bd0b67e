"""bench.py — DiLoCo inner-step throughput benchmark on MI355X.

Contract: `python bench.py --gpus N --steps K --warmup W` runs the flagship
training step (llama-150m DiLoCo worker, bf16, per-worker batch 512 x seq
1024 = 524,288 tokens per inner step — BASELINE.json configs[1]/[3]) on N
GPUs of one node, one rank per GPU over RCCL.  W untimed warmup steps, then
EXACTLY K timed steps bracketed by barrier + torch.cuda.synchronize on both
sides; elapsed = MAX over ranks; rank 0 prints ONE JSON line.

`value` = whole-job tokens/s over all N workers (weak scaling: each worker
owns its full batch; the only cross-worker exchange is the outer
pseudo-gradient all-reduce every H steps, reported as outer_sync_fraction).
Inputs are pre-generated on device before the timed region.
"""

from __future__ import annotations

import argparse
import json
import os
import time
from functools import partial

import torch

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))

MODELS = {
    # shapes per the reference's configs/ (SURVEY.md §8): llama-150m / llama-1b
    "llama-150m": dict(vocab_size=32000, hidden_size=1024, intermediate_size=2688,
                       num_hidden_layers=12, num_attention_heads=16, num_key_value_heads=16),
    "llama-1b": dict(vocab_size=32000, hidden_size=2048, intermediate_size=5632,
                     num_hidden_layers=22, num_attention_heads=32, num_key_value_heads=4),
    "llama-2m": dict(vocab_size=1024, hidden_size=64, intermediate_size=256,
                     num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=2),
}



def _dist_init():
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        import torch.distributed as dist

        dist.init_process_group("nccl" if torch.cuda.is_available() else "gloo")
        return dist
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--model", default="llama-150m", choices=list(MODELS))
    ap.add_argument("--h", type=int, default=50, help="inner steps per outer round")
    ap.add_argument("--per-device", type=int, default=64)
    ap.add_argument("--batch", type=int, default=512, help="per-worker batch (seqs/step)")
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--precision", default="bf16", choices=["bf16", "fp16"])
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--no-roofline", action="store_true",
                    help="skip the in-process roofline/MFMA-fraction measurements "
                         "(for clean rocprofv3 kernel traces of the step alone)")
    ap.add_argument("--cpu-baseline-only", action="store_true")
    ap.add_argument("--advance-steps", type=int, default=0,
                    help="pre-advance the DiLoCo progress tracker by N inner steps "
                         "(no compute) so a short timed window can cross an outer "
                         "boundary at large H; e.g. --h 500 --advance-steps 497 "
                         "--warmup 2 --steps 6 times steps 500-505 incl. the outer "
                         "round at 500 (the H=500 sustained leg)")
    args = ap.parse_args()

    dist = _dist_init()
    rank = dist.get_rank() if dist else 0
    world = dist.get_world_size() if dist else 1
    assert world == args.gpus or dist is None, (world, args.gpus)
    n_workers = max(world, 1)

    from opendiloco_amd.llama_config import LlamaModelConfig

    mcfg = LlamaModelConfig(**MODELS[args.model])

    if args.cpu_baseline_only:
        print(json.dumps(_cpu_baseline(args, mcfg)))
        return

    assert torch.cuda.is_available(), "bench needs a GPU (MI355X)"
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
    torch.cuda.set_device(device)

    from opendiloco_amd.gemm_tuning import enable_tuned_gemms

    enable_tuned_gemms()

    from opendiloco_amd.diloco import DiLoCoOptimizer
    from opendiloco_amd.model import LlamaForCausalLM
    from opendiloco_amd.optim import clip_grad_norm_flat_

    model = LlamaForCausalLM(mcfg).init_weights(seed=42).to(device)
    model.compute_dtype = torch.bfloat16 if args.precision == "bf16" else torch.float16
    model.train()

    opt = DiLoCoOptimizer(
        batch_size=args.batch, num_inner_steps=args.h,
        outer_optimizer=partial(torch.optim.SGD, lr=0.7, momentum=0.9, nesterov=True),
        inner_optimizer=partial(torch.optim.AdamW, lr=4e-4, weight_decay=0.1, betas=(0.9, 0.95)),
        params=model.parameters())

    assert args.batch % args.per_device == 0
    grad_acc = args.batch // args.per_device

    # synthetic C4-shaped token batches, resident in HBM before timing
    gen = torch.Generator(device="cpu").manual_seed(42 + 1337 * rank)
    micros = []
    for _ in range(grad_acc):
        ids = torch.randint(3, mcfg.vocab_size, (args.per_device, args.seq),
                            generator=gen, dtype=torch.int64).to(device)
        micros.append(dict(input_ids=ids, attention_mask=torch.ones_like(ids),
                           labels=ids.clone()))

    # run the main compute on a HIGH-priority stream: the side-stream dW
    # GEMMs (ops.py DK_DW_ASYNC) otherwise starve the main stream's small
    # kernels at workgroup-dispatch arbitration (measured: rmsnorm's reduce
    # stretched 12 -> 184 us under overlap).  DK_MAIN_PRIO=0 disables.
    main_stream = (torch.cuda.Stream(priority=-1)
                   if os.environ.get("DK_MAIN_PRIO", "1") != "0" else None)

    def one_step():
        import contextlib

        with torch.cuda.stream(main_stream) if main_stream else contextlib.nullcontext():
            for mb in micros:
                loss = model(**mb).loss / grad_acc
                loss.backward()
            clip_grad_norm_flat_(opt.flat.flat_grad, 1.0)
            opt.step()
            opt.zero_grad()
        if main_stream:
            torch.cuda.current_stream().wait_stream(main_stream)

    if args.advance_steps:
        # tracker-only advance: the inner steps are identical regardless of
        # the sample counter; this lets the timed region cross the H-step
        # outer boundary without paying H-3 untimed real steps first
        opt.tracker.report_local_progress(
            opt.local_epoch, samples_accumulated=args.advance_steps * args.batch)

    for _ in range(args.warmup):
        one_step()

    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if dist:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    tokens_per_step = args.batch * args.seq
    value = n_workers * args.steps * tokens_per_step / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    # ---- outer round cost (pseudo-grad + all-reduce + Nesterov + copy-back) ----
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    opt._update_global_epoch()
    torch.cuda.synchronize()
    outer_s = time.perf_counter() - t0
    outer_fraction = outer_s / (args.h * (ms_per_step / 1000.0) + outer_s)

    # ---- roofline of the dominant HBM-bound hand-written kernel (fused AdamW):
    # algorithmic traffic = 28 B/param (read p,g,m,v; write p,m,v), HIP events
    # on the launch stream ----
    roofline = None
    if rank == 0 and not args.no_roofline:
        from opendiloco_amd.ops import _ext

        ext = _ext()
        f = opt.flat
        ev0, ev1 = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
        for _ in range(3):
            ext.fused_adamw(f.flat_param, f.flat_grad, opt.inner_optimizer.flat_m,
                            opt.inner_optimizer.flat_v, 0.0, 0.9, 0.95, 1e-8, 0.0, 1)
        torch.cuda.synchronize()
        reps = 20
        ev0.record()
        for _ in range(reps):
            ext.fused_adamw(f.flat_param, f.flat_grad, opt.inner_optimizer.flat_m,
                            opt.inner_optimizer.flat_v, 0.0, 0.9, 0.95, 1e-8, 0.0, 1)
        ev1.record()
        torch.cuda.synchronize()
        dur_s = ev0.elapsed_time(ev1) / 1000.0 / reps
        algo_bytes = 28.0 * f.n
        peak = 8.0e12  # HBM3E spec peak (MI355X_MICROARCH.md; ~6.3e12 achievable)
        # measured per-launch HBM traffic from the committed PMC run
        # (profiles/round1_adamw_traffic.json: FETCH_SIZE x2-corrected + WRITE_SIZE,
        # measured at the llama-150m size; scaled by n for other sizes)
        traffic = None
        tf_path = os.path.join(REPO_ROOT, "profiles", "round1_adamw_traffic.json")
        if os.path.exists(tf_path):
            with open(tf_path) as tf:
                tj = json.load(tf)
            traffic = tj["bytes_per_param"] * f.n
        roofline = {
            "bound": "hbm",
            "achieved": algo_bytes / dur_s / 1e9,
            "peak": peak / 1e9,
            "unit": "GB/s",
            "frac": (algo_bytes / dur_s) / peak,
            "traffic": traffic,
            "kernel": "dk_fused_adamw",
            "launch_ms": dur_s * 1000.0,
            # MFMA-class fractions (attention + GEMM classes, HIP-event timed
            # at the bench shape) beside the AdamW HBM story
            "mfma": _mfma_fractions(args, mcfg),
        }

    cpu_baseline = None
    if rank == 0 and n_workers == 1 and not args.no_cpu_baseline:
        cpu_baseline = _cpu_baseline(args, mcfg)

    if rank == 0:
        line = {
            "metric": "tokens_per_second",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": n_workers,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published reference numbers
            "dtype": args.precision,
            "data": "synthetic",
            "config": {
                "workload": f"{args.model} DiLoCo, {n_workers} worker(s), H={args.h}, "
                            f"{args.precision}, total-batch {args.batch}/worker, seq {args.seq} "
                            f"(BASELINE.json configs[{1 if n_workers == 1 else 3}])",
                "model": args.model,
                "global_batch": args.batch * n_workers,
                "seq_len": args.seq,
                "parallelism": f"diloco-dp{n_workers}",
                "grad_acc": grad_acc,
                "per_device_batch": args.per_device,
                "params": mcfg.num_params(),
            },
            "tokens_per_second_per_worker": value / n_workers,
            "outer_sync_seconds": outer_s,
            "outer_sync_fraction": outer_fraction,
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(line))
    if dist:
        dist.barrier()
        dist.destroy_process_group()


def _mfma_fractions(args, mcfg) -> dict:
    """Attention fwd/bwd and GEMM-class achieved TFLOP/s vs the measured
    dense bf16 MFMA ceiling (2495 TF, MI355X_MICROARCH.md 32x32x16 row),
    HIP-event timed at the bench shape.  Read with DESIGN.md §3: the
    attention kernels run on 16x16x32 tiles, whose per-SIMD issue ceiling is
    ~half the 32x32 rate."""
    import torch.nn.functional as F

    from opendiloco_amd.ops import _ext

    ext = _ext()
    dtype = torch.bfloat16 if args.precision == "bf16" else torch.float16
    B, Hq, S = args.per_device, mcfg.num_attention_heads, args.seq
    D = mcfg.hidden_size // mcfg.num_attention_heads
    PEAK_TF = 2495.0

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
        return e0.elapsed_time(e1) / 1000.0 / reps

    out = {"peak_tflops": PEAK_TF, "peak_note": "measured dense bf16 32x32x16 ceiling"}
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=dtype)
    k, v = torch.randn_like(q), torch.randn_like(q)
    scale = D ** -0.5
    attn_flops = 4.0 * B * Hq * S * S * D / 2  # causal
    dt = timeit(lambda: ext.attn_fwd(q, k, v, scale))
    out["attn_fwd"] = {"tflops": attn_flops / dt / 1e12, "frac": attn_flops / dt / 1e12 / PEAK_TF}
    o, lse = ext.attn_fwd(q, k, v, scale)
    do = torch.randn_like(o)
    dt = timeit(lambda: ext.attn_bwd(do, q, k, v, o, lse, scale))
    out["attn_bwd"] = {"tflops": 2.5 * attn_flops / dt / 1e12,
                       "frac": 2.5 * attn_flops / dt / 1e12 / PEAK_TF}
    del q, k, v, o, lse, do
    # GEMM classes (library kernels) over the model's projection shapes
    R = args.per_device * args.seq
    h = mcfg.hidden_size
    shapes = [(3 * h, h, mcfg.num_hidden_layers), (h, h, mcfg.num_hidden_layers),
              (2 * mcfg.intermediate_size, h, mcfg.num_hidden_layers),
              (h, mcfg.intermediate_size, mcfg.num_hidden_layers),
              (mcfg.vocab_size, h, 1)]
    from opendiloco_amd.ops import dw_splitk_accum

    tot_t = {"fwd": 0.0, "dx": 0.0, "dw": 0.0}
    tot_f = 0.0
    for out_f, in_f, n in shapes:
        x = torch.randn(R, in_f, device="cuda", dtype=dtype)
        w = torch.randn(out_f, in_f, device="cuda", dtype=dtype)
        dy = torch.randn(R, out_f, device="cuda", dtype=dtype)
        wg = torch.zeros(out_f, in_f, device="cuda", dtype=torch.float32)
        flops = 2.0 * R * out_f * in_f
        tot_f += flops * n
        tot_t["fwd"] += timeit(lambda: F.linear(x, w)) * n
        tot_t["dx"] += timeit(lambda: dy @ w) * n
        # the dW the model actually runs: split-K batched + fp32 chunk reduce
        # (dw_stream_sync so the side-stream work is inside the timed region)
        from opendiloco_amd.ops import dw_stream_sync

        if not dw_splitk_accum(dy, x, [(wg, 0)]):
            tot_t["dw"] += timeit(lambda: dy.t() @ x) * n
        else:
            def _dw():
                dw_splitk_accum(dy, x, [(wg, 0)])
                dw_stream_sync()

            tot_t["dw"] += timeit(_dw) * n
        del x, w, dy, wg
    for kind in tot_t:
        tf = tot_f / tot_t[kind] / 1e12
        out[f"gemm_{kind}"] = {"tflops": tf, "frac": tf / PEAK_TF}
    return out


def _cpu_baseline(args, mcfg) -> dict:
    """Reference CPU path timed on host cores: the FULL BASELINE.json
    configs[0] loop — llama-150m(-shaped), 1 worker, H=1, per-device-batch 8,
    ONE real step = 1 micro fwd+bwd (8x1024 tokens) + clip + AdamW +
    scheduler + the outer block (pseudo-grad, 1-worker mean, Nesterov SGD,
    re-offload), restated from train_diloco_torch.py:272-353 with the
    oracle's own pieces (kind "port")."""
    import torch

    from oracle.diloco_oracle import (OracleConfig, _fake_batch,
                                      get_cosine_schedule_with_warmup,
                                      make_reference_model)

    import tempfile

    ocfg = OracleConfig(model_path="", fresh_init_seed=42, seq_length=args.seq,
                        vocab_size=mcfg.vocab_size)
    with tempfile.TemporaryDirectory() as d:
        with open(os.path.join(d, "config.json"), "w") as f:
            json.dump({"architectures": ["LlamaForCausalLM"], "model_type": "llama",
                       "use_cache": False, "torch_dtype": "float32",
                       "max_position_embeddings": 2048, "rms_norm_eps": 1e-5,
                       "hidden_size": mcfg.hidden_size,
                       "intermediate_size": mcfg.intermediate_size,
                       "num_attention_heads": mcfg.num_attention_heads,
                       "num_key_value_heads": mcfg.num_key_value_heads,
                       "num_hidden_layers": mcfg.num_hidden_layers,
                       "vocab_size": mcfg.vocab_size}, f)
        ocfg.model_path = d
        model = make_reference_model(ocfg).train()
    bs = 8  # configs[0]: per-device-batch 8, grad_acc 1, H = 1
    inner = torch.optim.AdamW(model.parameters(), lr=4e-4, weight_decay=0.1, betas=(0.9, 0.95))
    outer = torch.optim.SGD(model.parameters(), lr=0.7, momentum=0.9, nesterov=True)
    sched = get_cosine_schedule_with_warmup(inner, 1000, 88_000)
    offloaded = [p.data.detach().clone() for g in outer.param_groups for p in g["params"]]
    gen = torch.Generator().manual_seed(42)
    batch = _fake_batch(gen, bs, args.seq, mcfg.vocab_size)
    # one untimed warm micro-step (allocator/MKL warmup)
    model(**batch).loss.backward()
    inner.zero_grad()
    batch = _fake_batch(gen, bs, args.seq, mcfg.vocab_size)
    t0 = time.perf_counter()
    loss = model(**batch).loss  # train_diloco_torch.py:313
    loss.backward()  # :318
    torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)  # :323
    inner.step()  # :325
    sched.step()  # :327
    inner.zero_grad()  # :334
    # outer block (H=1: fires every real step), :336-353
    main_param = [p for g in inner.param_groups for p in g["params"]]
    for off, p in zip(offloaded, main_param):
        p.grad = off.data - p.data  # 1-worker all_reduce(AVG) = identity
        p.data = off.data.clone()
    outer.step()
    outer.zero_grad()
    offloaded = [p.data.detach().clone() for g in outer.param_groups for p in g["params"]]
    dt = time.perf_counter() - t0
    return {
        "value": bs * args.seq / dt,
        "unit": "tokens/s",
        "cores": torch.get_num_threads(),
        "kind": "port",
        "sample": f"full configs[0] real step ({bs}x{args.seq} tokens fwd+bwd+clip+AdamW"
                  f"+scheduler + outer pseudo-grad/Nesterov/re-offload at H=1), "
                  f"transformers fp32 on host cores ({dt:.1f}s)",
    }


if __name__ == "__main__":
    main()
