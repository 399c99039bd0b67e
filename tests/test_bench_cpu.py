"""CPU-side smoke of the bench.py contract: the driver runs
`python bench.py ...` at round end — a broken import/flag surface there is
a silent zero, so exercise the CPU-reachable legs here."""

import json
import os
import subprocess
import sys

import pytest


def test_cpu_baseline_only_runs(repo_root):
    """`--cpu-baseline-only` runs the full configs[0]-shaped loop (at the 2m
    shape for speed) and prints one JSON object with the contract keys."""
    r = subprocess.run(
        [sys.executable, "bench.py", "--cpu-baseline-only", "--model", "llama-2m",
         "--seq", "128"],
        cwd=repo_root, capture_output=True, text=True, timeout=600,
        env={**os.environ, "OMP_NUM_THREADS": "4"})
    assert r.returncode == 0, r.stderr[-2000:]
    line = json.loads(r.stdout.strip().splitlines()[-1])
    assert line["unit"] == "tokens/s"
    assert line["kind"] == "port"
    assert line["cores"] > 0
    assert line["value"] > 0
    assert "outer" in line["sample"]  # the timed region includes the outer block


def test_bench_requires_gpu_for_training_leg(repo_root):
    """Without a GPU the training leg must fail loudly (no CPU fallback on
    the product path), not silently measure something else."""
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU present")
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--no-cpu-baseline"],
        cwd=repo_root, capture_output=True, text=True, timeout=300)
    assert r.returncode != 0
    assert "needs a GPU" in (r.stdout + r.stderr)
