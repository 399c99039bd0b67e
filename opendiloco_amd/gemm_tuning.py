"""hipBLASLt/rocBLAS GEMM solution selection (torch TunableOp).

The dense projections run as library GEMMs; torch's default hipBLASLt
heuristics leave ~10% on the table for the llama-150m training shapes.  The
committed ``tunableop_gfx950.csv`` holds solutions tuned on an MI355X
(PYTORCH_TUNABLEOP_TUNING=1 over the bench shapes); ``enable_tuned_gemms``
loads it read-only so runs are reproducible.  Re-tune with
``PYTORCH_TUNABLEOP_TUNING=1 PYTORCH_TUNABLEOP_FILENAME=... python bench.py``.
"""

from __future__ import annotations

import os

TUNED_FILE = os.path.join(os.path.dirname(os.path.abspath(__file__)), "tunableop_gfx950.csv")


def enable_tuned_gemms(verbose: bool = False) -> bool:
    """Enable TunableOp with the shipped results (no tuning at runtime).
    No-op when the results file is missing or CUDA/ROCm is unavailable."""
    import torch

    if not torch.cuda.is_available() or not os.path.exists(TUNED_FILE):
        return False
    if os.environ.get("PYTORCH_TUNABLEOP_TUNING") == "1":
        return False  # explicit re-tuning run: leave env-driven behaviour alone
    t = torch.cuda.tunable
    t.enable(True)
    t.tuning_enable(False)
    ok = t.read_file(TUNED_FILE)
    if verbose:
        print(f"[gemm_tuning] loaded {TUNED_FILE}: {ok}")
    return bool(ok)
