"""TEST INFRASTRUCTURE — CPU oracle for the DiLoCo hot path.

This package is a CPU restatement of the reference's pure-torch DiLoCo loop
(`open_diloco/train_diloco_torch.py`, PrimeIntellect-ai/OpenDiloco @
2024-10-08), which the north_star names as the parity target.  It exists
ONLY as the checker: the only permitted importers are ``tests/``,
``__graft_entry__.smoke()`` and ``bench.py``'s ``cpu_baseline`` leg.  The
product path (``opendiloco_amd``) must never import, call or fall back to
anything in here.

Parity pinning, two legs:

1. **Weights**: ``tests/models/llama-2m`` holds the reference's checked-in
   ``tests/models/llama-2m-fresh`` fixture tensors, re-saved bit-exactly via
   ``transformers`` by ``oracle/gen_fixture.py`` (run in the build container,
   which holds the read-only reference checkout; the round-trip is verified
   tensor-by-tensor before writing).
2. **Execution**: ``oracle/run_reference.py`` EXECUTES the reference's own
   ``train_diloco_torch.py`` loop (exec of its actual source with minimal
   asserted patches: cpu device, gloo SUM/size for the NCCL-only AVG, stub
   wandb/cyclopts/data pipeline) on 1- and 2-proc gloo CPU and commits its
   per-step (Loss, lr) traces under ``tests/golden/reference_trace_*.json``;
   ``tests/test_oracle.py`` requires the oracle to match those traces
   BIT-EQUALLY — stricter than the reference's own test tolerances
   (tests/test_training/test_train.py:76-83: loss atol 1e-3, lr exact).

The model is ``transformers.LlamaForCausalLM`` itself, the exact model the
reference calls (train_fsdp.py:171-174, train_diloco_torch.py:183).  Golden
vectors for the wider test matrix are generated on the pinned oracle by
``oracle/gen_golden.py`` and committed under ``tests/golden/``.
"""

from oracle.diloco_oracle import run_diloco_oracle, OracleConfig  # noqa: F401
