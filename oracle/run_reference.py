"""Execute the REFERENCE's own DiLoCo loop (train_diloco_torch.py) on CPU.

This shim runs `/root/reference/open_diloco/train_diloco_torch.py` — the
file north_star names as the parity target — as close to verbatim as this
offline CPU container allows, and dumps its per-step (Loss, lr) trace.  The
trace is committed under ``tests/golden/reference_trace_*.json`` and pins the
oracle restatement (oracle/diloco_oracle.py) against the reference's OWN
execution, not against the builder's reading of it.

The reference file cannot run unmodified here:

  * stale import at :24 (``get_grad_norm``/``register_hooks_log_activations``
    were deleted from utils.py) → a stub ``open_diloco.utils`` module provides
    inert versions (the only call sites, :292/:330, are behind
    ``log_activations_steps`` which we leave None);
  * hard-coded ``"cuda"`` device (:183 ``.to(local_rank)``, :310
    ``.to("cuda")``) → textual patch to ``"cpu"``, each asserted to occur the
    expected number of times;
  * ``dist.all_reduce(op=ReduceOp.AVG)`` (:345) is NCCL-only; on gloo it is
    replaced by SUM + divide-by-world-size — arithmetically the same mean, and
    bit-equal for 2 ranks (fp32 addition is commutative);
  * wandb / cyclopts are absent → stub modules; the wandb stub CAPTURES every
    ``wandb.log(dict)`` call on rank 0, which is exactly the reference's
    per-step {"Loss", "step", "lr"} record (:368-375);
  * the C4 + Mistral-tokenizer data pipeline (:201-229) needs network → the
    namespace's ``AutoTokenizer``/``load_dataset``/``split_dataset_by_node``/
    ``DataCollatorForLanguageModeling``/``DataLoader`` names are replaced so
    ``train_dataloader`` yields the seeded fake-token batches of
    utils.py:163-167 (``randint(3, vocab)``, ones mask, labels=ids), generator
    seed ``seed + 1337*rank`` — the same scheme the oracle and the product CLI
    use, so traces are directly comparable.

Everything else — model load, optimizer/scheduler construction, the inner
loop, clipping, the outer pseudo-grad/all-reduce/Nesterov block — executes
the reference's OWN lines.

``/root/reference`` exists only in the build container; the committed traces
travel, the shim does not need to run on the GPU box.

Usage (writes tests/golden/reference_trace_w2_h3.json):
    python -m oracle.run_reference
This is TEST INFRASTRUCTURE (see oracle/__init__.py).
"""

from __future__ import annotations

import json
import os
import subprocess
import sys
import tempfile
import types

REFERENCE_FILE = "/root/reference/open_diloco/train_diloco_torch.py"
HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = os.path.join(HERE, "..", "tests", "golden")
REF_MODEL = "/root/reference/tests/models/llama-2m-fresh"

# the committed trace's run shape (mirrors tests/golden/llama2m_w2_h3.json)
TRACE_CASES = {
    "reference_trace_w2_h3": dict(
        nproc=2, batch_size=16, per_device_train_batch_size=8, seq_length=128,
        local_steps=3, max_steps=6, lr=4e-4, outer_lr=0.7,
        warmup_steps=1000, total_steps=88_000, seed=42, vocab_size=1024,
    ),
    "reference_trace_w1_h1": dict(
        nproc=1, batch_size=16, per_device_train_batch_size=8, seq_length=128,
        local_steps=1, max_steps=6, lr=4e-4, outer_lr=0.7,
        warmup_steps=1000, total_steps=88_000, seed=42, vocab_size=1024,
    ),
    # the reference's own e2e test shape (test_train.py:24-39: batch 16/8,
    # seq 1024) at H=5 — matches tests/golden/llama2m_w2_h5_seq1024.json
    "reference_trace_w2_h5_seq1024": dict(
        nproc=2, batch_size=16, per_device_train_batch_size=8, seq_length=1024,
        local_steps=5, max_steps=10, lr=1e-2, outer_lr=0.7,
        warmup_steps=1000, total_steps=88_000, seed=42, vocab_size=1024,
    ),
}


def _patched_source() -> str:
    """Read the reference file and apply the minimal textual patches.

    Each replacement asserts its expected occurrence count so any drift in
    the reference text is caught loudly instead of silently mis-patching.
    """
    with open(REFERENCE_FILE) as f:
        src = f.read()

    def sub(old: str, new: str, count: int) -> None:
        nonlocal src
        found = src.count(old)
        assert found == count, f"expected {count}x {old!r} in reference, found {found}"
        src = src.replace(old, new)

    # :183  model placement (local_rank device index -> cpu)
    sub(".to(local_rank)", '.to("cpu")', 1)
    # :94 (eval, unused) and :310 (train batch) device moves
    sub('.to("cuda")', '.to("cpu")', 2)
    # :345  gloo has no ReduceOp.AVG; SUM + divide is the same mean
    sub(
        "dist.all_reduce(tensor=param.grad, op=dist.ReduceOp.AVG)",
        "dist.all_reduce(tensor=param.grad); param.grad.div_(dist.get_world_size())",
        1,
    )
    return src


def _install_stub_modules(wandb_records: list) -> None:
    """sys.modules stubs for the imports that cannot resolve offline."""
    import importlib.machinery

    def _spec(m: types.ModuleType) -> types.ModuleType:
        # accelerate probes wandb via importlib.util.find_spec, which raises
        # on a module whose __spec__ is None
        m.__spec__ = importlib.machinery.ModuleSpec(m.__name__, None)
        return m
    # open_diloco.utils with the two stale names (:24); never called here
    utils = types.ModuleType("open_diloco.utils")
    utils.get_grad_norm = lambda model: {}
    utils.register_hooks_log_activations = lambda model: ([], {})
    pkg = types.ModuleType("open_diloco")
    pkg.utils = utils
    sys.modules["open_diloco"] = _spec(pkg)
    sys.modules["open_diloco.utils"] = _spec(utils)

    wandb = types.ModuleType("wandb")
    wandb.run = types.SimpleNamespace(id="refshim")
    wandb.init = lambda **kw: None
    wandb.log = lambda d: wandb_records.append({k: v for k, v in d.items()})
    wandb.finish = lambda: None
    sys.modules["wandb"] = _spec(wandb)

    cyclopts = types.ModuleType("cyclopts")

    class App:
        def default(self, fn):
            return fn

        def __call__(self, *a, **kw):
            raise RuntimeError("shim calls main() directly")

    cyclopts.App = App
    sys.modules["cyclopts"] = _spec(cyclopts)


class _FakeLoader:
    """Replaces the C4 DataLoader: seeded fake-token batches
    (open_diloco/utils.py:163-167 semantics + mlm=False collation)."""

    def __init__(self, rank: int, n_batches: int, bs: int, seq_len: int,
                 vocab: int, seed: int):
        import torch

        self._torch = torch
        self.gen = torch.Generator().manual_seed(seed + 1337 * rank)
        self.n_batches = n_batches
        self.bs, self.seq_len, self.vocab = bs, seq_len, vocab

    def __iter__(self):
        torch = self._torch
        for _ in range(self.n_batches):
            ids = torch.randint(3, self.vocab, (self.bs, self.seq_len),
                                generator=self.gen, dtype=torch.int64)
            yield {"input_ids": ids, "attention_mask": torch.ones_like(ids),
                   "labels": ids.clone()}


def run_worker() -> None:
    """Executed under torchrun: run the patched reference main() once."""
    import torch
    import torch.distributed as dist

    case = json.loads(os.environ["REF_SHIM_CASE"])
    out_path = os.environ["REF_SHIM_OUT"]
    rank = int(os.environ["RANK"])
    os.environ.setdefault("LOCAL_RANK", str(rank))

    records: list = []
    _install_stub_modules(records)
    dist.init_process_group(backend="gloo")

    src = _patched_source()
    ns: dict = {"__name__": "reference_train_diloco_torch", "__file__": REFERENCE_FILE}
    exec(compile(src, REFERENCE_FILE, "exec"), ns)

    grad_acc = case["batch_size"] // case["per_device_train_batch_size"]
    n_batches = case["max_steps"] * grad_acc

    # replace the network-bound data pipeline + wandb access-check plumbing
    ns["AutoTokenizer"] = types.SimpleNamespace(
        from_pretrained=lambda *a, **kw: types.SimpleNamespace(pad_token=None))
    ns["load_dataset"] = lambda *a, **kw: types.SimpleNamespace(
        shuffle=lambda seed: None,
        map=lambda *a2, **kw2: {"train": "train", "validation": "validation"})
    ns["split_dataset_by_node"] = lambda ds, world_size, rank: ds
    ns["DataCollatorForLanguageModeling"] = lambda **kw: None
    ns["DataLoader"] = lambda ds, collate_fn=None, batch_size=8: _FakeLoader(
        rank, n_batches, case["per_device_train_batch_size"],
        case["seq_length"], case["vocab_size"], case["seed"])
    ns["check_checkpoint_path_access"] = lambda *a, **kw: None

    with tempfile.TemporaryDirectory() as tmp:
        ns["main"](
            batch_size=case["batch_size"],
            per_device_train_batch_size=case["per_device_train_batch_size"],
            seq_length=case["seq_length"],
            precision="32-true",
            model_name_or_path=REF_MODEL,
            lr=case["lr"],
            outer_lr=case["outer_lr"],
            warmup_steps=case["warmup_steps"],
            total_steps=case["total_steps"],
            local_steps=case["local_steps"],
            checkpoint_path=tmp,
            project="refshim",
        )

    if rank == 0:
        payload = {
            "source": REFERENCE_FILE,
            "model": REF_MODEL,
            "config": case,
            "records": [
                {"step": r["step"], "Loss": r["Loss"], "lr": r["lr"]}
                for r in records
            ],
        }
        with open(out_path, "w") as f:
            json.dump(payload, f, indent=1)
        print(f"wrote {len(records)} records to {out_path}")
    dist.barrier()
    dist.destroy_process_group()


def launch(case_name: str, out_path: str | None = None, port: int = 29511) -> str:
    """Run one trace case via torchrun (gloo, CPU); returns the output path."""
    case = TRACE_CASES[case_name]
    out_path = out_path or os.path.join(GOLDEN, f"{case_name}.json")
    env = dict(os.environ)
    env["REF_SHIM_CASE"] = json.dumps(case)
    env["REF_SHIM_OUT"] = out_path
    env["OMP_NUM_THREADS"] = env.get("OMP_NUM_THREADS", "4")
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={case['nproc']}",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        os.path.abspath(__file__), "worker",
    ]
    subprocess.run(cmd, env=env, check=True,
                   cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    return out_path


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "worker":
        run_worker()
    else:
        for i, name in enumerate(TRACE_CASES):
            launch(name, port=29511 + i)
