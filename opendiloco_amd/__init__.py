"""opendiloco_amd — MI355X-native DiLoCo training framework.

A from-scratch rebuild of OpenDiLoCo's data-parallel hot path (the DiLoCo
round: H inner AdamW steps of Llama forward/backward per worker, then an
outer step of pseudo-gradient averaging + Nesterov SGD) for AMD Instinct
MI355X (gfx950): hand-written CDNA4 HIP kernels behind a C-ABI extension,
one process per GPU, a single RCCL all-reduce over xGMI for the outer step.

API surface mirrors the reference (PrimeIntellect-ai/OpenDiloco @ 2024-10-08):
  - ``DiLoCoOptimizer`` — open_diloco/hivemind_diloco.py:303
  - ``train_fsdp`` CLI  — open_diloco/train_fsdp.py
"""

__version__ = "0.1.0"

_LAZY = {
    "LlamaModelConfig": "opendiloco_amd.llama_config",
    "DiLoCoOptimizer": "opendiloco_amd.diloco",
    "AllReduceStrategy": "opendiloco_amd.diloco",
    "LlamaForCausalLM": "opendiloco_amd.model",
}


def __getattr__(name):
    if name in _LAZY:
        import importlib

        mod = importlib.import_module(_LAZY[name])
        return getattr(mod, name)
    raise AttributeError(name)
