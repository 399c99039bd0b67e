"""C-ABI boundary checks (no GPU needed): the kernel library loads and
exports every symbol declared in include/diloco_kernels.h."""

import ctypes
import os
import re

import pytest


def _header_symbols(repo_root):
    hdr = os.path.join(repo_root, "include", "diloco_kernels.h")
    with open(hdr) as f:
        text = f.read()
    syms = re.findall(r"^\s*(?:int|const char\*)\s+(dk_\w+)\s*\(", text, re.M)
    assert len(syms) >= 15, syms
    return syms


def test_kernel_lib_exports_all_header_symbols(repo_root):
    lib_path = os.path.join(repo_root, "opendiloco_amd", "libdiloco_kernels.so")
    if not os.path.exists(lib_path):
        pytest.skip("libdiloco_kernels.so not built (run python -m opendiloco_amd.build_ext)")
    lib = ctypes.CDLL(lib_path)
    missing = [s for s in _header_symbols(repo_root) if not hasattr(lib, s)]
    assert not missing, f"missing C-ABI symbols: {missing}"


def test_version_string(repo_root):
    lib_path = os.path.join(repo_root, "opendiloco_amd", "libdiloco_kernels.so")
    if not os.path.exists(lib_path):
        pytest.skip("libdiloco_kernels.so not built")
    lib = ctypes.CDLL(lib_path)
    lib.dk_version.restype = ctypes.c_char_p
    assert b"gfx950" in lib.dk_version()


def test_torch_binding_importable(repo_root):
    so = os.path.join(repo_root, "opendiloco_amd", "_diloco_C.so")
    if not os.path.exists(so):
        pytest.skip("_diloco_C.so not built")
    from opendiloco_amd.build_ext import load_binding

    mod = load_binding()
    for fn in ["rmsnorm_fwd", "rope", "swiglu_fwd", "ce_fwd", "attn_fwd", "attn_bwd",
               "attn_fwd_bsd", "attn_bwd_bsd", "qkv_rope_gather", "rope_scatter_",
               "accum_", "fused_adamw", "clip_grad_", "pseudo_grad", "outer_nesterov",
               "probe_mfma"]:
        assert hasattr(mod, fn), fn
