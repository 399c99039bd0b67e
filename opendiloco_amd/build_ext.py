"""In-tree build of the HIP kernel library + torch binding (gfx950).

Two artifacts, both living inside the package directory so they travel with
the repo snapshot to the GPU box:

  opendiloco_amd/libdiloco_kernels.so  — pure C-ABI kernel library
      (hipcc --offload-arch=gfx950, no torch dependency; the drop-in
      boundary declared in include/diloco_kernels.h)
  opendiloco_amd/_diloco_C.so          — thin torch (pybind) binding that
      links the C-ABI library (csrc/binding.cpp)

`python -m opendiloco_amd.build_ext` builds both.
"""

from __future__ import annotations

import os
import shutil
import subprocess
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
REPO_ROOT = os.path.dirname(PKG_DIR)
CSRC = os.path.join(PKG_DIR, "csrc")
KERNEL_SOURCES = ["elementwise.hip", "ce.hip", "attn.hip"]
KERNELS_LIB = os.path.join(PKG_DIR, "libdiloco_kernels.so")
BINDING_SO = os.path.join(PKG_DIR, "_diloco_C.so")


def _newest_mtime(paths):
    return max(os.path.getmtime(p) for p in paths)


def build_kernels_lib(force: bool = False) -> str:
    srcs = [os.path.join(CSRC, s) for s in KERNEL_SOURCES]
    hdrs = [os.path.join(CSRC, "dk_common.h"), os.path.join(REPO_ROOT, "include", "diloco_kernels.h")]
    if not force and os.path.exists(KERNELS_LIB) and os.path.getmtime(KERNELS_LIB) >= _newest_mtime(srcs + hdrs):
        return KERNELS_LIB
    cmd = [
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC", "-shared",
        f"-I{os.path.join(REPO_ROOT, 'include')}",
        *srcs, "-o", KERNELS_LIB,
    ]
    print("[build_ext]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return KERNELS_LIB


def build_binding(force: bool = False) -> str:
    src = os.path.join(CSRC, "binding.cpp")
    if (not force and os.path.exists(BINDING_SO)
            and os.path.getmtime(BINDING_SO) >= os.path.getmtime(src)
            and os.path.getmtime(BINDING_SO) >= os.path.getmtime(KERNELS_LIB)):
        return BINDING_SO
    from torch.utils import cpp_extension

    build_dir = os.path.join(PKG_DIR, "build")
    os.makedirs(build_dir, exist_ok=True)
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    cpp_extension.load(
        name="_diloco_C",
        sources=[src],
        extra_cflags=["-O2", "-D__HIP_PLATFORM_AMD__=1"],
        extra_include_paths=[os.path.join(REPO_ROOT, "include"), "/opt/rocm/include"],
        # rpath: $ORIGIN for the shipped copy living next to libdiloco_kernels.so
        # in the package dir; the absolute PKG_DIR covers the import done from
        # the build dir during this build (and resolves on the GPU box too,
        # where /root/repo is a symlink to the snapshot).
        extra_ldflags=[f"-L{PKG_DIR}", "-ldiloco_kernels", "-lrocblas", "-lhipblaslt", "-Wl,-rpath,$ORIGIN",
                       f"-Wl,-rpath,{PKG_DIR}"],
        build_directory=build_dir,
        is_python_module=False,  # just build; importing is done from the package copy
        verbose=False,
    )
    shutil.copy2(os.path.join(build_dir, "_diloco_C.so"), BINDING_SO)
    return BINDING_SO


def build(force: bool = False) -> None:
    build_kernels_lib(force=force)
    build_binding(force=force)


def load_binding():
    """Import the built pybind module from the package directory.

    Never triggers a rebuild: on a GPU box the .so files shipped with the
    snapshot are used as-is."""
    if not os.path.exists(BINDING_SO):
        raise ImportError(
            "opendiloco_amd HIP extension not built: run `python -m opendiloco_amd.build_ext` "
            f"(missing {BINDING_SO})"
        )
    import importlib.util

    import torch  # noqa: F401  (the binding links against torch libs)

    spec = importlib.util.spec_from_file_location("opendiloco_amd._diloco_C", BINDING_SO)
    mod = importlib.util.module_from_spec(spec)
    sys.modules["opendiloco_amd._diloco_C"] = mod
    spec.loader.exec_module(mod)
    return mod


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print("built:", KERNELS_LIB, BINDING_SO)
