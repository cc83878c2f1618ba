"""In-tree build of the fengshen_amd HIP extension for gfx950.

Produces fengshen_amd/ops/_C.so (importable as fengshen_amd.ops._C).  The .so
is built IN-TREE so it travels with the repo snapshot to GPU boxes (a JIT
cache under ~/.cache would not).  hipcc cross-compiles gfx950 without a GPU.

Usage: python -m fengshen_amd.ops.build
"""
from __future__ import annotations

import glob
import os
import shutil
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_OPS_DIR, "csrc")
_BUILD_DIR = os.path.join(_OPS_DIR, "_build")
_TARGET = os.path.join(_OPS_DIR, "_C.so")


def sources():
    srcs = [os.path.join(_CSRC, "bindings.cpp"),
            os.path.join(_CSRC, "kernels.hip")]
    fa = os.path.join(_CSRC, "flash_attn.hip")
    if os.path.exists(fa):
        srcs.append(fa)
    return srcs


def needs_rebuild() -> bool:
    if not os.path.exists(_TARGET):
        return True
    t = os.path.getmtime(_TARGET)
    deps = sources() + glob.glob(os.path.join(_CSRC, "*.h"))
    return any(os.path.getmtime(s) > t for s in deps)


def build(force: bool = False, verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    if not force and not needs_rebuild():
        if verbose:
            print(f"[fengshen_amd.ops.build] up to date: {_TARGET}")
        return _TARGET
    from torch.utils.cpp_extension import load

    os.makedirs(_BUILD_DIR, exist_ok=True)
    load(
        name="_C",
        sources=sources(),
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950",
                           "-ffast-math"],
        build_directory=_BUILD_DIR,
        verbose=verbose,
        is_python_module=False,
        keep_intermediates=True,
    )
    built = os.path.join(_BUILD_DIR, "_C.so")
    if not os.path.exists(built):
        cands = glob.glob(os.path.join(_BUILD_DIR, "_C*.so"))
        if not cands:
            raise RuntimeError(f"extension build produced no .so in {_BUILD_DIR}")
        built = cands[0]
    shutil.copy2(built, _TARGET)
    if verbose:
        print(f"[fengshen_amd.ops.build] built {_TARGET}")
    build_data_helpers(force=force, verbose=verbose)
    return _TARGET


_DATA_DIR = os.path.normpath(os.path.join(_OPS_DIR, "..", "data"))
_HELPERS_SRC = os.path.join(_DATA_DIR, "csrc", "helpers.cpp")
_HELPERS_TARGET = os.path.join(_DATA_DIR, "_helpers.so")


def build_data_helpers(force: bool = False, verbose: bool = True) -> str:
    """CPU pybind11 index-builder extension (ref helpers.cpp equivalent)."""
    if not force and os.path.exists(_HELPERS_TARGET) and \
            os.path.getmtime(_HELPERS_TARGET) > os.path.getmtime(_HELPERS_SRC):
        return _HELPERS_TARGET
    import subprocess
    import sysconfig

    import pybind11

    cmd = [
        "c++", "-O3", "-std=c++17", "-shared", "-fPIC",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        _HELPERS_SRC, "-o", _HELPERS_TARGET,
    ]
    subprocess.check_call(cmd)
    if verbose:
        print(f"[fengshen_amd.ops.build] built {_HELPERS_TARGET}")
    return _HELPERS_TARGET


if __name__ == "__main__":
    build(force="--force" in sys.argv)
