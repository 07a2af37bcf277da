"""Device ops: HIP/CDNA4 kernels + their torch reference implementations.

The HIP extension is built IN-TREE (cordum_amd/ops/_build/) so the .so
travels with the repo snapshot to GPU boxes. On a machine with a GPU the
extension is REQUIRED: device-path entry points raise if it is missing
rather than silently falling back to eager torch.
"""
from __future__ import annotations

import os
from pathlib import Path

import torch

_HERE = Path(__file__).resolve().parent
BUILD_DIR = _HERE / "_build"
SOURCES = [str(_HERE / "hip" / "cordum_kernels.hip")]
EXT_NAME = "cordum_hip_ops"

_ext = None
_load_error: Exception | None = None


def build_extension(verbose: bool = False):
    """Compile the HIP extension for gfx950 (cross-compiles without a GPU)."""
    global _ext, _load_error
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    BUILD_DIR.mkdir(parents=True, exist_ok=True)
    from torch.utils.cpp_extension import load

    _ext = load(
        name=EXT_NAME,
        sources=SOURCES,
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        verbose=verbose,
        with_cuda=True,
    )
    _load_error = None
    build_capv2_native()
    return _ext


def build_capv2_native(verbose: bool = False) -> bool:
    """Compile the native CAP v2 codec (plain C++/pybind11, no torch/HIP dep)
    in-tree next to its source so the .so travels with the snapshot."""
    import subprocess
    import sys
    import sysconfig

    import pybind11

    native_dir = _HERE.parent / "protocol" / "native"
    src = native_dir / "capv2_codec.cpp"
    out = native_dir / "_capv2_native.so"
    if out.exists() and out.stat().st_mtime >= src.stat().st_mtime:
        return True
    cmd = [
        "g++", "-O3", "-shared", "-fPIC", "-std=c++17",
        f"-I{sysconfig.get_paths()['include']}",
        f"-I{pybind11.get_include()}",
        str(src), "-o", str(out),
    ]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        if verbose:
            print(r.stderr, file=sys.stderr)
        return False
    return True


def _try_import_prebuilt():
    """Import an already-built .so without invoking the compiler."""
    global _ext, _load_error
    so = BUILD_DIR / f"{EXT_NAME}.so"
    if not so.exists():
        _load_error = FileNotFoundError(f"{so} not built")
        return None
    import importlib.util

    try:
        spec = importlib.util.spec_from_file_location(EXT_NAME, so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
        _load_error = None
        return mod
    except Exception as e:  # pragma: no cover
        _load_error = e
        return None


def get_ext(required: bool | None = None):
    """Returns the HIP extension module.

    required=None (default): required exactly when a GPU is visible — a GPU
    box must run the native kernels, never a silent eager fallback."""
    global _ext, _load_error
    if _ext is not None:
        return _ext
    _try_import_prebuilt()
    if _ext is None and torch.cuda.is_available():
        try:
            build_extension()
        except Exception as e:
            _load_error = e
    if required is None:
        required = torch.cuda.is_available()
    if _ext is None and required:
        raise RuntimeError(
            f"cordum_hip_ops extension unavailable on a GPU host: {_load_error}. "
            "Run cordum_amd.ops.build_extension() (hipcc, gfx950) first."
        )
    return _ext


def has_ext() -> bool:
    return get_ext(required=False) is not None
