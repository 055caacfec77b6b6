"""HIP/CDNA4 kernel extension loader.

Builds in-tree (sirius_amd/ops/_build) so the compiled .so travels with the
repo snapshot to GPU boxes. On a GPU machine the native kernels are
mandatory: a failed build/import raises instead of silently falling back
to eager torch (the CPU path keeps the torch reference implementations).
"""

from __future__ import annotations

import os

import torch

_ext = None
_ext_zgemm = None
_ext_radial = None
_tried = False
_tried_radial = False

_SRC_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "src")
_BUILD_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_build")


def build_extensions(verbose: bool = False):
    """Compile the HIP extensions for gfx950 (cross-compiles fine on CPU-only
    boxes; hipcc needs no GPU present)."""
    global _ext, _ext_zgemm, _tried
    _tried = True
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(_BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    _ext = load(
        name="sirius_amd_ops",
        sources=[os.path.join(_SRC_DIR, "scf_ops.hip")],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        build_directory=_BUILD_DIR,
        verbose=verbose,
    )
    zdir = os.path.join(_BUILD_DIR, "zgemm")
    os.makedirs(zdir, exist_ok=True)
    _ext_zgemm = load(
        name="sirius_amd_zgemm",
        sources=[os.path.join(_SRC_DIR, "zgemm_gram.hip")],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        build_directory=zdir,
        verbose=verbose,
    )
    build_radial(verbose=verbose)
    build_capi(verbose=verbose)
    return _ext


def build_capi(verbose: bool = False):
    """Build libsirius_amd.so (the embedded-CPython C API shim)."""
    import subprocess
    import sys

    api_dir = os.path.abspath(os.path.join(os.path.dirname(_SRC_DIR),
                                           os.pardir, "api"))
    src = os.path.join(api_dir, "sirius_amd_api.cpp")
    lib = os.path.join(api_dir, "libsirius_amd.so")
    if not os.path.exists(src):
        return None
    if os.path.exists(lib) and os.path.getmtime(lib) >= os.path.getmtime(src):
        return lib
    inc = subprocess.run(["python3-config", "--includes"],
                         capture_output=True, text=True)
    cmd = ["g++", "-O2", "-shared", "-fPIC", "-o", lib, src] \
        + inc.stdout.split() \
        + [f"-lpython{sys.version_info.major}.{sys.version_info.minor}"]
    subprocess.run(cmd, check=True)
    if verbose:
        print(f"built {lib}")
    return lib


def build_radial(verbose: bool = False):
    """Compile the CPU radial-ODE extension (LAPW radial solver)."""
    global _ext_radial, _tried_radial
    _tried_radial = True
    from torch.utils.cpp_extension import load

    rdir = os.path.join(_BUILD_DIR, "radial")
    os.makedirs(rdir, exist_ok=True)
    _ext_radial = load(
        name="sirius_amd_radial",
        sources=[os.path.join(_SRC_DIR, "radial_ode.cpp")],
        extra_cflags=["-O3"],
        build_directory=rdir,
        verbose=verbose,
    )
    return _ext_radial


def get_radial():
    """The radial-ODE solver extension (CPU; required for the FP-LAPW
    branch everywhere — raises if it cannot build)."""
    global _ext_radial
    if _ext_radial is None and not _tried_radial:
        build_radial()
    if _ext_radial is None:
        raise RuntimeError("sirius_amd radial-ODE extension failed to build")
    return _ext_radial


def get_ext(required: bool | None = None):
    """Return the extension module; None when unavailable on CPU.

    required defaults to torch.cuda.is_available(): on a GPU box the HIP
    kernels must load — failure raises loudly.
    """
    global _ext, _tried
    if _ext is not None:
        return _ext
    if required is None:
        required = torch.cuda.is_available()
    if not _tried:
        try:
            build_extensions()
        except Exception as e:
            _ext = None
            if required:
                raise RuntimeError(
                    f"sirius_amd HIP extension failed to build/load on a GPU "
                    f"machine — native kernels are mandatory there: {e}") from e
    if _ext is None and required:
        raise RuntimeError("sirius_amd HIP extension unavailable on GPU machine")
    return _ext


def get_zgemm(required: bool | None = None):
    """The MFMA fp64 zgemm extension (None on CPU).

    required defaults to torch.cuda.is_available(): on a GPU box the MFMA
    kernels are the compute path — a missing extension raises instead of
    silently falling back to rocBLAS."""
    if required is None:
        required = torch.cuda.is_available()
    get_ext(required)
    if _ext_zgemm is None and required:
        raise RuntimeError(
            "sirius_amd MFMA zgemm extension unavailable on GPU machine")
    return _ext_zgemm


def available() -> bool:
    try:
        return get_ext(required=False) is not None
    except Exception:
        return False
