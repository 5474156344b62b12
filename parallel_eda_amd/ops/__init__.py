"""Extension loading.

`cpu()` returns the pybind11 host-engine module (_pnr_cpu).
`hip()` returns the ctypes handle to the CDNA4 kernel library
(libpnr_hip.so) — and FAILS LOUDLY if a GPU is present but the library is
missing: the HIP path must be the one that runs on a GPU box, never a
silent CPU fallback.
"""
import ctypes
import os
from pathlib import Path

_PKG_DIR = Path(__file__).resolve().parent.parent

_cpu_mod = None
_hip_lib = None


def cpu():
    global _cpu_mod
    if _cpu_mod is None:
        from parallel_eda_amd import _pnr_cpu
        _cpu_mod = _pnr_cpu
    return _cpu_mod


def hip_lib_path() -> Path:
    if os.environ.get("PNR_HIP_DEBUG"):
        return _PKG_DIR / "libpnr_hip_dbg.so"
    return _PKG_DIR / "libpnr_hip.so"


def hip():
    """ctypes handle to the gfx950 kernel library."""
    global _hip_lib
    if _hip_lib is None:
        p = hip_lib_path()
        if not p.exists():
            raise RuntimeError(
                f"libpnr_hip.so not found at {p}; build it with "
                f"`python setup.py build_ext --inplace` (hipcc required). "
                f"Refusing to fall back to CPU on a GPU box.")
        _hip_lib = ctypes.CDLL(str(p), mode=ctypes.RTLD_GLOBAL)
    return _hip_lib


def have_gpu() -> bool:
    if os.environ.get("PNR_FORCE_CPU"):
        return False
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False
