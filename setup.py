"""In-tree build: `python setup.py build_ext --inplace`.

Builds:
  * parallel_eda_amd/_pnr_cpu.*.so  — host-side engine (pybind11, pure C++)
  * parallel_eda_amd/libpnr_hip.so  — CDNA4 HIP kernels (built by hipcc for
    gfx950; loaded via ctypes at first GPU use). Built by tools/build_hip.py,
    which this setup invokes when hipcc is available.
"""
import subprocess
import sys
from pathlib import Path

from setuptools import setup, Extension
import pybind11

ROOT = Path(__file__).resolve().parent

ext = Extension(
    "parallel_eda_amd._pnr_cpu",
    sources=["csrc/cpu/bindings.cpp"],
    depends=sorted(str(p) for p in (ROOT / "csrc" / "cpu").glob("*.*")),
    include_dirs=[pybind11.get_include(), "csrc/cpu"],
    extra_compile_args=["-O3", "-std=c++17", "-fvisibility=hidden", "-g0"],
    language="c++",
)

if __name__ == "__main__":
    setup(
        name="parallel_eda_amd",
        version="0.1.0",
        packages=["parallel_eda_amd"],
        ext_modules=[ext],
    )
    if "build_ext" in sys.argv:
        subprocess.check_call([sys.executable, str(ROOT / "tools" / "build_hip.py")])
