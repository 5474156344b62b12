#!/usr/bin/env python3
"""Build the CDNA4 HIP kernel library (gfx950) in-tree.

hipcc cross-compiles without a GPU; the resulting .so travels to the GPU box
with the repo snapshot. Output: parallel_eda_amd/libpnr_hip.so
"""
import shutil
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
SRC_DIR = ROOT / "csrc" / "hip"
OUT = ROOT / "parallel_eda_amd" / "libpnr_hip.so"


def main():
    hipcc = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
    if not Path(hipcc).exists():
        print("build_hip: hipcc not found; skipping HIP build", file=sys.stderr)
        return 0
    srcs = sorted(SRC_DIR.glob("*.hip"))
    if not srcs:
        print("build_hip: no .hip sources yet; skipping", file=sys.stderr)
        return 0
    # skip if up to date
    if OUT.exists():
        newest = max(s.stat().st_mtime for s in list(srcs) + list(SRC_DIR.glob("*.h")))
        if OUT.stat().st_mtime > newest:
            print(f"build_hip: {OUT.name} up to date")
            return 0
    cmd = [
        hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17",
        "-fPIC", "-shared", "-fvisibility=default",
        "-o", str(OUT),
    ] + [str(s) for s in srcs]
    print("build_hip:", " ".join(cmd))
    subprocess.check_call(cmd, cwd=ROOT)
    # debug variant with device-side bounds checks (loaded via
    # PNR_HIP_DEBUG=1; see ops/__init__.py)
    dbg = OUT.with_name("libpnr_hip_dbg.so")
    cmd_dbg = [
        hipcc, "--offload-arch=gfx950", "-O1", "-g", "-std=c++17",
        "-DPNR_DEBUG_BOUNDS", "-fPIC", "-shared", "-fvisibility=default",
        "-o", str(dbg),
    ] + [str(s) for s in srcs]
    subprocess.check_call(cmd_dbg, cwd=ROOT)
    return 0


if __name__ == "__main__":
    sys.exit(main())
