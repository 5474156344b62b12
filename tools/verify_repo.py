"""One-command repo verification (CPU box, no GPU needed).

Round-2 opening sanity on the CPU side before spending GPU minutes:
cold build, full CPU suite, het flow, a soak sample, and the bench
tool's CPU smoke. Exits nonzero on the first failure.

  python tools/verify_repo.py            # ~2-3 min
  python tools/verify_repo.py --fast     # skip the cold rebuild
"""
import argparse
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def run(name, cmd, timeout=1800):
    print(f"==== {name}: {' '.join(cmd)}", flush=True)
    t0 = time.time()
    r = subprocess.run(cmd, cwd=ROOT, timeout=timeout)
    print(f"==== {name} rc={r.returncode} ({time.time()-t0:.0f}s)",
          flush=True)
    if r.returncode != 0:
        sys.exit(f"verify_repo: {name} FAILED")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--fast", action="store_true")
    args = ap.parse_args()
    py = sys.executable
    if not args.fast:
        run("build", [py, "-c",
                      "import __graft_entry__; __graft_entry__.build()"])
    run("cpu-suite", [py, "-m", "pytest", "tests/", "-x", "-q",
                      "-m", "not gpu"])
    run("het-flow", [py, "-c",
        "from parallel_eda_amd.flow import run_flow; "
        "r = run_flow('mem32K', seed=2, fill=0.45); "
        "assert r.route.success"])
    run("soak-sample", [py, "tools/soak.py", "flows",
                        "--seeds", "9000:9030"])
    run("dist-soak-sample", [py, "tools/soak.py", "dist",
                             "--seeds", "9100:9106"])
    run("flow-bench-cpu", [py, "tools/bench_flow.py", "tseng",
                           "--placer", "cpu", "--router", "cpu"])
    print("verify_repo: ALL GREEN")
    return 0


if __name__ == "__main__":
    sys.exit(main())
