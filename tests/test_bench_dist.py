"""Launch-path test of bench.py exactly as the driver does (torchrun,
world 2) — CPU-oracle fallback over gloo."""
import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def test_bench_torchrun_world2():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", str(ROOT / "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--config", "tseng", "--fill", "0.3"],
        capture_output=True, text=True, timeout=600, env=env, cwd=ROOT)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert lines, r.stdout[-2000:]
    out = json.loads(lines[-1])
    assert out["n_gpus"] == 2
    assert out["value"] > 0
    assert out["scaling"] == "strong"


def test_bench_single_cpu_fallback():
    r = subprocess.run(
        [sys.executable, str(ROOT / "bench.py"), "--gpus", "1",
         "--steps", "2", "--warmup", "1", "--config", "tseng",
         "--fill", "0.3"],
        capture_output=True, text=True, timeout=600, cwd=ROOT)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    out = json.loads(lines[-1])
    assert out["n_gpus"] == 1 and out["value"] > 0
