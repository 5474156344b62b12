"""Round-2 opening GPU checklist, STAGED to the 90-GPU-minute budget:

  # core (~30-40 min incl. box setup): regressions + the two biggest A/Bs
  gpurun --timeout 2100 -- 'python tools/round2_entry.py core > gpurun_out/r2a.log 2>&1'
  # full tail (separate call, after reading r2a): bench + placer scale
  gpurun --timeout 2400 -- 'python tools/round2_entry.py full > gpurun_out/r2b.log 2>&1'

Order of information value per GPU-minute:
  core: gpu test suite -> calendar A/B (tseng quality, LU32 perf)
        -> partial-rip A/B (tseng) -> multi-domain STA -> het placer
  full: bitcoin bench w/ winning frontier -> LU32 partial A/B
        -> LU32 flow bench -> bgm-scale GPU anneal
Every stage keeps its own subprocess timeout so one hang can't eat the
whole call.
"""
import os, subprocess, sys, time
from pathlib import Path
ROOT = Path(__file__).resolve().parent.parent
os.chdir(ROOT)
STAGE = sys.argv[1] if len(sys.argv) > 1 else "core"

def run(name, cmd, env=None, timeout=600, stage="core"):
    if STAGE != stage and STAGE != "all":
        return 0
    print(f"\n===== {name}: {cmd}", flush=True)
    e = dict(os.environ); e.update(env or {})
    t0 = time.time()
    r = subprocess.run(cmd, shell=True, env=e, timeout=timeout)
    print(f"===== {name} rc={r.returncode} ({time.time()-t0:.0f}s)", flush=True)
    return r.returncode

run("gpu-tests", "python -m pytest tests/ -q -m gpu", timeout=900)
run("lu32-pingpong", "python tools/gpu_sweep_one.py 1.2 3.0 6", timeout=400)
run("lu32-calendar", "python tools/gpu_sweep_one.py 1.2 3.0 6",
    env={"PNR_CALENDAR": "1"}, timeout=400)
run("tseng-quality-calendar",
    "python -m pytest tests/test_gpu_router.py::test_gpu_route_matches_cpu_quality -q",
    env={"PNR_CALENDAR": "1"}, timeout=300)
run("btc-calendar", "python bench.py --config bitcoin_miner --steps 2 "
    "--warmup 2 --verbose", env={"PNR_CALENDAR": "1"}, timeout=900,
    stage="full")
run("partial-rip-tseng", "python tools/gpu_partial_ab.py tseng 0.6",
    timeout=300)
run("partial-rip-lu32", "python tools/gpu_partial_ab.py LU32PEEng 0.55",
    timeout=600, stage="full")
run("sta-domains-gpu", "python - <<'P'\n"
    "import sys; sys.path.insert(0, '.')\n"
    "import numpy as np\n"
    "from parallel_eda_amd.arch.archdef import get_arch\n"
    "from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch\n"
    "from parallel_eda_amd.timing.sta import STA\n"
    "from parallel_eda_amd.timing.gpu_sta import GpuSTA\n"
    "arch = get_arch('tseng')\n"
    "nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))\n"
    "rng = np.random.default_rng(3)\n"
    "bc = np.where(np.asarray(nl.block_is_seq) > 0,\n"
    "              rng.integers(0, 2, nl.num_blocks), -1).astype(np.int32)\n"
    "per = np.asarray([5e-9, 8e-9], dtype=np.float32)\n"
    "dly = rng.uniform(0.1e-9, 2e-9, nl.num_conns).astype(np.float32)\n"
    "wp_c, sl_c, cr_c = STA(nl, arch).analyze_domains(dly, bc, per)\n"
    "g = GpuSTA(nl, arch)\n"
    "wp_g, sl_g, cr_g = g.analyze_domains(dly, bc, per)\n"
    "print('slack match:', np.allclose(sl_c, sl_g, rtol=1e-4, atol=1e-12))\n"
    "print('crit match:', np.allclose(np.minimum(cr_c, 0.99), cr_g, rtol=1e-4))\n"
    "assert np.allclose(sl_c, sl_g, rtol=1e-4, atol=1e-12)\n"
    "P", timeout=300)
run("xcd-order-ab", "python tools/gpu_sweep_one.py 1.2 3.0 6",
    env={"PNR_XCD_ORDER": "1"}, timeout=400, stage="full")
run("flow-bench", "python tools/bench_flow.py LU32PEEng > "
    "gpurun_out/flow_lu32.json && cat gpurun_out/flow_lu32.json",
    timeout=900, stage="full")
run("het-gpu", "python - <<'P'\n"
    "import sys; sys.path.insert(0, '.')\n"
    "import numpy as np\n"
    "from parallel_eda_amd.arch.archdef import get_arch\n"
    "from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch\n"
    "from parallel_eda_amd.place.gpu_placer import anneal_place_gpu\n"
    "from parallel_eda_amd.place.placer import anneal_place\n"
    "# heterogeneous GPU placer (dormant round-1 path): legality + quality\n"
    "arch = get_arch('mem32K')\n"
    "nl = synth_netlist(spec_for_arch(arch, fill=0.45, seed=2))\n"
    "pl = anneal_place_gpu(nl, arch, seed=7, timing_tradeoff=0.0)\n"
    "tb = arch.tile_btype_grid(); gy = arch.ny + 2\n"
    "bt = np.asarray(nl.block_type)\n"
    "bad = sum(1 for b in range(nl.num_blocks)\n"
    "          if tb[pl.x[b] * gy + pl.y[b]] != bt[b])\n"
    "cpu = anneal_place(nl, arch, seed=7, timing_tradeoff=0.0)\n"
    "print(f'het GPU anneal: bb={pl.bb_cost:.0f} vs CPU {cpu.bb_cost:.0f} '\n"
    "      f'type-violations={bad}')\n"
    "assert bad == 0\n"
    "P", timeout=600)
run("placer-bgm", "python - <<'P'\n"
    "import sys; sys.path.insert(0, '.')\n"
    "import time, numpy as np\n"
    "from parallel_eda_amd.arch.archdef import get_arch\n"
    "from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch\n"
    "from parallel_eda_amd.place.gpu_placer import anneal_place_gpu\n"
    "arch = get_arch('bgm')\n"
    "nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=1))\n"
    "t0 = time.time()\n"
    "pl = anneal_place_gpu(nl, arch, seed=7, timing_tradeoff=0.0)\n"
    "print(f'bgm GPU anneal: bb={pl.bb_cost:.0f} temps={pl.stats[\"temps\"]} "
    "t={time.time()-t0:.0f}s')\n"
    "P", timeout=900, stage="full")
