"""Round-2 GPU probe 2: fixed-calendar A/B, LU32 full flow, bitcoin placer.

Run: gpurun --timeout 2100 -- 'python tools/r2probe2.py > gpurun_out/r2probe2.log 2>&1'
"""
import os
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
os.chdir(ROOT)


def run(name, cmd, env=None, timeout=600):
    print(f"\n===== {name}: {cmd}", flush=True)
    e = dict(os.environ)
    e.update(env or {})
    t0 = time.time()
    try:
        r = subprocess.run(cmd, shell=True, env=e, timeout=timeout)
        rc = r.returncode
    except subprocess.TimeoutExpired:
        rc = -9
        print(f"===== {name} TIMEOUT", flush=True)
    print(f"===== {name} rc={rc} ({time.time()-t0:.0f}s)", flush=True)
    return rc


run("gpu-tests", "python -m pytest tests/ -q -m gpu", timeout=900)
run("lu32-calendar-fixed", "python tools/gpu_sweep_one.py 1.2 3.0 5",
    env={"PNR_CALENDAR": "1"}, timeout=420)
run("gpu-tests-calendar", "python -m pytest tests/test_gpu_router.py -q",
    env={"PNR_CALENDAR": "1"}, timeout=300)
run("flow-lu32", "python bench.py --config LU32PEEng --fill 0.55 --steps 1 "
    "--warmup 0 --verbose", timeout=900)
run("btc-placer", "python - <<'P'\n"
    "import sys, time; sys.path.insert(0, '.')\n"
    "import numpy as np\n"
    "from parallel_eda_amd.arch.archdef import get_arch\n"
    "from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch\n"
    "from parallel_eda_amd.place.gpu_placer import anneal_place_gpu\n"
    "from parallel_eda_amd.timing.sta import STA\n"
    "arch = get_arch('bitcoin_miner')\n"
    "t0 = time.time()\n"
    "nl = synth_netlist(spec_for_arch(arch, fill=0.6, seed=1))\n"
    "print(f'synth {time.time()-t0:.1f}s blocks={nl.num_blocks} '\n"
    "      f'nets={nl.num_nets}', flush=True)\n"
    "sta = STA(nl, arch)\n"
    "t0 = time.time()\n"
    "pl = anneal_place_gpu(nl, arch, seed=7, timing_tradeoff=0.5, sta=sta)\n"
    "print(f'bitcoin GPU anneal: {time.time()-t0:.1f}s bb={pl.bb_cost:.0f} '\n"
    "      f'temps={pl.stats[\"temps\"]} move_lim={pl.stats[\"move_lim\"]}',\n"
    "      flush=True)\n"
    "P", timeout=1200)
print("\nPROBE2 DONE", flush=True)
