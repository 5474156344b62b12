import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np
from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.place.gpu_placer import anneal_place_gpu
from parallel_eda_amd.place.placer import anneal_place
arch2 = get_arch("tseng")
nl2 = synth_netlist(spec_for_arch(arch2, fill=0.5, seed=4))
pl_cpu = anneal_place(nl2, arch2, seed=7, timing_tradeoff=0.0)
print(f"CPU bb={pl_cpu.bb_cost:.1f}", flush=True)
for nm in (32, 64, 128):
    t0 = time.perf_counter()
    plg = anneal_place_gpu(nl2, arch2, seed=7, timing_tradeoff=0.0, n_moves=nm)
    print(f"n_moves={nm}: bb={plg.bb_cost:.1f} temps={plg.stats['temps']} "
          f"t={time.perf_counter()-t0:.1f}s", flush=True)
