"""GPU parameter sweep: router (astar_fac, delta_fac) + placer n_moves."""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np
from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_placed_netlist, synth_netlist, spec_for_arch
from parallel_eda_amd.route.router import net_rr_terminals
from parallel_eda_amd.route.gpu_router import GpuRouter
from parallel_eda_amd import rrgraph

# ---- router sweep on LU32 ----
arch = get_arch("LU32PEEng")
nl, pl = synth_placed_netlist(arch, fill=0.6, seed=1)
g = rrgraph.build_rr_graph(arch)
net_ids, src_rr, sink_ptr, sink_rr, ci = net_rr_terminals(nl, pl, g, arch)
for astar in (1.2, 1.6, 2.2):
    for delta in (1.5, 3.0, 6.0):
        r = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr,
                      astar_fac=astar, delta_fac=delta)
        crit = np.zeros(len(sink_rr), dtype=np.float32)
        pres = 0.0
        import torch
        t_hist = []
        over = -1
        for it in range(6):
            t0 = time.perf_counter()
            r.reset_search_stats()
            over, sd = r.route_iteration(crit, pres)
            t_hist.append(time.perf_counter() - t0)
            pres = 0.5 if pres == 0.0 else pres * 1.3
            r.update_acc(1.0)
        st = r.search_stats()
        wl = r.wirelength()
        print(f"astar={astar} delta={delta}: t_iter={t_hist[-1]:.2f}s "
              f"(first {t_hist[0]:.2f}) over={over} wl={wl} "
              f"scan/sink={st['scanned']//max(1,st['sinks'])}", flush=True)
        del r
        torch.cuda.empty_cache()

# ---- placer n_moves sweep on tseng ----
from parallel_eda_amd.place.gpu_placer import anneal_place_gpu
from parallel_eda_amd.place.placer import anneal_place
arch2 = get_arch("tseng")
nl2 = synth_netlist(spec_for_arch(arch2, fill=0.5, seed=4))
pl_cpu = anneal_place(nl2, arch2, seed=7, timing_tradeoff=0.0)
print(f"CPU bb={pl_cpu.bb_cost:.1f}", flush=True)
for nm in (32, 64, 128, 256):
    t0 = time.perf_counter()
    plg = anneal_place_gpu(nl2, arch2, seed=7, timing_tradeoff=0.0, n_moves=nm)
    print(f"n_moves={nm}: bb={plg.bb_cost:.1f} temps={plg.stats['temps']} "
          f"t={time.perf_counter()-t0:.1f}s", flush=True)
