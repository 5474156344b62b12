"""One router config per process: python tools/gpu_sweep_one.py astar delta [iters]"""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np
astar, delta = float(sys.argv[1]), float(sys.argv[2])
iters = int(sys.argv[3]) if len(sys.argv) > 3 else 6
from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_placed_netlist
from parallel_eda_amd.route.router import net_rr_terminals
from parallel_eda_amd.route.gpu_router import GpuRouter
from parallel_eda_amd import rrgraph
arch = get_arch("LU32PEEng")
nl, pl = synth_placed_netlist(arch, fill=0.6, seed=1)
g = rrgraph.build_rr_graph(arch)
net_ids, src_rr, sink_ptr, sink_rr, ci = net_rr_terminals(nl, pl, g, arch)
r = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr,
              astar_fac=astar, delta_fac=delta)
crit = np.zeros(len(sink_rr), dtype=np.float32)
pres = 0.0
t_hist = []
for it in range(iters):
    t0 = time.perf_counter()
    r.reset_search_stats()
    print(f"  iter {it} start", flush=True)
    over, sd = r.route_iteration(crit, pres)
    t_hist.append(time.perf_counter() - t0)
    print(f"  iter {it} done over={over} retries={r.last_retries}", flush=True)
    pres = 0.5 if pres == 0.0 else pres * 1.3
    r.update_acc(1.0)
st = r.search_stats()
print(f"astar={astar} delta={delta}: t_last={t_hist[-1]:.2f}s t_all={[round(t,2) for t in t_hist]} "
      f"over={over} wl={r.wirelength()} scan/sink={st['scanned']//max(1,st['sinks'])} "
      f"touched/sink={st['touched']//max(1,st['sinks'])}", flush=True)
print("top cost nets (net, scans, nsinks, bb_area):", r.top_cost_nets(12), flush=True)
tot = int(r.t_net_scans.sum().item())
topsum = sum(t[1] for t in r.top_cost_nets(12))
print(f"top12 share of last-iter scans: {topsum}/{tot} = {topsum/max(1,tot):.1%}", flush=True)
