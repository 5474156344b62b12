"""Debug: GPU route with per-iteration overuse detail.

Usage: python tools/gpu_debug_route.py [arch] [fill] [seed] [td]
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np
import torch

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.place.placer import anneal_place
from parallel_eda_amd.route.router import net_rr_terminals, ConnMap
from parallel_eda_amd.route.gpu_router import GpuRouter
from parallel_eda_amd.timing.sta import STA
from parallel_eda_amd import rrgraph

arch_name = sys.argv[1] if len(sys.argv) > 1 else "tiny"
fill = float(sys.argv[2]) if len(sys.argv) > 2 else 0.6
seed = int(sys.argv[3]) if len(sys.argv) > 3 else 2
td = len(sys.argv) > 4 and sys.argv[4] == "td"

arch = get_arch(arch_name)
nl = synth_netlist(spec_for_arch(arch, fill=fill, seed=seed))
pl = anneal_place(nl, arch, seed=seed, timing_tradeoff=0.0)
g = rrgraph.build_rr_graph(arch)

net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(nl, pl, g, arch)
router = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr)
sta = STA(nl, arch) if td else None
cmap = ConnMap(conn_index, sink_ptr, nl.num_conns, len(sink_rr))
crit = np.zeros(len(sink_rr), dtype=np.float32)
conn_delay = np.zeros(nl.num_conns, dtype=np.float32)
ty = np.asarray(g.type)
cap = np.asarray(g.capacity)

pres_fac = 0.0
for it in range(1, 41):
    overused, sd = router.route_iteration(crit, pres_fac)
    occ = router.t_occ.cpu().numpy()
    ov = np.nonzero(occ > cap)[0]
    detail = [(int(v), int(ty[v]), int(occ[v]), int(cap[v])) for v in ov[:8]]
    owners = []
    if len(ov):
        v0 = int(ov[0])
        for n in range(router.num_nets):
            nodes, _, _, _ = router.get_tree(n)
            cnt = int((nodes == v0).sum())
            if cnt:
                inbb = (router.bb[n, 0] <= np.asarray(g.xlow)[v0] <= router.bb[n, 2]
                        and router.bb[n, 1] <= np.asarray(g.ylow)[v0] <= router.bb[n, 3])
                dup = len(nodes) != len(np.unique(nodes))
                owners.append((n, cnt, bool(inbb), dup, tuple(router.bb[n])))
    print(f"iter {it} pres={pres_fac:.2f} overused={overused} detail={detail} "
          f"owners={owners}", flush=True)
    if td:
        cmap.conn_delays(sd, out=conn_delay)
        cpd, slack, c = sta.analyze(conn_delay)
        crit = cmap.sink_crit(c)
        print(f"  cpd={cpd*1e9:.3f}ns critmax={crit.max():.3f}", flush=True)
    if overused == 0:
        print(f"FEASIBLE at iter {it}")
        break
    pres_fac = 0.5 if pres_fac == 0.0 else pres_fac * 1.3
    router.update_acc(1.0)
print("recount:", router.check_occ_recount())
