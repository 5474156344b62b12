"""Distributed full place-and-route flow (BASELINE config 4's shape:
"bgm full place+route on 8 GPUs").

Composes the two distributed drivers over one process group:
  1. anneal_place_dist — strip-sharded SA with per-temperature fusion
  2. pathfinder_route_dist — spatially-partitioned PathFinder with
     occ all-reduce, selective reroute, elastic shrink
plus replicated STA. Every rank returns the identical result.

CPU engines under gloo here (world-2/4 tested); the same drivers take
the GPU engines over RCCL in round 2 (GpuPlacer.set_move_region and
GpuEngine are already wired).
"""
import numpy as np


def run_flow_dist(netlist, arch, rank=0, world_size=1, seed=7,
                  timing_driven=True, max_route_iters=60,
                  incremental=True, verbose=False):
    """Returns dict(place=Placement, route=dict from pathfinder_route_dist,
    wirelength, cpd) — identical on every rank."""
    from ..timing.sta import STA
    from ..route.router import net_rr_terminals, ConnMap
    from .. import rrgraph, ops
    from .dist import CpuEngine, DistRouteLoop, pathfinder_route_dist
    from .dist_place import anneal_place_dist

    sta = STA(netlist, arch) if timing_driven else None
    pl = anneal_place_dist(netlist, arch, rank=rank, world_size=world_size,
                           seed=seed,
                           timing_tradeoff=0.5 if timing_driven else 0.0,
                           sta=sta, verbose=verbose)
    g = rrgraph.build_rr_graph(arch)
    net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(
        netlist, pl, g, arch)
    cmap = ConnMap(conn_index, sink_ptr, netlist.num_conns, len(sink_rr))
    cpu = ops.cpu()
    router = cpu.SerialRouter(g, src_rr, sink_ptr, sink_rr, cpu.RouterOpts())
    engine = CpuEngine(router, g.num_nodes)
    xlow = np.asarray(g.xlow); ylow = np.asarray(g.ylow)
    bb = np.zeros((len(net_ids), 4), dtype=np.int16)
    for n in range(len(net_ids)):
        t = np.r_[src_rr[n], sink_rr[sink_ptr[n]:sink_ptr[n + 1]]]
        bb[n] = (xlow[t].min(), ylow[t].min(), xlow[t].max(), ylow[t].max())
    loop = DistRouteLoop(engine, len(net_ids), bb, len(sink_rr), sink_ptr,
                         rank=rank, world_size=world_size)
    res = pathfinder_route_dist(loop, cmap, sta, max_iters=max_route_iters,
                                incremental=incremental, verbose=verbose)
    # global wirelength from the (rank-identical) occupancy
    occ = engine.occ_tensor().cpu().numpy()
    ty = np.asarray(g.type)
    chan = (ty == 4) | (ty == 5)
    xl = np.asarray(g.xlow); xh = np.asarray(g.xhigh)
    yl = np.asarray(g.ylow); yh = np.asarray(g.yhigh)
    seg_len = (xh - xl + yh - yl + 1).astype(np.int64)
    wl = int((occ[chan] * seg_len[chan]).sum())
    return dict(place=pl, route=res, wirelength=wl, cpd=res["cpd"],
                success=res["success"])
