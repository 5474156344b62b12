"""Full place-and-route flows (BASELINE configs 4-5: full flow on 1-8
GPUs; the bench.py headline step).

Composes the two distributed drivers over one process group:
  1. anneal_place_dist — strip-sharded SA with per-temperature fusion
  2. pathfinder_route_dist — spatially-partitioned PathFinder with
     occ all-reduce, selective reroute, elastic shrink
plus replicated STA. Every rank returns the identical result.

run_flow_dist = CPU engines (gloo, world-2/4 tested); run_flow_gpu =
the same drivers over the CDNA4 kernels (GpuPlacer/GpuRouter), RCCL
collectives when world_size > 1.
"""
import time

import numpy as np


def run_flow_gpu(netlist, arch, g, dev_graph=None, rank=0, world_size=1,
                 device="cuda:0", seed=7, timing_driven=True,
                 max_route_iters=80, incremental=True, verbose=False,
                 inner_num=1.0, check=False):
    """One complete GPU place+route flow to feasibility (the headline
    benchmark step). The rr graph g (and optionally its device-resident
    upload dev_graph) is the fixed device model — passed in, like the
    FPGA itself. Everything netlist-dependent (placement anneal, net
    terminals, route trees, routing to 0 overused, STA) happens here.

    Returns dict(place, route, wirelength, cpd, success, phase_s) —
    rank-identical."""
    from ..timing.sta import STA
    from ..route.router import net_rr_terminals, ConnMap
    from ..route.gpu_router import GpuRouter, DevGraph
    from ..place.gpu_placer import anneal_place_gpu
    from .dist import GpuEngine, DistRouteLoop, pathfinder_route_dist
    from .dist_place import anneal_place_dist

    sta = STA(netlist, arch) if timing_driven else None
    tt = 0.5 if timing_driven else 0.0

    t0 = time.perf_counter()
    if world_size > 1:
        pl = anneal_place_dist(netlist, arch, rank=rank,
                               world_size=world_size, seed=seed,
                               timing_tradeoff=tt, sta=sta, engine="gpu",
                               device=device, verbose=verbose,
                               inner_num=inner_num)
    else:
        pl = anneal_place_gpu(netlist, arch, seed=seed, timing_tradeoff=tt,
                              sta=sta, device=device, verbose=verbose,
                              inner_num=inner_num)
    t_place = time.perf_counter() - t0

    t0 = time.perf_counter()
    net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(
        netlist, pl, g, arch)
    cmap = ConnMap(conn_index, sink_ptr, netlist.num_conns, len(sink_rr))
    if dev_graph is None:
        dev_graph = DevGraph(g, arch, device)
    router = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr,
                       device=device, dev_graph=dev_graph)
    engine = GpuEngine(router)
    loop = DistRouteLoop(engine, len(net_ids), router.bb, len(sink_rr),
                         sink_ptr, rank=rank, world_size=world_size)
    t_setup = time.perf_counter() - t0

    t0 = time.perf_counter()
    res = pathfinder_route_dist(
        loop, cmap, sta, max_iters=max_route_iters, incremental=incremental,
        verbose=verbose, intra_delay=float(arch.T_opin + arch.T_ipin))
    t_route = time.perf_counter() - t0
    if check and res["success"]:
        if not router.check_occ_recount():
            raise RuntimeError("flow: occ recount mismatch after routing")
    wl = router.wirelength()
    return dict(place=pl, route=res, wirelength=wl, cpd=res["cpd"],
                success=res["success"], router=router,
                phase_s={"place": t_place, "route_setup": t_setup,
                         "route": t_route})


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
