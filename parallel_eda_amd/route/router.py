"""PathFinder negotiated-congestion routing driver.

The outer rip-up-and-reroute loop (reference: route_timing.c:85
try_timing_driven_route; parallel outer loop
partitioning_multi_sink_delta_stepping_route.cxx:5939-6365) lives here and
is shared by the CPU oracle (csrc/cpu/route_serial.cpp) and the GPU
wavefront engine (route/gpu_router.py + csrc/hip/router_kernel.hip).
"""
from dataclasses import dataclass, field

import numpy as np

from ..arch.archdef import ArchDef
from .. import ops


def net_rr_terminals(netlist, placement, g, arch: ArchDef):
    """Map each net to rr (source SOURCE node, sink SINK nodes).

    Reference: parallel_route/init.cxx:244 init_nets (net_rr_terminals).
    Nets whose driver and all sinks share a tile are dropped (no routing
    needed); returns (net_ids, src_rr, sink_ptr, sink_rr).
    """
    ts = np.asarray(g.tile_source)
    tk = np.asarray(g.tile_sink)
    gy = arch.ny + 2
    bx, by = placement.x, placement.y
    tile_of_block = bx.astype(np.int64) * gy + by.astype(np.int64)

    net_ids, src_rr, sink_ptr, sink_rr = [], [], [0], []
    nd = netlist.net_driver
    sp = netlist.net_sink_ptr
    ss = netlist.net_sinks
    conn_index = []  # index into netlist.net_sinks for each routed sink
    for n in range(netlist.num_nets):
        drv_tile = tile_of_block[nd[n]]
        sinks = ss[sp[n]:sp[n + 1]]
        sink_tiles = tile_of_block[sinks]
        keep = sink_tiles != drv_tile
        if not keep.any():
            continue
        net_ids.append(n)
        src_rr.append(ts[drv_tile])
        kept_tiles = sink_tiles[keep]
        # dedupe sink tiles (multiple sinks on one tile route once; their
        # delays are equal through the shared SINK node)
        uniq, inv = np.unique(kept_tiles, return_inverse=True)
        sink_rr.extend(tk[uniq].tolist())
        sink_ptr.append(sink_ptr[-1] + len(uniq))
        idx = np.arange(sp[n], sp[n + 1])[keep]
        conn_index.append((idx, inv))
    return (np.asarray(net_ids, dtype=np.int64),
            np.asarray(src_rr, dtype=np.int32),
            np.asarray(sink_ptr, dtype=np.int64),
            np.asarray(sink_rr, dtype=np.int32),
            conn_index)


@dataclass
class RouteResult:
    success: bool
    iterations: int
    overused: int
    wirelength: int
    crit_path_delay: float
    stats: dict = field(default_factory=dict)
    router: object = None


def pathfinder_route(netlist, placement, g, arch: ArchDef, sta=None,
                     max_iters: int = 60, pres_fac_init: float = 0.5,
                     pres_fac_mult: float = 1.3, acc_fac: float = 1.0,
                     astar_fac: float = 1.2, verbose: bool = False,
                     engine: str = "cpu"):
    """Timing-driven PathFinder: route all nets to feasibility."""
    if engine == "gpu":
        from .gpu_router import pathfinder_route_gpu
        return pathfinder_route_gpu(
            netlist, placement, g, arch, sta=sta, max_iters=max_iters,
            pres_fac_init=pres_fac_init, pres_fac_mult=pres_fac_mult,
            acc_fac=acc_fac, astar_fac=astar_fac, verbose=verbose)
    cpu = ops.cpu()
    net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(
        netlist, placement, g, arch)
    opts = cpu.RouterOpts()
    opts.astar_fac = astar_fac
    router = cpu.SerialRouter(g, src_rr, sink_ptr, sink_rr, opts)

    n_rsinks = len(sink_rr)
    crit = np.zeros(n_rsinks, dtype=np.float32)
    conn_delay = np.zeros(netlist.num_conns, dtype=np.float32)
    pres_fac = pres_fac_init
    router.set_pres_fac(pres_fac)
    cpd = 0.0
    history = []
    it = 0
    overused = -1
    for it in range(1, max_iters + 1):
        overused = router.route_iteration(crit)
        history.append(dict(iter=it, overused=int(overused),
                            pops=router.heap_pops(), cpd=cpd))
        if verbose:
            print(f"iter {it}: overused={overused} pops={router.heap_pops()} "
                  f"cpd={cpd*1e9:.2f}ns")
        if sta is not None:
            # net delays -> connection delays -> STA -> criticality
            sd = router.sink_delays()
            for (idx, inv), k in zip(conn_index, range(len(net_ids))):
                conn_delay[idx] = sd[sink_ptr[k]:sink_ptr[k + 1]][inv]
            cpd, slack, c = sta.analyze(conn_delay)
            # map connection crits back to routed-sink crits (max over conns
            # sharing a sink tile)
            for (idx, inv), k in zip(conn_index, range(len(net_ids))):
                seg = np.zeros(sink_ptr[k + 1] - sink_ptr[k], dtype=np.float32)
                np.maximum.at(seg, inv, c[idx])
                crit[sink_ptr[k]:sink_ptr[k + 1]] = seg
        if overused == 0:
            break
        pres_fac = pres_fac_init if it == 1 else pres_fac * pres_fac_mult
        router.update_costs(pres_fac, acc_fac)

    ok = overused == 0
    if ok:
        valid, err = router.check_routed()
        if not valid:
            raise RuntimeError(f"check_route failed: {err}")
    return RouteResult(success=ok, iterations=it, overused=int(overused),
                       wirelength=int(router.total_wirelength()),
                       crit_path_delay=cpd,
                       stats={"history": history,
                              "num_routed_nets": len(net_ids)},
                       router=router)
