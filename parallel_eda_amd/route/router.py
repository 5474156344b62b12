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


class ConnMap:
    """Vectorized mapping between netlist connections and routed sinks.

    Routed sinks are deduped per (net, sink-tile); each netlist connection
    (net, sink-block) maps onto one routed sink. Used to scatter routed
    sink delays to connection delays for STA, and gather connection
    criticalities back (max per routed sink).
    """

    def __init__(self, conn_index, sink_ptr, num_conns, n_rsinks):
        conn_all, rsink_all = [], []
        for k, (idx, inv) in enumerate(conn_index):
            conn_all.append(idx)
            rsink_all.append(sink_ptr[k] + inv)
        if conn_all:
            self.conn = np.concatenate(conn_all).astype(np.int64)
            self.rsink = np.concatenate(rsink_all).astype(np.int64)
        else:
            self.conn = np.zeros(0, dtype=np.int64)
            self.rsink = np.zeros(0, dtype=np.int64)
        self.num_conns = num_conns
        self.n_rsinks = n_rsinks

    def conn_delays(self, sink_delays, out=None, fill=0.0):
        """Scatter routed-sink delays to connection delays. Connections
        with no routed sink (driver and sink share a tile) get `fill` —
        pass the intra-tile constant (arch.T_opin + arch.T_ipin) so STA
        sees the same delay the placer's delay model charges for
        same-tile connections (placer.analytic_delay_matrix d[0,0])."""
        if out is None:
            out = np.zeros(self.num_conns, dtype=np.float32)
        out[:] = fill
        out[self.conn] = sink_delays[self.rsink]
        return out

    def sink_crit(self, conn_crit, out=None, max_crit=0.99, crit_exp=1.0):
        """Max connection criticality per routed sink, clamped to max_crit.

        The clamp (reference: VPR max_criticality, router opts) is
        essential: at crit == 1.0 the congestion term (1-crit)*cong
        vanishes and fully-critical nets ignore pres_cost forever.
        """
        if out is None:
            out = np.zeros(self.n_rsinks, dtype=np.float32)
        else:
            out[:] = 0.0
        np.maximum.at(out, self.rsink, conn_crit[self.conn])
        if crit_exp != 1.0:
            np.power(out, crit_exp, out=out)
        np.minimum(out, max_crit, out=out)
        return out


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
                     engine: str = "cpu", rip_up_always: bool = False,
                     deterministic: bool = False, bb_factor: int = 4,
                     crit_exp: float = 1.0, max_criticality: float = 0.99,
                     incremental: bool = False,
                     crit_rip_threshold: float = 0.99,
                     incremental_start: int = 2,
                     full_resync_every: int = 2):
    """Timing-driven PathFinder: route all nets to feasibility.

    rip_up_always / deterministic reach the GPU engine (the CPU oracle
    rips all nets every iteration by construction and is serial, hence
    already deterministic)."""
    if engine == "gpu":
        from .gpu_router import pathfinder_route_gpu
        return pathfinder_route_gpu(
            netlist, placement, g, arch, sta=sta, max_iters=max_iters,
            pres_fac_init=pres_fac_init, pres_fac_mult=pres_fac_mult,
            acc_fac=acc_fac, astar_fac=astar_fac, verbose=verbose,
            rip_up_always=rip_up_always, deterministic=deterministic,
            bb_factor=bb_factor, crit_exp=crit_exp,
            max_criticality=max_criticality, incremental=incremental)
    cpu = ops.cpu()
    net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(
        netlist, placement, g, arch)
    opts = cpu.RouterOpts()
    opts.astar_fac = astar_fac
    router = cpu.SerialRouter(g, src_rr, sink_ptr, sink_rr, opts)

    n_rsinks = len(sink_rr)
    cmap = ConnMap(conn_index, sink_ptr, netlist.num_conns, n_rsinks)
    crit = np.zeros(n_rsinks, dtype=np.float32)
    conn_delay = np.zeros(netlist.num_conns, dtype=np.float32)
    intra_delay = float(arch.T_opin + arch.T_ipin)
    # iteration 1 is congestion-blind (VPR first_iter_pres_fac = 0;
    # matches the C++ oracle's documented schedule and the GPU driver)
    pres_fac = 0.0
    router.set_pres_fac(pres_fac)
    cpd = 0.0
    history = []
    it = 0
    overused = -1
    for it in range(1, max_iters + 1):
        rerouted = len(net_ids)
        resync = (full_resync_every > 0 and it > incremental_start and
                  (it - incremental_start) % full_resync_every == 0)
        if incremental and it > incremental_start and not resync:
            # selective + partial rip-up (reference phase-two +
            # route_tree_mark_congested_...): reroute congested nets and
            # nets with missing sinks; keep their clean subtrees
            todo = np.union1d(np.asarray(router.congested_nets()),
                              np.asarray(router.incomplete_nets()))
            if sta is not None and crit_rip_threshold <= 1.0:
                hot = np.nonzero(crit >= crit_rip_threshold)[0]
                if len(hot):
                    net_of_rsink = np.repeat(
                        np.arange(len(net_ids)), np.diff(sink_ptr))
                    todo = np.union1d(todo, net_of_rsink[hot])
            rerouted = len(todo)
            overused = router.route_subset_incremental(
                crit, todo.astype(np.int32),
                crit_rip_thr=crit_rip_threshold)
        else:
            overused = router.route_iteration(crit)
        history.append(dict(iter=it, overused=int(overused),
                            pops=router.heap_pops(), cpd=cpd,
                            rerouted=rerouted))
        if verbose:
            print(f"iter {it}: overused={overused} pops={router.heap_pops()} "
                  f"cpd={cpd*1e9:.2f}ns")
        if sta is not None:
            # net delays -> connection delays -> STA -> criticality
            cmap.conn_delays(router.sink_delays(), out=conn_delay,
                             fill=intra_delay)
            cpd, slack, c = sta.analyze(conn_delay)
            crit = cmap.sink_crit(c, max_crit=max_criticality,
                              crit_exp=crit_exp)
        if overused == 0 and router.unrouted_sinks() == 0:
            break
        pres_fac = pres_fac_init if it == 1 else pres_fac * pres_fac_mult
        router.update_costs(pres_fac, acc_fac)

    ok = overused == 0 and router.unrouted_sinks() == 0
    if router.unrouted_sinks() > 0 and verbose:
        print(f"WARNING: {router.unrouted_sinks()} sinks unreachable")
    if ok:
        valid, err = router.check_routed()
        if not valid:
            # forensics dump (reference: check_route_tree writes
            # error.dot on failure, router.cxx:145-200)
            import re
            from ..utils.debug_dump import write_tree_dot
            m = re.search(r"net (\d+)", err)
            if m:
                inet = int(m.group(1))
                nodes, parents, _sw, _d = router.tree(inet)
                write_tree_dot("error.dot", nodes, parents, g=g,
                               label=f"net {inet}: {err}")
                err += " (tree dumped to error.dot)"
            raise RuntimeError(f"check_route failed: {err}")
    return RouteResult(success=ok, iterations=it, overused=int(overused),
                       wirelength=int(router.total_wirelength()),
                       crit_path_delay=cpd,
                       stats={"history": history,
                              "num_routed_nets": len(net_ids)},
                       router=router)
