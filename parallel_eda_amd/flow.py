"""Full place-and-route flow driver.

Mirrors the reference's main flow (vpr/SRC/main.c:310 main ->
vpr_place_and_route -> place_and_route_new, place_and_route.c:51):
netlist -> SA placement -> PathFinder routing -> STA report.
"""
import time
from dataclasses import dataclass, field

from .arch.archdef import ArchDef, get_arch
from .io.synth import synth_netlist, spec_for_arch
from .place.placer import anneal_place
from .route.router import pathfinder_route
from .timing.sta import STA
from . import rrgraph


@dataclass
class FlowResult:
    placement: object
    route: object
    cpd: float
    wirelength: int
    times: dict = field(default_factory=dict)


def run_flow(arch_name: str = "tseng", seed: int = 1, timing_driven: bool = True,
             fill: float = 0.75, max_route_iters: int = 60, verbose: bool = False,
             engine: str = "cpu", netlist=None, arch: ArchDef = None):
    arch = arch or get_arch(arch_name)
    if netlist is None:
        netlist = synth_netlist(spec_for_arch(arch, fill=fill, seed=seed))
    sta = STA(netlist, arch) if timing_driven else None
    times = {}
    t0 = time.perf_counter()
    placement = anneal_place(netlist, arch, seed=seed,
                             timing_tradeoff=0.5 if timing_driven else 0.0,
                             sta=sta, verbose=verbose, engine=engine)
    times["place"] = time.perf_counter() - t0
    t0 = time.perf_counter()
    g = rrgraph.build_rr_graph(arch)
    times["rr_graph"] = time.perf_counter() - t0
    t0 = time.perf_counter()
    route = pathfinder_route(netlist, placement, g, arch, sta=sta,
                             max_iters=max_route_iters, verbose=verbose,
                             engine=engine)
    times["route"] = time.perf_counter() - t0
    return FlowResult(placement=placement, route=route,
                      cpd=route.crit_path_delay, wirelength=route.wirelength,
                      times=times)


def min_channel_width(netlist, placement, arch: ArchDef, w_lo=8, w_hi=None,
                      max_route_iters=40, engine="cpu", verbose=False):
    """Binary search for the minimum routable channel width.

    Reference: base/place_and_route.c:432 binary_search_place_and_route —
    placement fixed, the rr graph is rebuilt and routed at each candidate W.
    Returns (w_min, result_at_w_min).
    """
    import copy
    w_hi = w_hi or max(arch.W * 2, 32)
    best = None
    best_w = None

    def try_w(w):
        a = copy.copy(arch)
        a.W = w + (w % 2)
        g = rrgraph.build_rr_graph(a)
        res = pathfinder_route(netlist, placement, g, a, sta=None,
                               max_iters=max_route_iters, engine=engine)
        if verbose:
            print(f"W={a.W}: {'routed' if res.success else 'FAILED'} "
                  f"(overused={res.overused})")
        return res

    # make sure hi is routable; grow if not
    while True:
        res = try_w(w_hi)
        if res.success:
            best, best_w = res, w_hi
            break
        w_hi *= 2
        if w_hi > 4096:
            raise RuntimeError("unroutable even at W=4096")
    lo, hi = w_lo, w_hi
    while lo < hi:
        mid = (lo + hi) // 2
        res = try_w(mid)
        if res.success:
            best, best_w = res, mid
            hi = mid
        else:
            lo = mid + 1
    return best_w + (best_w % 2), best
