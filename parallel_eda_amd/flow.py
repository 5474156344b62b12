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
