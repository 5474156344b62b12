"""Determinism diff-tests (reference: SURVEY section 4.4 — deterministic
routers exist so a parallel run can be diffed against ground truth).

The GPU router's deterministic mode = fixed bb-disjoint wave schedule +
deterministic 64-bit atomicMin tie-breaks; two runs from the same seed
must produce bit-identical congestion state.
"""
import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.place.placer import anneal_place
from parallel_eda_amd.route.router import net_rr_terminals
from parallel_eda_amd import rrgraph


def _case():
    arch = get_arch("tseng")
    nl = synth_netlist(spec_for_arch(arch, fill=0.45, seed=13))
    pl = anneal_place(nl, arch, seed=13, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    return arch, nl, pl, g


def test_cpu_router_deterministic():
    arch, nl, pl, g = _case()
    from parallel_eda_amd.route.router import pathfinder_route
    r1 = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    r2 = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    assert r1.success and r2.success
    assert np.array_equal(np.asarray(r1.router.occ()),
                          np.asarray(r2.router.occ()))
    assert r1.wirelength == r2.wirelength


@pytest.mark.gpu
def test_gpu_router_deterministic_mode():
    arch, nl, pl, g = _case()
    from parallel_eda_amd.route.gpu_router import GpuRouter

    def run():
        net_ids, src_rr, sink_ptr, sink_rr, _ = net_rr_terminals(
            nl, pl, g, arch)
        r = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr,
                      deterministic=True)
        crit = np.zeros(len(sink_rr), dtype=np.float32)
        pres = 0.0
        for it in range(30):
            over, sd = r.route_iteration(crit, pres)
            if over == 0:
                break
            pres = 0.5 if pres == 0.0 else pres * 1.3
            r.update_acc(1.0)
        return over, r.t_occ.cpu().numpy(), r.wirelength(), sd

    over1, occ1, wl1, sd1 = run()
    over2, occ2, wl2, sd2 = run()
    assert over1 == over2
    assert np.array_equal(occ1, occ2), "deterministic mode: occ differs"
    assert wl1 == wl2
    assert np.array_equal(sd1, sd2), "deterministic mode: delays differ"
