import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.place.placer import anneal_place
from parallel_eda_amd.route.router import pathfinder_route, net_rr_terminals
from parallel_eda_amd.timing.sta import STA
from parallel_eda_amd import rrgraph
from parallel_eda_amd.flow import run_flow


@pytest.fixture(scope="module")
def tiny_placed():
    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.6, seed=2))
    pl = anneal_place(nl, arch, seed=2, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    return arch, nl, pl, g


def test_terminals(tiny_placed):
    arch, nl, pl, g = tiny_placed
    net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(nl, pl, g, arch)
    assert len(net_ids) > 0
    ty = np.asarray(g.type)
    assert (ty[src_rr] == 0).all()   # SOURCE
    assert (ty[sink_rr] == 1).all()  # SINK


def test_route_tiny_bb_only(tiny_placed):
    arch, nl, pl, g = tiny_placed
    res = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    assert res.success, f"unroutable: overused={res.overused}"
    assert res.wirelength > 0
    # check_routed ran inside pathfinder_route on success


def test_route_timing_driven(tiny_placed):
    arch, nl, pl, g = tiny_placed
    sta = STA(nl, arch)
    res = pathfinder_route(nl, pl, g, arch, sta=sta, max_iters=40)
    assert res.success
    assert res.crit_path_delay > 0
    # routed delays should give a plausible cpd (ns scale)
    assert 1e-10 < res.crit_path_delay < 1e-6


def test_full_flow_tseng():
    """BASELINE config 1: tseng-scale synthetic, CPU single-thread flow."""
    res = run_flow("tseng", seed=1, timing_driven=True, fill=0.5,
                   max_route_iters=50)
    assert res.route.success
    assert res.cpd > 0
    assert res.wirelength > 0
