import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.place.placer import anneal_place
from parallel_eda_amd.route.router import pathfinder_route, net_rr_terminals
from parallel_eda_amd.utils.stats import StatsWriter, routing_stats
from parallel_eda_amd.flow import min_channel_width
from parallel_eda_amd import rrgraph


@pytest.fixture(scope="module")
def tiny_routed():
    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))
    pl = anneal_place(nl, arch, seed=3, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    res = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    assert res.success
    net_ids, *_ = net_rr_terminals(nl, pl, g, arch)
    return arch, nl, pl, g, res, net_ids


def test_stats_writer(tmp_path, tiny_routed):
    arch, nl, pl, g, res, net_ids = tiny_routed
    sw = StatsWriter(str(tmp_path / "stats"))
    sw.iteration(1, 5, rerouted=10, heap_pops=100, cpd=1e-9)
    sw.iteration(2, 0, rerouted=2, heap_pops=50, cpd=0.9e-9)
    data = sw.final(True, res.wirelength, 0.9e-9)
    assert data["iterations"] == 2
    assert (tmp_path / "stats" / "iter_stats.txt").exists()
    assert (tmp_path / "stats" / "final_stats.txt").exists()


def test_routing_stats(tiny_routed):
    arch, nl, pl, g, res, net_ids = tiny_routed
    st = routing_stats(g, arch, net_ids, lambda k: res.router.tree(k))
    assert st["total_wirelength"] == res.wirelength
    assert st["total_segments"] > 0
    assert st["total_bends"] >= 0


def test_min_channel_width():
    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))
    pl = anneal_place(nl, arch, seed=3, timing_tradeoff=0.0)
    w, res = min_channel_width(nl, pl, arch, w_lo=4, w_hi=16,
                               max_route_iters=30)
    assert res.success
    assert 4 <= w <= 32
    # min width is genuinely minimal-ish: W-2 must fail or equal lower bound
    if w > 4:
        import copy
        a2 = copy.copy(arch)
        a2.W = w - 2
        g2 = rrgraph.build_rr_graph(a2)
        res2 = pathfinder_route(nl, pl, g2, a2, sta=None, max_iters=30)
        assert not res2.success


def test_serial_num_and_mem(tiny_routed):
    from parallel_eda_amd.utils.stats import routing_serial_num, mem_usage_mb
    arch, nl, pl, g, res, net_ids = tiny_routed
    s1 = routing_serial_num(net_ids, lambda k: res.router.tree(k))
    s2 = routing_serial_num(net_ids, lambda k: res.router.tree(k))
    assert s1 == s2 and len(s1) == 16
    assert mem_usage_mb() > 10


def test_verilog_writer(tmp_path, tiny_routed):
    from parallel_eda_amd.io.verilog import write_verilog
    arch, nl, pl, g, res, net_ids = tiny_routed
    nl.names = [f"b{i}" for i in range(nl.num_blocks)]
    p = tmp_path / "out.v"
    write_verilog(p, nl)
    text = p.read_text()
    assert text.startswith("module")
    assert "endmodule" in text
    assert text.count("wire ") == nl.num_nets
