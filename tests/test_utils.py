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


def test_path_codec_roundtrip():
    """Bit-packed path codec (reference: path_codec.h): encode routed
    paths as edge indices at ceil(log2(max_deg+1)) bits/hop, round-trip
    exactly, and beat 32-bit node lists by >2x."""
    import numpy as np
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.router import pathfinder_route
    from parallel_eda_amd.utils.path_codec import PathCodec, encode_tree_paths
    from parallel_eda_amd import rrgraph

    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))
    pl = anneal_place(nl, arch, seed=3, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    res = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    assert res.success
    codec = PathCodec(g.row_ptr, g.edge_dst)
    ty = np.asarray(g.type)
    total_nodes = 0
    total_words = 0
    checked = 0
    from parallel_eda_amd.route.router import net_rr_terminals
    net_ids, _, _, _, _ = net_rr_terminals(nl, pl, g, arch)
    for inet in range(min(40, len(net_ids))):
        nodes, parents, sw, delay = res.router.tree(inet)
        nodes = np.asarray(nodes)
        parents = np.asarray(parents)
        sink_mask = ty[nodes] == 1
        packed = encode_tree_paths(codec, nodes, parents, sink_mask)
        for k, p in zip(np.nonzero(sink_mask)[0], packed):
            # reconstruct the root->sink path and compare
            rev = []
            i = int(k)
            while i >= 0:
                rev.append(int(nodes[i]))
                i = int(parents[i])
            ref = rev[::-1]
            got = codec.decode(ref[0], p)
            assert got == ref
            total_nodes += len(ref)
            total_words += len(p)
            checked += 1
    assert checked > 10
    packed_bits = total_words * 64
    assert packed_bits < total_nodes * 32 / 2, (packed_bits, total_nodes * 32)


def test_sdf_writer(tmp_path):
    import numpy as np
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.io.verilog import write_verilog, write_sdf

    arch = get_arch("tiny_het")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=2))
    delay = np.full(nl.num_conns, 1.25e-9, dtype=np.float32)
    vf = tmp_path / "a.v"
    sf = tmp_path / "a.sdf"
    write_verilog(str(vf), nl)
    write_sdf(str(sf), nl, arch, delay)
    v = vf.read_text()
    s = sf.read_text()
    assert "module" in v and "ram_seq" in v   # het instances emitted
    assert s.count("(INTERCONNECT ") == nl.num_conns
    assert "(1250.0:1250.0:1250.0)" in s      # 1.25 ns in ps
    assert '(CELLTYPE "dsp")' in s
    # DSP comb delay annotated
    d = arch.T_dsp * 1e12
    assert f"({d:.1f}:{d:.1f}:{d:.1f})" in s


def test_stats_net_costs(tmp_path):
    from parallel_eda_amd.utils.stats import StatsWriter
    sw = StatsWriter(str(tmp_path))
    sw.net_costs(3, [0, 5, 0, 7], rank=1)
    body = (tmp_path / "net_cost_iter_3_rank_1.txt").read_text()
    assert body == "1 5\n3 7\n"


def test_cli_settings_file(tmp_path):
    from parallel_eda_amd.__main__ import main
    cfg = tmp_path / "s.toml"
    cfg.write_text('fill = 0.4\nmax_router_iterations = 50\n')
    rc = main(["--synth", "tiny", "--seed", "3",
               "--settings", str(cfg)])
    assert rc == 0
    # unknown keys rejected
    bad = tmp_path / "b.toml"
    bad.write_text('no_such_option = 1\n')
    import pytest as _pt
    with _pt.raises(SystemExit):
        main(["--synth", "tiny", "--settings", str(bad)])


def test_tree_dot_dump(tmp_path):
    from parallel_eda_amd.utils.debug_dump import write_tree_dot
    f = tmp_path / "t.dot"
    write_tree_dot(str(f), [10, 11, 12], [-1, 0, 0], label="x")
    body = f.read_text()
    assert "digraph" in body and "n0 -> n1" in body and "n0 -> n2" in body


def test_power_estimate(tmp_path):
    """Block-level power model (reference: power/power.c power_total —
    dynamic + leakage from activities): components positive, total adds
    up, routing power scales with activity, .act file honored."""
    import numpy as np
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.router import pathfinder_route
    from parallel_eda_amd.utils.power import (estimate_power,
                                              read_activity_file,
                                              write_power_report)
    from parallel_eda_amd import rrgraph

    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))
    pl = anneal_place(nl, arch, seed=3, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    res = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    assert res.success
    p = estimate_power(nl, arch, g, res.router)
    assert p["total_W"] > 0
    assert abs(p["total_W"] - (p["routing_W"] + p["logic_W"] +
                               p["clock_W"] + p["leakage_W"])) < 1e-12
    # doubling activity doubles routing power
    p2 = estimate_power(nl, arch, g, res.router,
                        activities=np.full(nl.num_nets, 0.30))
    assert p2["routing_W"] == pytest.approx(2 * p["routing_W"], rel=1e-6)
    # activity file
    nl.names = [f"b{i}" for i in range(nl.num_blocks)]
    f = tmp_path / "x.act"
    drv0 = f"b{int(nl.net_driver[0])}"
    f.write_text(f"{drv0} 0.5\n")
    act = read_activity_file(str(f), nl)
    assert act[0] == 0.5 and act[1] == 0.15
    rpt = tmp_path / "p.rpt"
    write_power_report(str(rpt), p)
    assert "total_W" in rpt.read_text()


def test_svg_rendering(tmp_path):
    """Headless placement/routing SVG dumps (reference: graphics.c/draw.c
    interactive view -> SURVEY's dump-to-image)."""
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.router import pathfinder_route
    from parallel_eda_amd.utils.draw import (write_placement_svg,
                                             write_routing_svg)
    from parallel_eda_amd import rrgraph
    arch = get_arch("tiny_het")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=5))
    pl = anneal_place(nl, arch, seed=5, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    res = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=60)
    f1 = tmp_path / "p.svg"; f2 = tmp_path / "r.svg"
    write_placement_svg(str(f1), pl, nl, arch)
    write_routing_svg(str(f2), g, arch, res.router)
    a, b = f1.read_text(), f2.read_text()
    assert a.startswith("<svg") and a.rstrip().endswith("</svg>")
    assert b.count("<line") > 10           # wires drawn
    assert a.count("<rect") >= nl.num_blocks


def test_hip_build_artifacts_present():
    """Both gfx950 kernel libraries (release + debug-bounds) must be
    built in-tree — the GPU box receives them via the repo snapshot, and
    ops.hip() refuses to run without them (no silent CPU fallback)."""
    from parallel_eda_amd import ops
    import parallel_eda_amd as pkg
    from pathlib import Path
    root = Path(pkg.__file__).parent
    assert (root / "libpnr_hip.so").exists()
    assert (root / "libpnr_hip_dbg.so").exists()
    assert ops.hip_lib_path().name == "libpnr_hip.so"
    import os
    os.environ["PNR_HIP_DEBUG"] = "1"
    try:
        assert ops.hip_lib_path().name == "libpnr_hip_dbg.so"
    finally:
        del os.environ["PNR_HIP_DEBUG"]
