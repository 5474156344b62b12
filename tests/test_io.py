import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.arch.xml_parser import parse_arch_xml
from parallel_eda_amd.io.blif import parse_blif
from parallel_eda_amd.io.pack import pack_blif
from parallel_eda_amd.io.place_file import write_place, read_place
from parallel_eda_amd.io.route_file import write_route
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.place.placer import anneal_place
from parallel_eda_amd.route.router import pathfinder_route, net_rr_terminals
from parallel_eda_amd.timing.sta import STA
from parallel_eda_amd import rrgraph

BLIF = """
# test circuit
.model top
.inputs a b c clk
.outputs y z
.names a b n1
11 1
.names n1 c n2
11 1
.latch n2 q re clk 0
.names q n1 y
11 1
.names q z
1 1
.end
"""

ARCH_XML = """
<architecture>
 <complexblocklist>
  <pb_type name="io" capacity="4">
   <input name="outpad" num_pins="1"/>
   <output name="inpad" num_pins="1"/>
  </pb_type>
  <pb_type name="clb">
   <input name="I" num_pins="22"/>
   <output name="O" num_pins="6"/>
   <fc default_in_type="frac" default_in_val="0.15"
       default_out_type="frac" default_out_val="0.1"/>
   <T_setup value="6.6e-11"/>
   <delay_constant max="2.6e-10"/>
  </pb_type>
 </complexblocklist>
 <segmentlist>
  <segment length="4" Rmetal="201" Cmetal="3.1e-14"/>
 </segmentlist>
 <switchlist>
  <switch type="mux" name="0" R="551" Cin="7.7e-16" Tdel="5.8e-11"/>
 </switchlist>
</architecture>
"""


def test_blif_parse_and_pack():
    m = parse_blif(BLIF)
    assert m.inputs == ["a", "b", "c", "clk"]
    assert m.outputs == ["y", "z"]
    kinds = sorted(p.kind for p in m.prims)
    assert kinds.count("latch") == 1
    assert kinds.count("names") == 4
    arch = get_arch("tiny")
    nl, cluster_of, names = pack_blif(m, arch, n_ble=4)
    assert nl.num_blocks >= 7  # 4 in + 2 out + >=1 clb
    # packed netlist is routable end-to-end on a tiny grid
    from parallel_eda_amd.timing.sta import STA
    sta = STA(nl, arch)  # levelizes => acyclic
    assert sta.num_levels >= 1


def test_arch_xml_parse():
    a = parse_arch_xml(ARCH_XML, nx=10, ny=10, W=40)
    assert a.nx == 10 and a.W == 40
    assert a.clb_in == 22 and a.clb_out == 6
    assert a.io_cap == 4
    assert a.L == 4
    assert a.R_wire == pytest.approx(201.0)
    assert a.fc_in == max(1, round(0.15 * 40))
    assert a.T_sw == pytest.approx(5.8e-11)
    # graph builds and validates from an XML-derived arch
    g = rrgraph.build_rr_graph(a)
    assert rrgraph.check_rr_graph(g, a)


def test_place_route_files_roundtrip(tmp_path):
    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))
    pl = anneal_place(nl, arch, seed=3, timing_tradeoff=0.0)
    pfile = tmp_path / "out.place"
    write_place(pfile, pl, nl, arch)
    pl2 = read_place(pfile, nl)
    assert np.array_equal(pl.x, pl2.x)
    assert np.array_equal(pl.y, pl2.y)
    assert np.array_equal(pl.slot, pl2.slot)

    g = rrgraph.build_rr_graph(arch)
    res = pathfinder_route(nl, pl, g, arch, sta=None, max_iters=40)
    assert res.success
    net_ids, src_rr, sink_ptr, sink_rr, _ = net_rr_terminals(nl, pl, g, arch)
    rfile = tmp_path / "out.route"
    write_route(rfile, g, arch, net_ids, lambda k: res.router.tree(k),
                netlist=nl)
    text = rfile.read_text()
    assert "Routing:" in text
    assert text.count("Net ") == len(net_ids)
    assert "SOURCE" in text and "SINK" in text and "CHAN" in text


def test_net_file_roundtrip(tmp_path):
    from parallel_eda_amd.io.net_file import write_net, read_net, check_netlist
    arch = get_arch("tiny")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=5))
    nl.names = [f"b{i}" for i in range(nl.num_blocks)]
    errs, dangling = check_netlist(nl)
    assert not errs
    p = tmp_path / "t.net"
    write_net(p, nl)
    nl2 = read_net(p)
    assert nl2.num_blocks == nl.num_blocks
    assert nl2.num_nets == nl.num_nets
    # same connectivity as multisets of (driver, sinks)
    def canon(n):
        out = []
        for i in range(n.num_nets):
            s = sorted(n.net_sinks[n.net_sink_ptr[i]:n.net_sink_ptr[i+1]].tolist())
            out.append((int(n.net_driver[i]), tuple(s)))
        return sorted(out)
    assert canon(nl) == canon(nl2)


def test_cli_synth_flow(tmp_path):
    from parallel_eda_amd.__main__ import main
    rc = main(["--synth", "tiny", "--fill", "0.5", "--seed", "3",
               "--out_place", str(tmp_path / "o.place"),
               "--out_route", str(tmp_path / "o.route"),
               "--stats_dir", str(tmp_path / "stats")])
    assert rc == 0
    assert (tmp_path / "o.place").exists()
    assert (tmp_path / "o.route").exists()
    assert (tmp_path / "stats" / "final_stats.txt").exists()


def test_cli_blif_xml_flow(tmp_path):
    """End-to-end: BLIF + arch.xml through the CLI (pack/place/route)."""
    from parallel_eda_amd.__main__ import main
    blif = tmp_path / "c.blif"
    blif.write_text(BLIF)
    xml = tmp_path / "a.xml"
    xml.write_text(ARCH_XML)
    rpt = tmp_path / "t.rpt"
    sdc = tmp_path / "c.sdc"
    sdc.write_text("create_clock -period 10.0 -name clk\n")
    rc = main([str(blif), str(xml), "--route_chan_width", "20",
               "--sdc", str(sdc), "--timing_report", str(rpt)])
    assert rc == 0
    assert "Critical path delay" in rpt.read_text()


BLIF2CLK = """
.model two
.inputs a b clkA clkB
.outputs y z
.names a n1
1 1
.latch n1 q1 re clkA 0
.names b n2
1 1
.latch n2 q2 re clkB 0
.names q1 y
1 1
.names q2 z
1 1
.end
"""


def test_two_clock_blif_flow(tmp_path):
    from parallel_eda_amd.__main__ import main
    from parallel_eda_amd.io.pack import pack_blif
    from parallel_eda_amd.io.blif import parse_blif
    m = parse_blif(BLIF2CLK)
    arch = get_arch("tiny")
    nl, _, _ = pack_blif(m, arch, n_ble=2)
    assert len(nl.clock_names) == 2
    assert (np.asarray(nl.block_clock) >= 0).sum() >= 2
    blif = tmp_path / "c.blif"; blif.write_text(BLIF2CLK)
    xml = tmp_path / "a.xml"; xml.write_text(ARCH_XML)
    sdc = tmp_path / "c.sdc"
    sdc.write_text("create_clock -period 8.0 -name clkA\n"
                   "create_clock -period 4.0 -name clkB\n")
    rc = main([str(blif), str(xml), "--route_chan_width", "20",
               "--sdc", str(sdc)])
    assert rc == 0


def test_bench_flow_tool(tmp_path):
    """tools/bench_flow.py (place+route wall-clock, the BASELINE metric)
    emits a valid JSON line and succeeds on the CPU path."""
    import json
    import subprocess
    import sys
    from pathlib import Path
    root = Path(__file__).resolve().parent.parent
    r = subprocess.run([sys.executable, str(root / "tools" / "bench_flow.py"),
                        "tseng", "--placer", "cpu", "--router", "cpu"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    data = json.loads(r.stdout.strip().splitlines()[-1])
    assert data["success"] and data["value"] > 0
    assert data["higher_is_better"] is False
    assert set(data["phase_s"]) == {"netlist_synth", "place", "rr_build",
                                    "route"}


def test_fixed_layout_xml():
    """<layout width height> fixes the grid; auto layouts size from the
    netlist (reference: SetupGrid.c fixed vs auto)."""
    from parallel_eda_amd.arch.xml_parser import (parse_arch_xml,
                                                  size_grid_for_netlist)
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.arch.archdef import get_arch
    a = parse_arch_xml('<architecture><layout width="20" height="10"/>'
                       '</architecture>')
    assert (a.nx, a.ny) == (20, 10)
    nl = synth_netlist(spec_for_arch(get_arch("tiny"), fill=0.5, seed=1))
    size_grid_for_netlist(nl, a)
    assert (a.nx, a.ny) == (20, 10)     # fixed layout wins
    b = parse_arch_xml('<architecture><layout auto="1.0"/></architecture>')
    size_grid_for_netlist(nl, b)
    assert b.nx >= 2                    # auto-sized


def test_cli_kitchen_sink(tmp_path):
    """Integration smoke: the widest CLI surface in one flow — synth het
    arch, settings file, pad pinning, criticality knobs, incremental
    routing, and every output artifact at once (catches option
    interaction regressions)."""
    from parallel_eda_amd.__main__ import main
    cfg = tmp_path / "s.toml"
    cfg.write_text("fill = 0.5\nmax_router_iterations = 60\n")
    out = {k: tmp_path / k for k in
           ("o.net", "o.place", "o.route", "o.v", "o.sdf", "p.rpt",
            "pl.svg", "rt.svg", "t.rpt")}
    stats = tmp_path / "stats"
    rc = main(["--synth", "tiny_het", "--seed", "5",
               "--settings", str(cfg),
               "--criticality_exp", "1.0", "--max_criticality", "0.98",
               "--route_incremental",
               "--out_net", str(out["o.net"]),
               "--out_place", str(out["o.place"]),
               "--out_route", str(out["o.route"]),
               "--out_verilog", str(out["o.v"]),
               "--out_sdf", str(out["o.sdf"]),
               "--power_report", str(out["p.rpt"]),
               "--draw_place", str(out["pl.svg"]),
               "--draw_route", str(out["rt.svg"]),
               "--timing_report", str(out["t.rpt"]),
               "--stats_dir", str(stats), "--echo_routes"])
    assert rc == 0
    for k, f in out.items():
        assert f.exists() and f.stat().st_size > 0, k
    assert (stats / "iter_stats.txt").exists()
    assert any(p.name.startswith("routes_iter_")
               for p in stats.iterdir())


def test_big_blif_end_to_end(tmp_path):
    """Scale integration for the front end: ~1.6k primitives (names,
    latches, RAM subckts, an 8-adder carry chain) through parse -> pack
    (cluster size from the arch num_pb) -> auto grid sizing -> macro-
    aware placement -> routing, via the CLI's positional path."""
    import numpy as np
    from parallel_eda_amd.__main__ import main
    rng = np.random.default_rng(99)
    n_in, n_out = 24, 12
    sigs = [f"i{k}" for k in range(n_in)]
    lines = [".model big", ".inputs " + " ".join(sigs),
             ".outputs " + " ".join(f"o{k}" for k in range(n_out))]
    for k in range(2000):
        nin = int(rng.integers(2, 5))
        pick = rng.choice(len(sigs), nin, replace=False)
        lines += [".names " + " ".join(sigs[int(i)] for i in pick)
                  + f" n{k}", "1" * nin + " 1"]
        sigs.append(f"n{k}")
    for k in range(200):
        src = sigs[int(rng.integers(len(sigs)))]
        lines.append(f".latch {src} q{k} re clk 0")
        sigs.append(f"q{k}")
    for k in range(30):
        a = sigs[int(rng.integers(len(sigs)))]
        b = sigs[int(rng.integers(len(sigs)))]
        lines.append(f".subckt single_port_ram addr0={a} data0={b} "
                     f"clk=clk out0=m{k}")
        sigs.append(f"m{k}")
    prev = None
    for k in range(8):
        a = sigs[int(rng.integers(len(sigs)))]
        cin = f"cin=c{k-1} " if prev else ""
        lines.append(f".subckt adder a0={a} {cin}sumout=as{k} cout=c{k}")
        sigs.append(f"as{k}")
        prev = f"c{k}"
    for k in range(n_out):
        lines += [f".names {sigs[int(rng.integers(len(sigs)))]} o{k}",
                  "1 1"]
    lines.append(".end")
    xml = """<architecture><layout auto="1.0"/>
 <switchlist><switch type="mux" name="0" R="551" Cin="0.77e-15"
   Tdel="58e-12"/></switchlist>
 <segmentlist><segment length="4" Rmetal="101"
   Cmetal="22.5e-15"/></segmentlist>
 <complexblocklist>
  <pb_type name="io" capacity="8"><input name="o" num_pins="1"/>
    <output name="i" num_pins="1"/></pb_type>
  <pb_type name="clb"><input name="I" num_pins="22"/>
    <output name="O" num_pins="8"/>
    <pb_type name="ble" num_pb="8"><input name="in" num_pins="6"/>
      <output name="out" num_pins="1"/></pb_type>
    <fc default_in_type="frac" default_in_val="0.15"
        default_out_type="frac" default_out_val="0.1"/></pb_type>
  <pb_type name="memory"><input name="addr" num_pins="16"/>
    <output name="out" num_pins="8"/>
    <gridlocations><loc type="col" start="2" repeat="6"
      priority="2"/></gridlocations></pb_type>
  <pb_type name="mult_36"><input name="a" num_pins="16"/>
    <output name="out" num_pins="8"/>
    <gridlocations><loc type="col" start="4" repeat="9"
      priority="2"/></gridlocations></pb_type>
 </complexblocklist></architecture>"""
    bf = tmp_path / "big.blif"
    xf = tmp_path / "arch.xml"
    bf.write_text("\n".join(lines))
    xf.write_text(xml)
    rc = main([str(bf), str(xf), "--route_chan_width", "80",
               "--max_router_iterations", "80"])
    assert rc == 0


def test_arch_xml_segment_distribution():
    """Multiple <segment> lengths map onto the two-length fabric: the
    longest becomes L, the length-1 frequency share becomes w_l1
    (reference: <segmentlist> distributions in VTR arches)."""
    from parallel_eda_amd.arch.xml_parser import parse_arch_xml
    xml = """<architecture>
 <complexblocklist>
  <pb_type name="io" capacity="2">
   <input name="outpad" num_pins="1"/><output name="inpad" num_pins="1"/>
  </pb_type>
  <pb_type name="clb">
   <input name="I" num_pins="16"/><output name="O" num_pins="4"/>
  </pb_type>
 </complexblocklist>
 <segmentlist>
  <segment length="1" freq="0.25" Rmetal="50" Cmetal="1e-14"/>
  <segment length="4" freq="0.75" Rmetal="101" Cmetal="2.25e-14"/>
 </segmentlist>
 <switchlist>
  <switch type="mux" name="0" R="551" Cin="7.7e-16" Tdel="5.8e-11"/>
 </switchlist>
</architecture>"""
    a = parse_arch_xml(xml, nx=20, ny=20, W=40)
    assert a.L == 4
    assert a.w_l1 == 10   # 25% of W=40, even
    assert a.R_wire == 101.0   # timing from the workhorse (longest) segment
    # single-segment arch keeps auto behavior
    a2 = parse_arch_xml(xml.replace(
        '<segment length="1" freq="0.25" Rmetal="50" Cmetal="1e-14"/>', ""),
        nx=20, ny=20, W=40)
    assert a2.L == 4 and a2.w_l1 == -1


def test_route_file_roundtrip_and_analysis(tmp_path):
    """Write a routing, read it back (--route_file analysis flow), and
    re-derive identical structure + Elmore delays (reference: VPR's
    ROUTE_NEVER read-route-and-analyze path)."""
    import numpy as np
    from parallel_eda_amd.__main__ import main

    blif = tmp_path / "c.blif"
    blif.write_text(BLIF)
    xml = tmp_path / "a.xml"
    xml.write_text(ARCH_XML)
    rfile = tmp_path / "c.route"
    pfile = tmp_path / "c.place"
    rc = main([str(blif), str(xml), "--route_chan_width", "20",
               "--out_route", str(rfile), "--out_place", str(pfile)])
    assert rc == 0 and rfile.exists()
    # analysis-only flow over the emitted files
    rc2 = main([str(blif), str(xml), "--route_chan_width", "20",
                "--place_file", str(pfile), "--route_file", str(rfile),
                "--timing_report", str(tmp_path / "t.rpt")])
    assert rc2 == 0
    assert (tmp_path / "t.rpt").exists()
    # a corrupted traceback is rejected
    txt = rfile.read_text().splitlines()
    for i, ln in enumerate(txt):
        if ln.startswith("IPIN"):
            del txt[i:i + 2]   # drop an IPIN+SINK pair -> missing sink
            break
    bad = tmp_path / "bad.route"
    bad.write_text("\n".join(txt) + "\n")
    rc3 = main([str(blif), str(xml), "--route_chan_width", "20",
                "--place_file", str(pfile), "--route_file", str(bad)])
    assert rc3 == 1


def test_route_file_delay_fidelity():
    """read_route + tree_elmore_delays reproduce the router's own
    per-sink delays exactly (same per-hop Elmore model)."""
    import numpy as np
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_placed_netlist
    from parallel_eda_amd import rrgraph
    from parallel_eda_amd.route.router import (pathfinder_route,
                                               net_rr_terminals)
    from parallel_eda_amd.timing.sta import STA
    from parallel_eda_amd.io.route_file import (write_route, read_route,
                                                tree_elmore_delays)
    arch = get_arch("tseng")
    nl, pl = synth_placed_netlist(arch, fill=0.4, seed=21)
    g = rrgraph.build_rr_graph(arch)
    res = pathfinder_route(nl, pl, g, arch, sta=STA(nl, arch), max_iters=40)
    assert res.success
    net_ids, src_rr, sink_ptr, sink_rr, _ = net_rr_terminals(nl, pl, g, arch)
    import tempfile, os
    path = os.path.join(tempfile.mkdtemp(), "t.route")
    write_route(path, g, arch, net_ids, lambda k: res.router.tree(k),
                netlist=nl)
    names, trees = read_route(path, g, arch)
    assert len(trees) == len(net_ids)
    sd_router = np.asarray(res.router.sink_delays())
    for k, (nodes, parents) in enumerate(trees):
        rn, rp, rs, rd = res.router.tree(k)
        assert set(nodes.tolist()) == set(np.asarray(rn).tolist()), k
        d = tree_elmore_delays(g, nodes, parents)
        pos = {int(v): i for i, v in enumerate(nodes)}
        for s in range(sink_ptr[k], sink_ptr[k + 1]):
            v = int(sink_rr[s])
            assert abs(d[pos[v]] - sd_router[s]) < 1e-12, (k, s)


def test_cli_net_input(tmp_path):
    """A packed .net is accepted directly as the circuit (reference: VPR
    reads circuit.net and skips packing)."""
    from parallel_eda_amd.__main__ import main
    blif = tmp_path / "c.blif"
    blif.write_text(BLIF)
    xml = tmp_path / "a.xml"
    xml.write_text(ARCH_XML)
    netf = tmp_path / "c.net"
    rc = main([str(blif), str(xml), "--route_chan_width", "20",
               "--out_net", str(netf), "--place_only"])
    assert rc == 0 and netf.exists()
    rc2 = main([str(netf), str(xml), "--route_chan_width", "20"])
    assert rc2 == 0
