"""Golden real-format flow test.

A hand-authored VTR-style arch.xml + BLIF pair runs through the full CLI
(pack -> place -> route) and the emitted .net / .place / .route files are
byte-compared against committed goldens. Any behavioral change to the
packer, placer, router, rr-graph builder, or writers shows up here and
must be accompanied by a deliberate tools/regen_goldens.py run.

Reference formats: vpr/SRC/base/read_netlist.c (.net),
read_place.c / print_place (.place), route_common.c:1322 print_route
(.route)."""
import filecmp
from pathlib import Path

import pytest

G = Path(__file__).parent / "golden"


def _run_flow(tmp_path):
    from parallel_eda_amd.__main__ import main
    tmp_path.mkdir(parents=True, exist_ok=True)
    outs = {k: tmp_path / f"out.{k}" for k in ("net", "place", "route")}
    rc = main([str(G / "golden.blif"), str(G / "golden_arch.xml"),
               "--route_chan_width", "12", "--seed", "3",
               "--timing_tradeoff", "0.5",
               "--out_net", str(outs["net"]),
               "--out_place", str(outs["place"]),
               "--out_route", str(outs["route"])])
    assert rc == 0
    return outs


def test_golden_flow_outputs(tmp_path):
    outs = _run_flow(tmp_path)
    for kind, path in outs.items():
        golden = G / f"golden.{kind}"
        assert golden.exists(), f"missing golden {golden}"
        if not filecmp.cmp(path, golden, shallow=False):
            import difflib
            a = golden.read_text().splitlines()
            b = path.read_text().splitlines()
            diff = "\n".join(list(difflib.unified_diff(
                a, b, str(golden), str(path), lineterm=""))[:40])
            pytest.fail(f".{kind} output differs from golden "
                        f"(regen with tools/regen_goldens.py if the change "
                        f"is deliberate):\n{diff}")


def test_golden_flow_deterministic(tmp_path):
    """Two runs produce byte-identical outputs (serial flow invariant)."""
    o1 = _run_flow(tmp_path / "a")
    o2 = _run_flow(tmp_path / "b")
    for kind in o1:
        assert filecmp.cmp(o1[kind], o2[kind], shallow=False), kind


def test_packer_single_clock_legality():
    """A 2-clock BLIF must never mix clocks inside one cluster
    (reference: cluster_legality.c clock feasibility)."""
    from parallel_eda_amd.io.blif import parse_blif
    from parallel_eda_amd.io.pack import pack_blif
    from parallel_eda_amd.arch.archdef import get_arch
    blif = """.model two_clk
.inputs c1 c2 a b
.outputs x y
.names a b d1\n11 1
.names a b d2\n01 1
.latch d1 x re c1 0
.latch d2 y re c2 0
.end
"""
    m = parse_blif(blif)
    arch = get_arch("tseng")
    nl, cluster_of, names = pack_blif(m, arch, n_ble=arch.clb_n_ble)
    # find each latch's cluster: latches drive x and y
    kinds = {}
    for i, p in enumerate(m.prims):
        if p.kind == "latch":
            kinds[p.clock] = cluster_of[i]
    assert kinds["c1"] != kinds["c2"], \
        "latches on different clocks packed into one cluster"


def test_packer_output_pin_legality():
    """Cluster external outputs never exceed clb_out."""
    import numpy as np
    from parallel_eda_amd.io.blif import parse_blif
    from parallel_eda_amd.io.pack import pack_blif
    from parallel_eda_amd.arch.archdef import ArchDef
    # 8 independent 1-LUT functions, all model outputs; clb_out=2 forces
    # at least 4 clusters
    lines = [".model outs", ".inputs a b", ".outputs " +
             " ".join(f"o{i}" for i in range(8))]
    for i in range(8):
        lines += [f".names a b o{i}", "11 1" if i % 2 else "01 1"]
    lines.append(".end")
    m = parse_blif("\n".join(lines))
    arch = ArchDef(name="t", nx=6, ny=6, W=20, L=1, clb_in=16, clb_out=2,
                   clb_n_ble=8, io_cap=4)
    nl, cluster_of, names = pack_blif(m, arch, n_ble=8)
    counts = {}
    for i, p in enumerate(m.prims):
        counts.setdefault(cluster_of[i], 0)
        counts[cluster_of[i]] += 1
    assert max(counts.values()) <= 2, counts
    assert len(counts) >= 4
