import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, SynthSpec, NetlistPy
from parallel_eda_amd.timing.sta import STA


def test_synth_netlist_basic():
    nl = synth_netlist(SynthSpec(n_clb=50, n_in=5, n_out=5, seed=3))
    assert nl.num_blocks == 60
    assert nl.num_nets == 55  # every input pad + every CLB drives one net
    # all sinks valid block ids, no self-loop driver==sink
    for n in range(nl.num_nets):
        s = nl.net_sinks[nl.net_sink_ptr[n]:nl.net_sink_ptr[n + 1]]
        assert len(s) >= 1
        assert (s != nl.net_driver[n]).all()


def test_sta_levelizes_and_analyzes():
    arch = get_arch("tiny")
    nl = synth_netlist(SynthSpec(n_clb=40, n_in=4, n_out=4, seed=5))
    sta = STA(nl, arch)
    assert sta.num_levels >= 1
    d = np.full(nl.num_conns, 1e-9, dtype=np.float32)
    cpd, slack, crit = sta.analyze(d)
    assert cpd > 0
    assert crit.min() >= 0.0 and crit.max() <= 1.0
    # some connection must be fully critical
    assert crit.max() > 0.99
    # zero delays => smaller cpd
    cpd0, _, _ = sta.analyze(np.zeros_like(d))
    assert cpd0 < cpd


def test_sta_hand_case():
    # in_pad(0) -> comb A(2) -> comb B(3) -> out_pad(1); delays 1ns each conn
    block_type = [0, 0, 1, 1]
    block_is_seq = [1, 1, 0, 0]
    # nets: 0: pad0 -> A ; 1: A -> B ; 2: B -> pad1
    driver = [0, 2, 3]
    sptr = [0, 1, 2, 3]
    sinks = [2, 3, 1]
    nl = NetlistPy(block_type, block_is_seq, driver, sptr, sinks)
    arch = get_arch("tiny")
    sta = STA(nl, arch)
    d = np.full(3, 1e-9, dtype=np.float32)
    cpd, slack, crit = sta.analyze(d)
    # cpd = T_seq_out + 3 conn delays + 2 comb delays + T_seq_in
    expect = arch.T_seq_out + 3e-9 + 2 * arch.T_clb + arch.T_seq_in
    assert cpd == pytest.approx(expect, rel=1e-5)
    # single path: every connection critical, slack ~ 0
    assert np.allclose(crit, 1.0, atol=1e-5)
    assert np.allclose(slack, 0.0, atol=1e-12)


def test_timing_report_and_sdc(tmp_path):
    from parallel_eda_amd.timing.report import (write_timing_report,
                                                critical_paths, parse_sdc)
    arch = get_arch("tiny")
    nl = synth_netlist(SynthSpec(n_clb=40, n_in=4, n_out=4, seed=5))
    nl.names = [f"b{i}" for i in range(nl.num_blocks)]
    sta = STA(nl, arch)
    rng = np.random.default_rng(1)
    d = (rng.random(nl.num_conns) * 1e-9).astype(np.float32)
    cpd, paths = critical_paths(nl, sta, d, k=3)
    assert cpd > 0 and len(paths) >= 1
    # most critical path's endpoint slack ~ 0
    assert abs(paths[0]["endpoint_slack"]) < cpd * 0.05
    p = tmp_path / "timing.rpt"
    write_timing_report(p, nl, sta, d)
    text = p.read_text()
    assert "Critical path delay" in text and "Path 0" in text
    assert parse_sdc("create_clock -period 5.0 -name clk [get_ports clk]\n"
                     ) == pytest.approx(5e-9)
    assert parse_sdc("# nothing\n") is None


def test_multi_clock_domains():
    """Two-domain hand case: in0(clkA) -> A(comb) -> ff1(clkA),
    in1(clkB) -> B(comb) -> ff2(clkB), cross path A -> ff2."""
    from parallel_eda_amd.io.synth import NetlistPy
    arch = get_arch("tiny")
    # blocks: 0=in0(A) 1=in1(B) 2=A comb 3=B comb 4=ff1(A) 5=ff2(B)
    block_type = [0, 0, 1, 1, 1, 1]
    block_is_seq = [1, 1, 0, 0, 1, 1]
    # nets: 0: in0->A; 1: in1->B; 2: A->{ff1, ff2}; 3: B->ff2
    driver = [0, 1, 2, 3]
    sptr = [0, 1, 2, 4, 5]
    sinks = [2, 3, 4, 5, 5]
    nl = NetlistPy(block_type, block_is_seq, driver, sptr, sinks)
    sta = STA(nl, arch)
    block_clock = np.array([0, 1, -1, -1, 0, 1], dtype=np.int32)
    periods = np.array([5e-9, 2e-9], dtype=np.float32)
    d = np.full(5, 1e-9, dtype=np.float32)
    wp, slack, crit = sta.analyze_domains(d, block_clock, periods)
    # path in0->A->ff1 (A-domain): arrival = Tout + 1n + Tclb + 1n
    arrA = arch.T_seq_out + 1e-9 + arch.T_clb + 1e-9
    # conn 2 (A->ff1, clkA): slack = 5n - Tsu - arrA
    assert slack[2] == pytest.approx(5e-9 - arch.T_seq_in - arrA, rel=1e-5)
    # conn 3 (A->ff2, clkB): tighter 2n constraint
    assert slack[3] == pytest.approx(2e-9 - arch.T_seq_in - arrA, rel=1e-5)
    # crit of the cross-domain conn must exceed the same-domain one
    assert crit[3] > crit[2]
    # worst achieved period comes from the tight B constraint paths
    assert wp == pytest.approx(arrA + arch.T_seq_in, rel=1e-5)


def test_sta_random_dag_property():
    """Property test: on random acyclic netlists the STA cpd must equal
    an independent longest-path computation over the same delay model,
    and slack of every connection on a critical path must be ~0."""
    import numpy as np
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_netlist, SynthSpec
    from parallel_eda_amd.timing.sta import STA

    arch = get_arch("tiny")
    for seed in (3, 17, 42):
        nl = synth_netlist(SynthSpec(n_clb=60, n_in=4, n_out=4,
                                     avg_fanout=3.0, max_fanin=6,
                                     seed=seed))
        sta = STA(nl, arch)
        rng = np.random.default_rng(seed)
        delay = rng.uniform(0.1e-9, 2e-9,
                            nl.num_conns).astype(np.float32)
        cpd, slack, crit = sta.analyze(delay)

        # independent longest-path: arrival at block outputs
        nb = nl.num_blocks
        seq = np.asarray(nl.block_is_seq).astype(bool)
        # conn lists
        conns = []
        for n in range(nl.num_nets):
            drv = int(nl.net_driver[n])
            for c in range(int(nl.net_sink_ptr[n]), int(nl.net_sink_ptr[n + 1])):
                conns.append((drv, int(nl.net_sinks[c]), float(delay[c])))
        arr = {}

        def arrival(b, depth=0):
            if b in arr:
                return arr[b]
            assert depth < nb + 1
            if seq[b]:
                arr[b] = arch.T_seq_out
                return arr[b]
            best = 0.0
            for (d, s, dl) in conns:
                if s == b:
                    best = max(best, arrival(d, depth + 1) + dl)
            arr[b] = best + arch.T_clb
            return arr[b]

        ref_cpd = 0.0
        for b in range(nb):
            if not seq[b]:
                continue
            a = 0.0
            for (d, s, dl) in conns:
                if s == b:
                    a = max(a, arrival(d) + dl)
            ref_cpd = max(ref_cpd, a + arch.T_seq_in)
        assert cpd == pytest.approx(ref_cpd, rel=1e-5), seed
        # max criticality must be ~1 and its conn slack ~0
        assert crit.max() == pytest.approx(1.0, abs=1e-4)
        assert abs(slack[np.argmax(crit)]) <= 1e-12 + 1e-4 * cpd


def test_sdc_pair_constraints():
    """set_false_path and set_multicycle_path between clock domains
    (reference read_sdc.c): a false-path pair is not analyzed; a
    multicycle pair's setup constraint is N periods."""
    import numpy as np
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.timing.sta import STA
    from parallel_eda_amd.timing.report import (parse_sdc_constraints,
                                                pair_constraints)
    sdc = parse_sdc_constraints("""
create_clock -period 5.0 -name clkA
create_clock -period 8.0 -name clkB
set_false_path -from [get_clocks clkA] -to [get_clocks clkB]
set_multicycle_path 2 -from [get_clocks clkB] -to [get_clocks clkA]
set_input_delay -clock clkA 1.0
""")
    assert sdc["clocks"] == {"clkA": 5e-9, "clkB": 8e-9}
    assert sdc["false_paths"] == [("clkA", "clkB")]
    assert sdc["multicycle"] == [("clkB", "clkA", 2)]
    assert abs(sdc["input_delay"]["clkA"] - 1e-9) < 1e-15
    skip, mult = pair_constraints(sdc, ["clkA", "clkB"])
    assert skip[0, 1] == 1 and skip.sum() == 1
    assert mult[1, 0] == 2.0 and mult.sum() == 5.0

    arch = get_arch("tseng")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=3))
    rng = np.random.default_rng(3)
    bc = np.where(np.asarray(nl.block_is_seq) > 0,
                  rng.integers(0, 2, nl.num_blocks), -1).astype(np.int32)
    periods = np.asarray([5e-9, 8e-9], dtype=np.float32)
    dly = rng.uniform(0.1e-9, 2e-9, nl.num_conns).astype(np.float32)
    sta = STA(nl, arch)
    wp0, sl0, cr0 = sta.analyze_domains(dly, bc, periods)
    # all pairs false -> nothing analyzed: zero slack everywhere
    all_skip = np.ones((2, 2), dtype=np.uint8)
    wpf, slf, crf = sta.analyze_domains(dly, bc, periods,
                                        pair_skip=all_skip)
    assert np.all(crf == 0.0)
    # multicycle 2 on every pair: slacks can only grow
    m2 = np.full((2, 2), 2.0, dtype=np.float32)
    wpm, slm, crm = sta.analyze_domains(dly, bc, periods, pair_mult=m2)
    assert np.all(slm >= sl0 - 1e-12)
    assert wpm <= wp0 + 1e-12 or crm.max() <= cr0.max() + 1e-6
    # skipping one pair never worsens slack on still-constrained conns
    # (conns whose only pair was skipped become unconstrained -> slack 0)
    wps, sls, crs = sta.analyze_domains(dly, bc, periods, pair_skip=skip)
    constrained = sls != 0.0
    assert np.all(sls[constrained] >= sl0[constrained] - 1e-12)
    assert crs.max() <= cr0.max() + 1e-6
