"""Property/fuzz tests: random small fabrics and netlists through the
validators and the full CPU flow (hypothesis-free, seeded-deterministic)."""
import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import ArchDef
from parallel_eda_amd.io.synth import synth_netlist, SynthSpec
from parallel_eda_amd.place.placer import anneal_place
from parallel_eda_amd.route.router import pathfinder_route
from parallel_eda_amd.timing.sta import STA
from parallel_eda_amd import rrgraph


def random_arch(rng):
    # realistic W/L: at least 2 tracks per (direction, stagger) class,
    # else the Fs=3 switch digraph can be reducible (validator rejects)
    L = int(rng.integers(1, 5))
    W = 2 * int(rng.integers(max(4, 2 * L), 17))
    return ArchDef(
        name=f"fuzz", nx=int(rng.integers(3, 9)), ny=int(rng.integers(3, 9)),
        W=W, L=L,
        fc_in=int(rng.integers(2, min(9, W + 1))),
        fc_out=int(rng.integers(2, min(9, W + 1))),
        clb_in=int(rng.integers(4, 12)), clb_out=int(rng.integers(1, 4)),
        io_cap=int(rng.integers(1, 4)))


@pytest.mark.parametrize("seed", [11, 23, 37, 51, 68])
def test_random_fabric_validates(seed):
    rng = np.random.default_rng(seed)
    arch = random_arch(rng)
    g = rrgraph.build_rr_graph(arch)
    assert rrgraph.check_rr_graph(g, arch), arch
    # degree bounded (SB Fs=3 in + fc_in taps + pins)
    assert g.degree_max <= max(arch.clb_in, arch.clb_out,
                               arch.io_cap) + arch.fc_out + 8


@pytest.mark.parametrize("seed", [5, 29])
def test_random_flow_end_to_end(seed):
    rng = np.random.default_rng(seed)
    arch = random_arch(rng)
    # keep the netlist small enough to be routable at the random W
    n_clb = max(4, int(arch.nx * arch.ny * 0.4))
    nl = synth_netlist(SynthSpec(
        n_clb=n_clb, n_in=2, n_out=2, avg_fanout=2.0,
        max_fanin=arch.clb_in, seed=seed))
    if int((nl.block_type == 0).sum()) > arch.num_io_slots():
        pytest.skip("io overflow for this random arch")
    g = rrgraph.build_rr_graph(arch)
    try:
        rrgraph.check_rr_graph(g, arch)
    except rrgraph.RRGraphError as e:
        pytest.skip(f"fuzzed fabric rejected by validator: {e}")
    pl = anneal_place(nl, arch, seed=seed, timing_tradeoff=0.0)
    sta = STA(nl, arch)
    res = pathfinder_route(nl, pl, g, arch, sta=sta, max_iters=80)
    if not res.success:
        # a narrow random channel may be genuinely unroutable; then the
        # flow must still terminate cleanly with overuse reported
        assert res.overused > 0
    else:
        assert res.crit_path_delay > 0
        ok, err = res.router.check_routed()
        assert ok, err
