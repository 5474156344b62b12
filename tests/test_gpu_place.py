"""GPU SA placer tests (MI355X)."""
import numpy as np
import pytest

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
from parallel_eda_amd.timing.sta import STA

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def tseng_case():
    arch = get_arch("tseng")
    nl = synth_netlist(spec_for_arch(arch, fill=0.5, seed=4))
    return arch, nl


def test_gpu_refresh_matches_cpu_cost(tseng_case):
    """GPU net-cost evaluation == CPU oracle on the same placement."""
    from parallel_eda_amd.place.gpu_placer import GpuPlacer
    from parallel_eda_amd import ops
    arch, nl = tseng_case
    gp = GpuPlacer(nl, arch, seed=11)
    bb_gpu, _ = gp.refresh_costs()
    # replicate the same placement in the CPU oracle and compare
    cpu = ops.cpu()
    sp = cpu.SerialPlacer(nl.cpp(), arch.nx, arch.ny, arch.io_cap,
                          np.zeros(0, dtype=np.float32), 1)
    pl = gp.placement()
    sp.set_placement(pl.x.astype(np.int32), pl.y.astype(np.int32),
                     pl.slot.astype(np.int32))
    assert bb_gpu == pytest.approx(sp.bb_cost(), rel=1e-4)


def test_gpu_anneal_improves_and_legal(tseng_case):
    from parallel_eda_amd.place.gpu_placer import GpuPlacer, anneal_place_gpu
    arch, nl = tseng_case
    init = GpuPlacer(nl, arch, seed=7).bb_cost
    pl = anneal_place_gpu(nl, arch, seed=7, timing_tradeoff=0.0)
    assert pl.bb_cost < init * 0.8, f"{pl.bb_cost} vs init {init}"


def test_gpu_anneal_quality_vs_cpu(tseng_case):
    """GPU anneal reaches within 15% of the CPU oracle's HPWL."""
    from parallel_eda_amd.place.gpu_placer import anneal_place_gpu
    from parallel_eda_amd.place.placer import anneal_place
    arch, nl = tseng_case
    pl_cpu = anneal_place(nl, arch, seed=7, timing_tradeoff=0.0)
    pl_gpu = anneal_place_gpu(nl, arch, seed=7, timing_tradeoff=0.0)
    hist = pl_gpu.stats["history"]
    assert pl_gpu.bb_cost <= pl_cpu.bb_cost * 1.15, (
        f"GPU bb {pl_gpu.bb_cost:.1f} vs CPU {pl_cpu.bb_cost:.1f}; "
        f"temps={pl_gpu.stats['temps']} hist[:4]={hist[:4]} "
        f"hist[-4:]={hist[-4:]}")


def test_gpu_timing_anneal(tseng_case):
    from parallel_eda_amd.place.gpu_placer import anneal_place_gpu
    arch, nl = tseng_case
    sta = STA(nl, arch)
    pl = anneal_place_gpu(nl, arch, seed=7, timing_tradeoff=0.5, sta=sta)
    assert pl.td_cost > 0


def test_gpu_macro_moves(tseng_case):
    """Carry-chain macros move rigidly through the batched GPU anneal
    (reference: place_macro.c + try_swap macro branch). Offsets are
    preserved at every temperature and the final placement is legal."""
    import numpy as np
    from parallel_eda_amd.place.gpu_placer import GpuPlacer, anneal_place_gpu
    arch, nl = tseng_case
    clbs = np.nonzero(np.asarray(nl.block_type) == 1)[0]
    assert len(clbs) >= 7
    # two vertical chains of 3 and 4 CLBs
    macros = [[(int(clbs[0]), 0, 0), (int(clbs[1]), 0, 1),
               (int(clbs[2]), 0, 2)],
              [(int(clbs[3]), 0, 0), (int(clbs[4]), 0, 1),
               (int(clbs[5]), 0, 2), (int(clbs[6]), 0, 3)]]
    pl = anneal_place_gpu(nl, arch, seed=11, timing_tradeoff=0.0,
                          inner_num=0.5, macros=macros)
    bx = np.asarray(pl.x); by = np.asarray(pl.y)
    for grp in macros:
        b0, dx0, dy0 = grp[0]
        for (b, dx, dy) in grp[1:]:
            assert bx[b] - bx[b0] == dx - dx0
            assert by[b] - by[b0] == dy - dy0
    # the chains must have actually moved from the deterministic
    # first-fit legalization anchor (1,1): bb cost improved over init
    placer = GpuPlacer(nl, arch, seed=11, macros=macros)
    init_bb = placer.bb_cost
    assert pl.bb_cost < init_bb
