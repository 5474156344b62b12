"""Placement driver: adaptive-schedule simulated annealing.

The outer temperature loop lives here (shared by the CPU oracle and the GPU
engine); per-temperature move batches run in the engine. Schedule semantics
follow the reference placer (vpr/SRC/place/place.c:310 try_place;
starting_t:1045; update_t:983; update_rlim:969; exit_crit:1023).
"""
from dataclasses import dataclass, field

import numpy as np

from ..arch.archdef import ArchDef
from .. import ops


@dataclass
class Placement:
    x: np.ndarray
    y: np.ndarray
    slot: np.ndarray
    bb_cost: float = 0.0
    td_cost: float = 0.0
    stats: dict = field(default_factory=dict)


def analytic_delay_matrix(arch: ArchDef) -> np.ndarray:
    """Delay lookup delta_delay[|dx|, |dy|], shape (nx+2, ny+2).

    The reference builds this by routing dummy nets at every (dx,dy)
    (timing_place_lookup.c:981 compute_delay_lookup_tables); we start from
    the same per-hop Elmore model the router charges, which for our
    all-buffered fabric is exact per segment. (A router-measured matrix is
    a planned refinement on GPU — batched wavefront routes.)
    """
    nx, ny = arch.nx, arch.ny
    seg_delay = arch.T_sw + arch.C_wire * arch.L * (arch.R_sw + 0.5 * arch.R_wire * arch.L)
    dx = np.arange(nx + 2)[:, None]
    dy = np.arange(ny + 2)[None, :]
    dist = dx + dy
    nseg = np.ceil(dist / arch.L)
    d = arch.T_opin + nseg * seg_delay + arch.T_ipin
    d[0, 0] = arch.T_opin + arch.T_ipin  # same tile
    return d.astype(np.float32)


def anneal_place(netlist, arch: ArchDef, seed: int = 7, timing_tradeoff: float = 0.5,
                 inner_num: float = 1.0, sta=None, crit_exp: float = 1.0,
                 verbose: bool = False, engine: str = "cpu",
                 delay_matrix: str = "analytic", fixed=None,
                 macros=None) -> Placement:
    """Run the full SA schedule; returns final Placement.

    sta: optional TimingGraph wrapper (timing.sta.STA) for criticality
    refresh each temperature; None => pure bounding-box placement.
    fixed: optional (ids, x, y, slot) arrays pinning blocks to locations
    (reference: -pad_loc_file / fix_pins); pinned blocks never move.
    macros: optional list of [(blk, dx, dy), ...] groups placed and moved
    atomically at fixed relative offsets (reference: place_macro.c carry
    chains); the first member is the head at (0, 0). CPU engine only.
    """
    if engine == "gpu":
        from .gpu_placer import anneal_place_gpu
        return anneal_place_gpu(netlist, arch, seed=seed,
                                timing_tradeoff=timing_tradeoff,
                                inner_num=inner_num, sta=sta,
                                crit_exp=crit_exp, verbose=verbose,
                                fixed=fixed, delay_matrix=delay_matrix)
    cpu = ops.cpu()
    if sta is not None and timing_tradeoff > 0:
        if delay_matrix == "routed":
            from .delay_matrix import routed_delay_matrix
            dm = routed_delay_matrix(arch)
        else:
            dm = analytic_delay_matrix(arch)
    else:
        dm = np.zeros(0, dtype=np.float32)
    tb = (arch.tile_btype_grid() if arch.is_heterogeneous()
          else np.empty(0, dtype=np.int8))
    placer = cpu.SerialPlacer(netlist.cpp(), arch.nx, arch.ny, arch.io_cap,
                              np.ascontiguousarray(dm.ravel()), seed, tb)
    if macros:
        if engine != "cpu":
            raise NotImplementedError("macros: CPU engine only (round 2: GPU)")
        ptr = [0]
        mb, mdx, mdy = [], [], []
        for grp in macros:
            for (b, dx, dy) in grp:
                mb.append(b); mdx.append(dx); mdy.append(dy)
            ptr.append(len(mb))
        placer.set_macros(np.asarray(ptr, dtype=np.int64),
                          np.asarray(mb, dtype=np.int32),
                          np.asarray(mdx, dtype=np.int32),
                          np.asarray(mdy, dtype=np.int32))
    if fixed is not None:
        ids, fx, fy, fs = fixed
        placer.fix_blocks(np.asarray(ids, dtype=np.int32),
                          np.asarray(fx, dtype=np.int32),
                          np.asarray(fy, dtype=np.int32),
                          np.asarray(fs, dtype=np.int32))
    nb = netlist.num_blocks
    move_lim = max(64, int(inner_num * (nb ** 1.3333)))
    rlim = float(max(arch.nx, arch.ny))
    tt = timing_tradeoff if sta is not None else 0.0

    crit = np.zeros(netlist.num_conns, dtype=np.float32)

    def refresh_crit():
        nonlocal crit
        if sta is None or tt <= 0:
            return
        delays = placer.conn_delays()
        cpd, slack, c = sta.analyze(delays)
        crit = (np.asarray(c) ** crit_exp).astype(np.float32)
        placer.set_crit(crit)

    refresh_crit()
    # In timing mode, normalize both cost terms by their value at the last
    # refresh (combined cost ~ 1, VPR-style); in bb-only mode use raw deltas.
    norm_mode = tt > 0

    def norms():
        if norm_mode:
            return (max(placer.bb_cost(), 1e-12), max(placer.td_cost(), 1e-30))
        return (1.0, 1.0)

    bb_norm, td_norm = norms()
    # starting temperature = 20 * std(move deltas) (place.c:1045 starting_t)
    n_trial = max(64, nb)
    placer.run_moves(1e30, rlim, n_trial, tt, bb_norm, td_norm)
    t = 20.0 * placer.last_delta_std()
    if t <= 0:
        t = 1.0

    history = []
    itemp = 0
    while True:
        refresh_crit()
        bb_norm, td_norm = norms()
        srate = placer.run_moves(t, rlim, move_lim, tt, bb_norm, td_norm)
        cost = placer.bb_cost()
        history.append((t, cost, srate, rlim))
        if verbose:
            print(f"T={t:.3e} bb={cost:.1f} td={placer.td_cost():.3e} "
                  f"acc={srate:.2f} rlim={rlim:.1f}")
        # update_t (place.c:983)
        if srate > 0.96:
            t *= 0.5
        elif srate > 0.8:
            t *= 0.9
        elif srate > 0.15 or rlim > 1:
            t *= 0.95
        else:
            t *= 0.8
        # update_rlim (place.c:969)
        rlim = rlim * (1.0 - 0.44 + srate)
        rlim = min(max(rlim, 1.0), float(max(arch.nx, arch.ny)))
        itemp += 1
        # exit criterion (place.c:1023): t below 0.5% of per-net cost
        exit_cost = 1.0 if norm_mode else cost
        if t < 0.005 * exit_cost / max(1, netlist.num_nets):
            break
        if itemp > 500:
            break
    # final quench at T=0
    placer.run_moves(0.0, 1.0, move_lim, tt, bb_norm, td_norm)
    ok, err = placer.check_place()
    if not ok:
        raise RuntimeError(f"check_place failed: {err}")
    x, y, s = placer.placement()
    return Placement(np.asarray(x), np.asarray(y), np.asarray(s),
                     bb_cost=placer.bb_cost(), td_cost=placer.td_cost(),
                     stats={"temps": itemp, "move_lim": move_lim,
                            "history": history})
