"""Multi-GPU / multi-process simulated-annealing placement.

SURVEY §7 step 6's second half: the SA placer reuses the router's
decomposition — the grid is cut into per-rank COLUMN STRIPS, each rank
anneals only blocks inside its strip (moves confined to the strip, so
concurrent ranks touch disjoint grid cells), and placements fuse with
one masked all-reduce per temperature. Strip boundaries alternate
between two offsets so blocks migrate across cuts over temperatures
(the reference's MPI router rotates work the same way; the serial
placer semantics transfer because within a strip this IS the serial
placer).

Deterministic: the shared seed gives every rank the identical initial
placement, per-rank move streams only touch owned blocks, the masked
sum reconstructs the identical merged placement everywhere, and the
adaptive schedule consumes the all-reduced global acceptance rate.

Engines: the CPU oracle (SerialPlacer.set_move_region, gloo tests) and
the GPU batched-move kernels (_GpuPlacerAdapter, one GPU per rank over
RCCL) — the fusion and schedule logic is engine-independent.
"""
import numpy as np


def _strips(nx, ws, phase):
    """Partition columns 0..nx+1 into ws contiguous strips; phase 1
    shifts the cuts by half a strip (first/last strips shorter)."""
    total = nx + 2
    w = max(1, total // ws)
    cuts = [0]
    off = (w // 2) if phase else 0
    for r in range(1, ws):
        cuts.append(min(total - 1, r * w + off))
    cuts.append(total)
    return [(cuts[r], cuts[r + 1] - 1) for r in range(ws)]


class _GpuPlacerAdapter:
    """Exposes the SerialPlacer driver API on top of place.gpu_placer.
    GpuPlacer so anneal_place_dist runs the SAME fusion/schedule logic
    with the CDNA4 batched-move kernels (BASELINE config 4: grid-
    partitioned SA on 8 GPUs over RCCL)."""

    def __init__(self, netlist, arch, seed, timing, device, macros=None):
        from ..place.gpu_placer import GpuPlacer
        self.p = GpuPlacer(netlist, arch, seed=seed, timing=timing,
                           device=device, macros=macros)
        self.nl = netlist
        self.arch = arch
        self._delta_std = 0.0

    def reseed(self, s):
        self.p.seed = int(s) & 0xFFFFFFFF

    def set_move_region(self, lo, hi):
        self.p.set_move_region(lo, hi)

    def run_moves(self, T, rlim, n, tt, bb_norm, td_norm):
        srate, att = self.p.run_batches(T, rlim, n, tt, bb_norm, td_norm)
        c = self.p.t_counters.cpu().numpy()
        self._att, self._acc = int(c[0]), int(c[1])
        flags = self.p.t_mv_flags
        sel = flags >= 1
        if int(sel.sum().item()) >= 8:
            d = (1.0 - tt) * self.p.t_mv_dbb[sel] / bb_norm
            if tt > 0:
                d = d + tt * self.p.t_mv_dtd[sel] / td_norm
            self._delta_std = float(d.std().item())
        self.p.refresh_costs()
        return srate

    def last_delta_std(self):
        return self._delta_std

    def last_accepts(self):
        return self._acc

    def last_valid_attempts(self):
        return self._att

    def placement(self):
        return (self.p.t_bx.cpu().numpy(), self.p.t_by.cpu().numpy(),
                self.p.t_bslot.cpu().numpy())

    def set_placement(self, x, y, sl):
        import torch
        x = np.asarray(x, dtype=np.int32)
        y = np.asarray(y, dtype=np.int32)
        sl = np.asarray(sl, dtype=np.int32)
        grid = np.full(self.p.gx * self.p.gy * self.p.cap, -1, dtype=np.int32)
        flat = (x.astype(np.int64) * self.p.gy + y) * self.p.cap + sl
        grid[flat] = np.arange(self.nl.num_blocks, dtype=np.int32)
        dev = self.p.device
        self.p.t_bx.copy_(torch.from_numpy(x).to(dev))
        self.p.t_by.copy_(torch.from_numpy(y).to(dev))
        self.p.t_bslot.copy_(torch.from_numpy(sl).to(dev))
        self.p.t_grid.copy_(torch.from_numpy(grid).to(dev))
        self.p.refresh_costs()

    def bb_cost(self):
        return self.p.bb_cost

    def td_cost(self):
        return self.p.td_cost

    def set_crit(self, conn_crit):
        self.p.set_crit(conn_crit)

    def conn_delays(self):
        from ..place.placer import analytic_delay_matrix
        nl = self.nl
        bx, by, _ = self.placement()
        dm = (self.p._dm if self.p._dm is not None
              else analytic_delay_matrix(self.arch))
        net_of_conn = np.repeat(np.arange(nl.num_nets),
                                np.diff(nl.net_sink_ptr))
        drv = nl.net_driver
        dx = np.abs(bx[nl.net_sinks] - bx[drv[net_of_conn]])
        dy = np.abs(by[nl.net_sinks] - by[drv[net_of_conn]])
        return dm[dx, dy].astype(np.float32)

    def check_place(self):
        return self.p.check_place()


def anneal_place_dist(netlist, arch, rank=0, world_size=1, seed=7,
                      timing_tradeoff=0.0, sta=None, inner_num=1.0,
                      crit_exp=1.0, verbose=False, macros=None,
                      engine="cpu", device="cuda:0"):
    """Distributed SA anneal. Returns the (rank-identical) Placement.

    macros: carry-chain groups (see anneal_place); applied BEFORE the
    per-rank reseed so every rank starts from the identical macro-legal
    placement, and boundary-straddling macros simply sit out the
    temperatures whose strips split them (C++ region guard).

    engine: "cpu" = SerialPlacer oracle (gloo tests), "gpu" = CDNA4
    batched-move kernels (one GPU per rank over RCCL). Macro moves are
    CPU-engine-only."""
    import torch
    import torch.distributed as dist
    from ..place.placer import Placement, analytic_delay_matrix
    from .. import ops

    tt = timing_tradeoff if sta is not None else 0.0
    if engine == "gpu":
        placer = _GpuPlacerAdapter(netlist, arch, seed, tt > 0, device,
                                   macros=macros)
    else:
        cpu = ops.cpu()
        if sta is not None and tt > 0:
            dm = analytic_delay_matrix(arch)
        else:
            dm = np.zeros(0, dtype=np.float32)
        tb = (arch.tile_btype_grid() if arch.is_heterogeneous()
              else np.empty(0, dtype=np.int8))
        placer = cpu.SerialPlacer(netlist.cpp(), arch.nx, arch.ny,
                                  arch.io_cap,
                                  np.ascontiguousarray(dm.ravel()), seed, tb)
        if macros:
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
    # identical initial placement everywhere (same seed; set_macros
    # consumed the same draws on every rank), then diverge the streams
    placer.reseed((seed + 1) * 1_000_003 + rank)
    nb = netlist.num_blocks
    ws = world_size
    move_lim = max(64, int(inner_num * (nb ** 1.3333)))
    rlim = float(max(arch.nx, arch.ny))

    crit = np.zeros(netlist.num_conns, dtype=np.float32)

    def refresh_crit():
        nonlocal crit
        if sta is None or tt <= 0:
            return
        cpd, slack, c = sta.analyze(placer.conn_delays())
        crit = (np.asarray(c) ** crit_exp).astype(np.float32)
        placer.set_crit(crit)

    # collectives run on the compute device: RCCL (nccl backend) requires
    # device tensors; gloo takes CPU tensors
    comm_dev = device if (engine == "gpu" and world_size > 1 and
                          dist.is_initialized() and
                          dist.get_backend() == "nccl") else "cpu"

    def fuse(owner_lo, owner_hi):
        """Masked all-reduce merge: every block is owned by exactly one
        rank (by its pre-move column), so summing owned positions
        reconstructs the full placement identically on every rank."""
        if ws <= 1:
            return
        x, y, sl = [np.asarray(a, dtype=np.int64)
                    for a in placer.placement()]
        own = (x0_pre >= owner_lo) & (x0_pre <= owner_hi)
        buf = torch.from_numpy(np.concatenate([
            np.where(own, x, 0), np.where(own, y, 0),
            np.where(own, sl, 0)])).to(comm_dev)
        dist.all_reduce(buf, op=dist.ReduceOp.SUM)
        m = buf.cpu().numpy()
        placer.set_placement(m[:nb].astype(np.int32),
                             m[nb:2 * nb].astype(np.int32),
                             m[2 * nb:].astype(np.int32))

    def global_srate():
        acc = placer.last_accepts()
        att = placer.last_valid_attempts()
        if ws > 1:
            t = torch.tensor([acc, att], dtype=torch.int64, device=comm_dev)
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
            acc, att = int(t[0]), int(t[1])
        return acc / max(1, att), att

    refresh_crit()
    norm_mode = tt > 0

    def norms():
        if norm_mode:
            return (max(placer.bb_cost(), 1e-12), max(placer.td_cost(), 1e-30))
        return (1.0, 1.0)

    # starting T from a probe pass on the own strip
    x0_pre = np.asarray(placer.placement()[0])
    lo, hi = _strips(arch.nx, ws, 0)[rank] if ws > 1 else (-1, -1)
    placer.set_move_region(lo, hi)
    bb_norm, td_norm = norms()
    placer.run_moves(1e30, rlim, max(64, nb), tt, bb_norm, td_norm)
    t = 20.0 * placer.last_delta_std()
    if ws > 1:
        tt_t = torch.tensor([t], dtype=torch.float64, device=comm_dev)
        dist.all_reduce(tt_t, op=dist.ReduceOp.MAX)
        t = float(tt_t[0])
    fuse(lo, hi)
    if t <= 0:
        t = 1.0

    itemp = 0
    history = []
    while True:
        refresh_crit()
        bb_norm, td_norm = norms()
        x0_pre = np.asarray(placer.placement()[0])
        if ws > 1:
            lo, hi = _strips(arch.nx, ws, itemp % 2)[rank]
        placer.set_move_region(lo, hi)
        placer.run_moves(t, rlim, move_lim, tt, bb_norm, td_norm)
        srate, att = global_srate()
        fuse(lo, hi)
        cost = placer.bb_cost()
        history.append((t, cost, srate, rlim))
        if verbose and rank == 0:
            print(f"[dist-sa] T={t:.3e} bb={cost:.1f} acc={srate:.2f} "
                  f"att={att} rlim={rlim:.1f}")
        if srate > 0.96:
            t *= 0.5
        elif srate > 0.8:
            t *= 0.9
        elif srate > 0.15 or rlim > 1:
            t *= 0.95
        else:
            t *= 0.8
        rlim = min(max(rlim * (1.0 - 0.44 + srate), 1.0),
                   float(max(arch.nx, arch.ny)))
        itemp += 1
        exit_cost = 1.0 if norm_mode else cost
        if t < 0.005 * exit_cost / max(1, netlist.num_nets):
            break
        if itemp > 500:
            break
    # quench on own strip, then final fuse
    x0_pre = np.asarray(placer.placement()[0])
    placer.run_moves(0.0, 1.0, move_lim, tt, bb_norm, td_norm)
    fuse(lo, hi)
    placer.set_move_region(-1, -1)
    ok, err = placer.check_place()
    if not ok:
        raise RuntimeError(f"dist check_place failed: {err}")
    x, y, sl = placer.placement()
    return Placement(np.asarray(x), np.asarray(y), np.asarray(sl),
                     bb_cost=placer.bb_cost(), td_cost=placer.td_cost(),
                     stats={"temps": itemp, "move_lim": move_lim,
                            "history": history, "engine": f"dist-{engine} x{ws}"})
