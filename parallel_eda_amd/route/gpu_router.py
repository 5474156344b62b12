"""GPU PathFinder engine: HBM-resident rr graph + wavefront router kernel.

Host side of csrc/hip/router_kernel.hip: uploads the SoA graph, sizes the
per-slot search scratch, runs the rip-up-and-reroute outer loop with
criticality-ordered sinks, and recovers failed nets by growing their
bounding boxes (reference: bb_factor growth; route failures re-tried).
"""
import numpy as np

from ..arch.archdef import ArchDef
from ..ops import hip_api
from .router import net_rr_terminals, RouteResult


def _torch():
    import torch
    return torch


def xcd_interleaved_order(bb, order, n_xcd=8):
    """Reorder a concurrent-launch queue so each XCD's workgroups pull
    spatially-clustered nets. MI355X dispatches consecutive workgroups
    round-robin across the 8 XCDs (each with a private L2); with the
    atomic-cursor queue, consecutive pops therefore land on DIFFERENT
    XCDs. Sorting nets by bb-center Morton code clusters them spatially,
    and an 8-way deal (net i -> position (i%%8)*ceil(n/8) + i//8) maps
    each spatial cluster onto ONE XCD's pop sequence — so an XCD's L2
    sees one region of the rr graph instead of all of them. Flag-gated
    (PNR_XCD_ORDER=1) pending a round-2 rocprofv3 L2-hit A/B; a pure
    permutation, so results are unchanged in concurrent mode either way
    (same net set, same atomics)."""
    cx = (bb[order, 0].astype(np.int64) + bb[order, 2])
    cy = (bb[order, 1].astype(np.int64) + bb[order, 3])

    def spread(v):
        v = (v | (v << 8)) & 0x00FF00FF
        v = (v | (v << 4)) & 0x0F0F0F0F
        v = (v | (v << 2)) & 0x33333333
        v = (v | (v << 1)) & 0x55555555
        return v

    morton = spread(cx & 0xFFFF) | (spread(cy & 0xFFFF) << 1)
    spatial = order[np.argsort(morton, kind="stable")]
    n = len(spatial)
    chunk = (n + n_xcd - 1) // n_xcd
    # gather: queue position p (popped by XCD p % n_xcd) takes from
    # spatial chunk (p % n_xcd) — so each XCD walks one Morton run
    pos = np.arange(n)
    src = (pos % n_xcd) * chunk + pos // n_xcd
    ok = src < n
    out = np.empty(n, dtype=spatial.dtype)
    out[pos[ok]] = spatial[src[ok]]
    if not ok.all():
        used = np.zeros(n, dtype=bool)
        used[src[ok]] = True
        out[pos[~ok]] = spatial[~used]
    return out


def schedule_bb_waves(bb, net_ids, areas, nx, ny, cell=8):
    """Greedy bb-disjoint wave schedule (ParaDRo-style: reference builds a
    bb-overlap graph and colors it, partitioning_multi_sink...:3598; we
    rasterize bbs onto a coarse cell grid and greedily assign each net the
    first wave whose cells are all free — deterministic, O(nets x cells)).
    Nets in the same wave have disjoint cell footprints, hence disjoint
    bbs, hence disjoint search state and congestion reads. Largest-area
    first keeps the giant nets from serializing the tail. Pure host-side
    numpy — unit-tested on CPU (tests/test_route.py)."""
    ncx = (nx + 2 + cell - 1) // cell
    ncy = (ny + 2 + cell - 1) // cell
    next_free = np.zeros(ncx * ncy, dtype=np.int32)
    order = net_ids[np.argsort(-areas[net_ids], kind="stable")]
    wave_of = np.zeros(len(order), dtype=np.int32)
    for i, n in enumerate(order):
        cx0 = bb[n, 0] // cell; cy0 = bb[n, 1] // cell
        cx1 = bb[n, 2] // cell; cy1 = bb[n, 3] // cell
        blockv = next_free[:].reshape(ncx, ncy)[cx0:cx1 + 1, cy0:cy1 + 1]
        w = int(blockv.max())
        blockv[:] = np.maximum(blockv, w + 1)
        wave_of[i] = w
    n_waves = int(wave_of.max()) + 1 if len(order) else 0
    return [order[wave_of == w] for w in range(n_waves)]


class DevGraph:
    """Device-resident SoA rr graph, shared across GpuRouter instances
    (the graph depends only on the arch; repeated flows re-place and
    re-route but never re-upload the 12.7M-node/78M-edge CSR)."""

    def __init__(self, g, arch: ArchDef, device="cuda:0"):
        torch = _torch()
        if g.num_edges >= 2**31:
            raise ValueError("edge count exceeds int32 CSR limit")
        self.g = g
        self.device = device

        def up(a):
            return torch.from_numpy(np.ascontiguousarray(a)).to(device)

        self.t_type = up(np.asarray(g.type))
        xlow = np.asarray(g.xlow); ylow = np.asarray(g.ylow)
        self.t_xlow = up(xlow); self.t_ylow = up(ylow)
        self.t_xhigh = up(np.asarray(g.xhigh))
        self.t_yhigh = up(np.asarray(g.yhigh))
        self.t_cap = up(np.asarray(g.capacity))
        self.t_R = up(np.asarray(g.node_R)); self.t_C = up(np.asarray(g.node_C))
        self.t_row_ptr = up(np.asarray(g.row_ptr).astype(np.int32))
        self.t_edge_dst = up(np.asarray(g.edge_dst))
        self.t_edge_sw = up(np.asarray(g.edge_sw))
        self.t_sw_R = up(np.asarray(g.sw_R))
        self.t_sw_Tdel = up(np.asarray(g.sw_Tdel))
        self.t_base_cost = up(np.asarray(g.base_cost))

        # dense per-tile node index (anchor = (xlow, ylow))
        gy = arch.ny + 2
        tile = xlow.astype(np.int64) * gy + ylow.astype(np.int64)
        order = np.argsort(tile, kind="stable")
        sorted_tile = tile[order]
        starts = np.r_[0, np.nonzero(np.diff(sorted_tile))[0] + 1]
        counts = np.diff(np.r_[starts, len(tile)])
        idx_in_tile = np.empty(len(tile), dtype=np.int32)
        pos = np.arange(len(tile)) - np.repeat(starts, counts)
        idx_in_tile[order] = pos.astype(np.int32)
        self.npt = int(counts.max())
        self.t_idx_in_tile = up(idx_in_tile)


class GpuRouter:
    def __init__(self, g, arch: ArchDef, src_rr, sink_ptr, sink_rr,
                 device="cuda:0", astar_fac=1.8, n_small_slots=1024,
                 n_large_slots=128, bb_margin=4, max_rounds=200000,
                 delta_fac=1.5, deterministic=False,
                 concurrent_threshold=48, occ=None, dev_graph=None):
        torch = _torch()
        self.torch = torch
        self.device = device
        self.arch = arch
        self.g = g
        self.num_nets = len(src_rr)
        import os as _os
        astar_fac = float(_os.environ.get("PNR_ASTAR", astar_fac))
        delta_fac = float(_os.environ.get("PNR_DELTA", delta_fac))
        # deterministic mode needs an ADMISSIBLE lookahead (astar_fac<=1):
        # with an inflated heuristic the termination point depends on
        # which paths the racing relaxations discovered first
        self.astar_fac = min(astar_fac, 1.0) if deterministic else astar_fac
        self.bb_margin = bb_margin
        self.max_rounds = max_rounds
        self.delta_fac = delta_fac
        self.deterministic = deterministic
        self.concurrent_threshold = concurrent_threshold

        # ---- graph (SoA, device-resident, shareable) ----
        dg = dev_graph if dev_graph is not None else DevGraph(g, arch, device)
        if dg.g is not g or dg.device != device:
            raise ValueError("dev_graph built for a different graph/device")
        self.dev_graph = dg
        for name in ("t_type", "t_xlow", "t_ylow", "t_xhigh", "t_yhigh",
                     "t_cap", "t_R", "t_C", "t_row_ptr", "t_edge_dst",
                     "t_edge_sw", "t_sw_R", "t_sw_Tdel", "t_base_cost",
                     "t_idx_in_tile", "npt"):
            setattr(self, name, getattr(dg, name))

        def up(a, dtype=None):
            t = torch.from_numpy(np.ascontiguousarray(a))
            if dtype is not None:
                t = t.to(dtype)
            return t.to(device)

        # ---- nets ----
        self.src_rr = np.asarray(src_rr, dtype=np.int32)
        self.sink_ptr = np.asarray(sink_ptr, dtype=np.int32)
        self.sink_rr = np.asarray(sink_rr, dtype=np.int32)
        self.n_sinks = len(self.sink_rr)
        self.t_net_src = up(self.src_rr)
        self.t_sink_ptr = up(self.sink_ptr)

        # bounding boxes (terminal bb + margin)
        self.bb_margin_per_net = np.full(self.num_nets, bb_margin, dtype=np.int32)
        self.bb = self._compute_bbs()
        self.t_bb = up(self.bb)
        self._bb_version = 0
        self._waves_cache = None

        # slot classes: the small class covers the 95th-percentile net bb
        # (state stays cheap, touched-list reset keeps clears O(visited));
        # the rest use whole-chip state in the large class.
        areas = self._bb_areas(self.bb)
        p95 = int(np.percentile(areas, 95)) if len(areas) else 64
        self.bb_max_small_area = int(min(max(p95, 64), 16384))
        self.n_small_slots = n_small_slots
        self.n_large_slots = n_large_slots
        self.small_cap = self.bb_max_small_area * self.npt
        self.large_cap = ((arch.nx + 2) * (arch.ny + 2)) * self.npt
        # if the whole chip fits in a small slot, drop the large class size
        if self.large_cap <= self.small_cap:
            self.large_cap = self.small_cap

        # with frontier dedup an overflow means > f_cap LIVE nodes; the
        # pres-0 first iteration's tie plateaus get close at Titan scale,
        # so size generously (memory is 16 B/entry: 2*2^18 / slot = 8 MiB)
        self.f_cap_small = 1 << 18
        self.f_cap_large = 1 << 22
        self.t_cap_small = self.small_cap
        self.t_cap_large = self.large_cap

        # sanity-check the planned scratch against free VRAM before
        # allocating (protects N-GPU runs from silent overcommit)
        planned = (n_small_slots * self.small_cap + n_large_slots * self.large_cap) * 8 \
            + (n_small_slots * 2 * self.f_cap_small +
               n_large_slots * 2 * self.f_cap_large) * 16 \
            + (n_small_slots * self.t_cap_small +
               n_large_slots * self.t_cap_large) * 4
        try:
            free, total = torch.cuda.mem_get_info(device)
            if planned > 0.6 * free:
                import sys
                print(f"GpuRouter: planned scratch {planned/2**30:.1f} GiB vs "
                      f"{free/2**30:.1f} GiB free — shrinking slots",
                      file=sys.stderr)
                while planned > 0.5 * free and n_small_slots > 128:
                    n_small_slots //= 2
                    self.n_small_slots = n_small_slots
                    planned = (n_small_slots * self.small_cap + n_large_slots * self.large_cap) * 8 \
                        + (n_small_slots * 2 * self.f_cap_small +
                           n_large_slots * 2 * self.f_cap_large) * 16 \
                        + (n_small_slots * self.t_cap_small +
                           n_large_slots * self.t_cap_large) * 4
        except Exception:
            pass

        nb = torch.int64
        self.t_state = torch.empty(
            n_small_slots * self.small_cap + n_large_slots * self.large_cap,
            dtype=nb, device=device)
        self.t_frontier = torch.empty(
            (n_small_slots * 2 * self.f_cap_small +
             n_large_slots * 2 * self.f_cap_large) * 4,
            dtype=torch.float32, device=device)
        self.t_touched = torch.empty(
            n_small_slots * self.t_cap_small + n_large_slots * self.t_cap_large,
            dtype=torch.int32, device=device)
        # per-state-entry in-queue flag (frontier dedup; starts clear and
        # is restored clear by the touched-list reset)
        self.t_inq = torch.zeros(self.t_state.numel(), dtype=torch.uint8,
                                 device=device)

        # trees: per-net capacity
        nsinks = np.diff(self.sink_ptr)
        bw = self.bb[:, 2] - self.bb[:, 0] + 1
        bh = self.bb[:, 3] - self.bb[:, 1] + 1
        cap = 32 + nsinks * (2 * (bw.astype(np.int64) + bh.astype(np.int64)) + 64)
        self.tree_off = np.r_[0, np.cumsum(cap)].astype(np.int64)
        total_tree = int(self.tree_off[-1])
        self.t_tree_off = up(self.tree_off)
        self.t_tree_node = torch.zeros(total_tree, dtype=torch.int32, device=device)
        self.t_tree_parent = torch.zeros(total_tree, dtype=torch.int32, device=device)
        self.t_tree_sw = torch.zeros(total_tree, dtype=torch.int8, device=device)
        self.t_tree_delay = torch.zeros(total_tree, dtype=torch.float32, device=device)
        self.t_tree_len = torch.zeros(self.num_nets, dtype=torch.int32, device=device)
        self.t_sink_delay = torch.zeros(self.n_sinks, dtype=torch.float32, device=device)

        # congestion
        if occ is None:
            self.t_occ = torch.zeros(g.num_nodes, dtype=torch.int32, device=device)
        else:
            self.t_occ = occ
        self.t_acc = torch.ones(g.num_nodes, dtype=torch.float32, device=device)
        self.t_fail = torch.zeros(self.num_nets, dtype=torch.int32, device=device)
        # per-net completeness: True until the net has been routed with all
        # sinks reached (the distributed driver must never call
        # a flow successful while a net exhausted its bb-growth retries or
        # was ripped for ownership hand-off and never rerouted)
        self.incomplete = np.ones(self.num_nets, dtype=bool)
        self.t_cursors = torch.zeros(2, dtype=torch.int32, device=device)
        self.t_stats = torch.zeros(8, dtype=torch.int64, device=device)
        self.t_net_scans = torch.zeros(self.num_nets, dtype=torch.int64,
                                       device=device)
        self.t_overuse = torch.zeros(8, dtype=torch.int32, device=device)

        # lookahead constants (same as serial oracle)
        self.seg_delay = float(arch.T_sw + arch.C_wire * arch.L *
                               (arch.R_sw + 0.5 * arch.R_wire * arch.L))
        self.ipin_delay = float(arch.T_ipin)
        self.seg_base = float(arch.base_costs()[4])
        self.ipin_base = float(arch.base_costs()[3])

        # init state to INF once
        self._fill_state()
        self.lib = hip_api.lib()

        # multi-workgroup straggler engine (csrc/hip/router_mwg.hip):
        # whole-device search for nets whose bb covers a large fraction of
        # the chip — one workgroup per net serializes the endgame
        # (measured 106 s for 123 nets at bitcoin scale). Buffers lazy.
        self.mwg_area_threshold = max(
            ((arch.nx + 2) * (arch.ny + 2)) // 8, self.bb_max_small_area * 2)
        self.mwg_max_per_launch = 64
        self._mwg_bufs = None

    def _mwg_buffers(self):
        if self._mwg_bufs is None:
            torch = self.torch
            f_cap = int(min(1 << 23, max(1 << 20, self.g.num_nodes)))
            state = torch.empty(self.g.num_nodes, dtype=torch.int64,
                                device=self.device)
            rc = self.lib.pnr_fill_u64_launch(
                ct_ptr(state), 0xFFFFFFFFFFFFFFFF, state.numel(),
                self._stream())
            hip_api.check(rc, "mwg_fill_state")
            frA = torch.empty(f_cap * 4, dtype=torch.float32,
                              device=self.device)
            frB = torch.empty(f_cap * 4, dtype=torch.float32,
                              device=self.device)
            inq = torch.zeros(self.g.num_nodes, dtype=torch.uint8,
                              device=self.device)
            ctrl = torch.zeros(16, dtype=torch.int32, device=self.device)
            self._mwg_bufs = (state, inq, frA, frB, ctrl, f_cap)
        return self._mwg_bufs

    # ---- helpers ----
    def _compute_bbs(self):
        xlow = np.asarray(self.g.xlow); ylow = np.asarray(self.g.ylow)
        nx, ny = self.arch.nx, self.arch.ny
        sx = xlow[self.src_rr].astype(np.int32)
        sy = ylow[self.src_rr].astype(np.int32)
        kx = xlow[self.sink_rr].astype(np.int32)
        ky = ylow[self.sink_rr].astype(np.int32)
        seg = self.sink_ptr[:-1]
        has = self.sink_ptr[1:] > seg
        # reduceat over each net's sink slice. A SENTINEL element keeps
        # every start index valid (trailing zero-sink nets have
        # seg == len(kx)); clamping indices instead would silently
        # truncate the PREVIOUS net's segment (caught by
        # tests/test_property.py hypothesis fuzzing). Empty slices
        # produce the sentinel and are masked back to the source below.
        BIG = np.int32(1 << 30)
        xmin = np.minimum.reduceat(np.r_[kx, BIG], seg)
        xmax = np.maximum.reduceat(np.r_[kx, -BIG], seg)
        ymin = np.minimum.reduceat(np.r_[ky, BIG], seg)
        ymax = np.maximum.reduceat(np.r_[ky, -BIG], seg)
        xmin = np.where(has, np.minimum(xmin, sx), sx)
        xmax = np.where(has, np.maximum(xmax, sx), sx)
        ymin = np.where(has, np.minimum(ymin, sy), sy)
        ymax = np.where(has, np.maximum(ymax, sy), sy)
        m = self.bb_margin_per_net
        bb = np.empty((self.num_nets, 4), dtype=np.int16)
        bb[:, 0] = np.maximum(0, xmin - m)
        bb[:, 1] = np.maximum(0, ymin - m)
        bb[:, 2] = np.minimum(nx + 1, xmax + m)
        bb[:, 3] = np.minimum(ny + 1, ymax + m)
        return bb

    def _bb_areas(self, bb):
        return ((bb[:, 2] - bb[:, 0] + 1).astype(np.int64) *
                (bb[:, 3] - bb[:, 1] + 1).astype(np.int64))

    def _fill_state(self):
        rc = hip_api.lib().pnr_fill_u64_launch(
            ct_ptr(self.t_state), 0xFFFFFFFFFFFFFFFF, self.t_state.numel(),
            self._stream())
        hip_api.check(rc, "fill_state")
        if hasattr(self, "t_inq"):
            self.t_inq.zero_()

    def _stream(self):
        return self.torch.cuda.current_stream().cuda_stream

    # ---- ParaDRo-style wave schedule (reference:
    #      partitioning_multi_sink...cxx:3563-4450 overlap graph + coloring).
    # Nets whose bbs share a coarse grid cell go in different waves; one
    # kernel launch per wave. Within a wave bbs are disjoint, so concurrent
    # nets never read or write each other's congestion => the whole
    # iteration is DETERMINISTIC (fixed schedule, like the reference's
    # partitioning router).
    def _schedule_waves(self, net_ids):
        # cache the full-set schedule; it only changes when bbs grow
        full = len(net_ids) == self.num_nets
        if full and getattr(self, "_waves_cache", None) is not None \
                and self._waves_cache[0] == self._bb_version:
            return self._waves_cache[1]
        waves = self._schedule_waves_impl(net_ids)
        if full:
            self._waves_cache = (self._bb_version, waves)
        return waves

    def _schedule_waves_impl(self, net_ids):
        areas = self._bb_areas(self.bb)
        return schedule_bb_waves(self.bb, net_ids, areas,
                                 self.arch.nx, self.arch.ny)

    # ---- one PathFinder iteration ----
    def route_iteration(self, crit, pres_fac, net_subset=None, fail_ok=False,
                        partial=False, force_waves=False):
        """crit: per-sink criticality aligned with sink_rr (original order).
        Returns (overused_count, sink_delays aligned with original order).
        net_subset: optional array of net ids to (re)route; others keep
        their route trees (multi-GPU partitioning / selective reroute).
        fail_ok: exhausted retries leave the stragglers partially routed
        instead of raising (multi-rank benches must not kill a rank
        mid-collective).
        partial: partial rip-up (keep clean subtrees, skip connected
        sinks; reference route_tree_mark_congested_...; production in
        the incremental flows, GPU-validated round 2).
        force_waves: route even medium sets on the bb-disjoint wave
        schedule (the drivers' stall breaker for stagnant concurrent
        endgames)."""
        self._partial = partial
        torch = self.torch
        import time as _time
        if not hasattr(self, "prof"):
            self.prof = dict(prep=0.0, sched=0.0, kernel=0.0, bbgrow=0.0,
                             launches=0)
        _tp = _time.perf_counter()
        # order sinks by decreasing criticality within each net
        net_of_sink = np.repeat(np.arange(self.num_nets), np.diff(self.sink_ptr))
        perm = np.lexsort((-crit, net_of_sink))
        t_sink_rr = torch.from_numpy(self.sink_rr[perm]).to(self.device)
        t_crit = torch.from_numpy(np.ascontiguousarray(crit[perm], dtype=np.float32)).to(self.device)
        t_sink_orig = torch.from_numpy(perm.astype(np.int32)).to(self.device)

        todo = (np.arange(self.num_nets, dtype=np.int64) if net_subset is None
                else np.asarray(net_subset, dtype=np.int64))
        attempted = todo
        attempts = 0
        self.last_retries = []
        # Scratch-overflow retry ladder (frontier/touched/path-cap, codes
        # 1/5/6, are SLOT limits, not bb problems — growing the bb makes
        # the overflowing search bigger): small slot -> large slot
        # (concurrent, 32x frontier) -> MWG (whole-device, serial).
        force_large = np.zeros(self.num_nets, dtype=bool)
        force_mwg = np.zeros(self.num_nets, dtype=bool)
        # congestion-aware lookahead: scale the heuristic's wire cost by
        # the mean effective cost of used wires so A* stays focused when
        # pres/acc inflate edge costs far beyond base (keeps the estimate
        # admissible-ish w.r.t. congested regions; disable with
        # cong_aware_lookahead=False)
        # A/B at LU32 measured 5.5x WORSE with this on (scan/sink 45k ->
        # 430k): inflating h without scaling the bucket width explodes the
        # kept-entry rescans. Off by default; revisit with the bucketed
        # frontier (docs/ROADMAP.md item 1).
        if getattr(self, "cong_aware_lookahead", False) and pres_fac > 0:
            t = self.torch
            chan = (self.t_type == 4) | (self.t_type == 5)
            used = chan & (self.t_occ > 0)
            if bool(used.any().item()):
                over = (self.t_occ + 1 - self.t_cap.to(t.int32)).clamp(min=0)
                pres_t = 1.0 + pres_fac * over.to(t.float32)
                m = float((self.t_acc * pres_t)[used].mean().item())
                self._cong_mult = float(np.clip(m, 1.0, 64.0))
            else:
                self._cong_mult = 1.0
        else:
            self._cong_mult = 1.0
        self.prof["prep"] += _time.perf_counter() - _tp
        while True:
            self.t_fail.zero_()
            # straggler split: huge-bb nets go to the multi-workgroup
            # engine (whole-device per-net search) instead of a wave /
            # concurrent slot. Applied in the endgame and on late retries;
            # capped so a large qualifying set can't serialize the launch.
            mwg_nets = None
            if not self.deterministic:
                big = np.zeros(0, dtype=todo.dtype)
                if len(todo) <= self.concurrent_threshold or attempts >= 2:
                    ar = self._bb_areas(self.bb)[todo]
                    big = todo[ar >= self.mwg_area_threshold]
                    if len(big) > self.mwg_max_per_launch:
                        big = big[np.argsort(
                            -self._bb_areas(self.bb)[big],
                            kind="stable")][:self.mwg_max_per_launch]
                forced = todo[force_mwg[todo]]
                if len(big) or len(forced):
                    sel = np.isin(todo, np.union1d(big, forced))
                    mwg_nets = todo[sel]
                    todo = todo[~sel]
            # Large reroute sets run as ONE concurrent launch (net-level
            # parallelism with atomic congestion, reference locking_route
            # family); small/endgame sets get the deterministic bb-disjoint
            # wave schedule (ParaDRo family), which is what resolves the
            # last contested nodes.
            import os as _os
            if attempts > 0 and _os.environ.get("PNR_RETRY_SERIAL"):
                waves = [np.asarray([n]) for n in todo]   # bisection mode
            elif not self.deterministic and (
                    (len(todo) > self.concurrent_threshold
                     and not force_waves) or attempts > 0):
                # one concurrent launch; biggest work first for load balance
                areas_t = self._bb_areas(self.bb)[todo]
                nsk = (self.sink_ptr[todo + 1] - self.sink_ptr[todo]).astype(np.int64)
                order = todo[np.argsort(-(areas_t * nsk), kind="stable")]
                if _os.environ.get("PNR_XCD_ORDER"):
                    order = xcd_interleaved_order(self.bb, order)
                waves = [order]
            else:
                _ts = _time.perf_counter()
                waves = self._schedule_waves(todo)
                self.prof["sched"] += _time.perf_counter() - _ts
            _tk = _time.perf_counter()
            if mwg_nets is not None:
                state, inq, frA, frB, ctrl, f_cap = self._mwg_buffers()
                empty_q = torch.zeros(0, dtype=torch.int32, device=self.device)
                margs = self._make_args(t_sink_rr, t_crit, t_sink_orig,
                                        empty_q, empty_q, pres_fac)
                for inet in mwg_nets:
                    s0 = int(self.sink_ptr[inet])
                    s1 = int(self.sink_ptr[inet + 1])
                    rc = self.lib.pnr_mwg_route_net(
                        hip_api.ct.byref(margs), int(inet), s0, s1,
                        ct_ptr(state), ct_ptr(inq), ct_ptr(frA),
                        ct_ptr(frB), f_cap, ct_ptr(ctrl), 48,
                        self._stream())
                    hip_api.check(rc, "mwg_route_net")
                self.prof["launches"] += len(mwg_nets)
            areas = self._bb_areas(self.bb)
            import os
            dbg = os.environ.get("PNR_ROUTE_DEBUG")
            for wi, wave in enumerate(waves):
                is_small = (areas[wave] <= self.bb_max_small_area) & \
                    ~force_large[wave]
                small = wave[is_small].astype(np.int32)
                large = wave[~is_small].astype(np.int32)
                q_small = torch.from_numpy(small).to(self.device)
                q_large = torch.from_numpy(large).to(self.device)
                self.t_cursors.zero_()
                args = self._make_args(t_sink_rr, t_crit, t_sink_orig,
                                       q_small, q_large, pres_fac)
                if dbg:
                    print(f"    [launch] attempt={attempts} wave={wi}/{len(waves)} "
                          f"small={len(small)} large={len(large)} "
                          f"nets={wave[:4].tolist() if len(wave) <= 8 else ''}",
                          flush=True)
                rc = self.lib.pnr_route_nets(hip_api.ct.byref(args), self._stream())
                hip_api.check(rc, "route_nets")
                if dbg:
                    torch.cuda.synchronize(self.device)
                    print(f"    [done]", flush=True)
                # no host sync between waves: stream order serializes them
            torch.cuda.synchronize(self.device)
            _dtk = _time.perf_counter() - _tk
            self.prof["kernel"] += _dtk
            self.prof["launches"] += len(waves)
            fail = self.t_fail.cpu().numpy()
            failed = np.nonzero(fail)[0]
            if _os.environ.get("PNR_ATTEMPT_LOG"):
                import sys as _sys
                cd, cn = np.unique(fail[failed], return_counts=True)
                print(f"    [attempt {attempts}] nets={len(todo)} "
                      f"mwg={0 if mwg_nets is None else len(mwg_nets)} "
                      f"waves={len(waves)} t={_dtk:.2f}s "
                      f"fails={dict(zip(cd.tolist(), cn.tolist()))}",
                      file=_sys.stderr, flush=True)
            if len(failed) == 0:
                self.incomplete[attempted] = False
                break
            attempts += 1
            self.last_retries.append(
                (len(failed), np.unique(fail[failed]).tolist()))
            if attempts > 6:
                if fail_ok:
                    import sys
                    print(f"router: giving up on {len(failed)} nets after "
                          f"retries (codes {np.unique(fail[failed])})",
                          file=sys.stderr, flush=True)
                    self.incomplete[attempted] = False
                    self.incomplete[failed] = True
                    break
                raise RuntimeError(
                    f"router: {len(failed)} nets failed after retries "
                    f"(codes {np.unique(fail[failed])})")
            # Scratch-limit failures (frontier/touched/path-cap overflow,
            # codes 1/6/5) retry on the MWG engine at the SAME bb; genuine
            # reachability failures grow the bb and retry.
            _tb = _time.perf_counter()
            codes = fail[failed]
            # deterministic mode has no MWG path: it must keep growing
            # (the doubled bb moves the net to the large whole-chip class,
            # whose scratch is bigger — the round-1 behavior)
            scratch = (np.isin(codes, (1, 5, 6)) if not self.deterministic
                       else np.zeros(len(codes), dtype=bool))
            sc = failed[scratch]
            # scratch failure in the small class: retry in the large class
            # (concurrent); failure in the large class: MWG engine
            was_large = (self._bb_areas(self.bb)[sc] >
                         self.bb_max_small_area) | force_large[sc]
            force_mwg[sc[was_large]] = True
            force_large[sc] = True
            grow = failed[~scratch]
            if len(grow):
                self.bb_margin_per_net[grow] = np.minimum(
                    self.bb_margin_per_net[grow] * 2 + 4,
                    max(self.arch.nx, self.arch.ny) + 2)
                self.bb = self._compute_bbs()
                self.t_bb.copy_(torch.from_numpy(self.bb).to(self.device))
                self._bb_version += 1
            # state may be dirty for failed slots; refill (cheap)
            self._fill_state()
            self.prof["bbgrow"] += _time.perf_counter() - _tb
            todo = failed

        overused = int((self.t_occ > self.t_cap.to(torch.int32)).sum().item())
        sink_delays = self.t_sink_delay.cpu().numpy()
        return overused, sink_delays

    def search_stats(self):
        s = self.t_stats.cpu().numpy()
        return dict(rounds=int(s[0]), scanned=int(s[1]), sinks=int(s[2]),
                    touched=int(s[3]))

    def reset_search_stats(self):
        self.t_stats.zero_()
        self.t_net_scans.zero_()

    def top_cost_nets(self, k=10):
        import torch
        v, idx = torch.topk(self.t_net_scans, min(k, self.num_nets))
        idx = idx.cpu().numpy(); v = v.cpu().numpy()
        nsk = self.sink_ptr[idx + 1] - self.sink_ptr[idx]
        areas = self._bb_areas(self.bb)[idx]
        return [(int(i), int(s), int(n), int(a))
                for i, s, n, a in zip(idx, v, nsk, areas)]

    def incomplete_nets(self):
        """Nets with unreached sinks: exhausted bb-growth retries under
        fail_ok, or ripped for ownership hand-off and not yet rerouted."""
        return np.nonzero(self.incomplete)[0]

    def rip_up_nets(self, net_ids):
        """Remove the given nets' routes (ownership hand-off)."""
        torch = self.torch
        self.incomplete[np.asarray(net_ids, dtype=np.int64)] = True
        ids = torch.from_numpy(np.asarray(net_ids, dtype=np.int32)).to(self.device)
        rc = self.lib.pnr_rip_up_nets(
            ct_ptr(self.t_tree_off), ct_ptr(self.t_tree_node),
            ct_ptr(self.t_tree_len), ct_ptr(ids), len(net_ids),
            ct_ptr(self.t_occ), self._stream())
        hip_api.check(rc, "rip_up_nets")
        torch.cuda.synchronize(self.device)

    def congested_nets(self):
        """Nets whose tree touches an overused node (selective-reroute set;
        reference: build_phase_two congested-nets-only)."""
        torch = self.torch
        flags = torch.zeros(self.num_nets, dtype=torch.uint8, device=self.device)
        rc = self.lib.pnr_flag_congested_nets(
            ct_ptr(self.t_tree_off), ct_ptr(self.t_tree_node),
            ct_ptr(self.t_tree_len), ct_ptr(self.t_occ), ct_ptr(self.t_cap),
            self.num_nets, ct_ptr(flags), self._stream())
        hip_api.check(rc, "flag_congested_nets")
        torch.cuda.synchronize(self.device)
        return np.nonzero(flags.cpu().numpy())[0]

    def update_acc(self, acc_fac):
        rc = self.lib.pnr_update_acc(
            ct_ptr(self.t_occ), ct_ptr(self.t_cap), ct_ptr(self.t_acc),
            hip_api.ct.c_float(acc_fac), self.g.num_nodes, self._stream())
        hip_api.check(rc, "update_acc")

    def wirelength(self):
        torch = self.torch
        chan = (self.t_type == 4) | (self.t_type == 5)
        length = (self.t_xhigh - self.t_xlow + self.t_yhigh - self.t_ylow + 1).to(torch.int64)
        return int((self.t_occ.to(torch.int64) * length * chan.to(torch.int64)).sum().item())

    def check_occ_recount(self):
        """Debug cross-check: occ == recount over route trees."""
        torch = self.torch
        recount = torch.zeros_like(self.t_occ)
        ids = torch.arange(self.num_nets, dtype=torch.int32, device=self.device)
        rc = self.lib.pnr_recount_occ(
            ct_ptr(self.t_tree_off), ct_ptr(self.t_tree_node),
            ct_ptr(self.t_tree_len), ct_ptr(ids), self.num_nets,
            ct_ptr(recount), self._stream())
        hip_api.check(rc, "recount_occ")
        torch.cuda.synchronize(self.device)
        return bool((recount == self.t_occ).all().item())

    def get_tree(self, inet):
        o0, o1 = int(self.tree_off[inet]), int(self.tree_off[inet + 1])
        ln = int(self.t_tree_len[inet].item())
        return (self.t_tree_node[o0:o0 + ln].cpu().numpy(),
                self.t_tree_parent[o0:o0 + ln].cpu().numpy(),
                self.t_tree_sw[o0:o0 + ln].cpu().numpy(),
                self.t_tree_delay[o0:o0 + ln].cpu().numpy())

    def _make_args(self, t_sink_rr, t_crit, t_sink_orig, q_small, q_large,
                   pres_fac):
        a = hip_api.RouteLaunchArgs()
        a.type = ct_ptr(self.t_type); a.xlow = ct_ptr(self.t_xlow)
        a.ylow = ct_ptr(self.t_ylow); a.xhigh = ct_ptr(self.t_xhigh)
        a.yhigh = ct_ptr(self.t_yhigh); a.capacity = ct_ptr(self.t_cap)
        a.R = ct_ptr(self.t_R); a.C = ct_ptr(self.t_C)
        a.row_ptr = ct_ptr(self.t_row_ptr); a.edge_dst = ct_ptr(self.t_edge_dst)
        a.edge_sw = ct_ptr(self.t_edge_sw); a.sw_R = ct_ptr(self.t_sw_R)
        a.sw_Tdel = ct_ptr(self.t_sw_Tdel); a.base_cost = ct_ptr(self.t_base_cost)
        a.idx_in_tile = ct_ptr(self.t_idx_in_tile)
        a.num_nodes = self.g.num_nodes; a.nx = self.arch.nx; a.ny = self.arch.ny
        a.L = self.arch.L; a.npt = self.npt
        a.net_src = ct_ptr(self.t_net_src); a.sink_ptr = ct_ptr(self.t_sink_ptr)
        a.sink_rr = ct_ptr(t_sink_rr); a.crit = ct_ptr(t_crit)
        a.sink_orig = ct_ptr(t_sink_orig)
        a.bb = ct_ptr(self.t_bb); a.num_nets = self.num_nets
        a.tree_off = ct_ptr(self.t_tree_off); a.tree_node = ct_ptr(self.t_tree_node)
        a.tree_parent = ct_ptr(self.t_tree_parent); a.tree_sw = ct_ptr(self.t_tree_sw)
        a.tree_delay = ct_ptr(self.t_tree_delay); a.tree_len = ct_ptr(self.t_tree_len)
        a.sink_delay = ct_ptr(self.t_sink_delay)
        a.astar_fac = self.astar_fac; a.pres_fac = pres_fac
        a.seg_delay = self.seg_delay; a.ipin_delay = self.ipin_delay
        a.seg_base = self.seg_base; a.ipin_base = self.ipin_base
        a.delta_fac = self.delta_fac
        a.cong_mult = getattr(self, "_cong_mult", 1.0)
        a.max_rounds = self.max_rounds
        import os as _os
        a.strict_term = 1 if (self.deterministic or
                              _os.environ.get("PNR_FORCE_STRICT")) else 0
        a.queue_small = ct_ptr(q_small); a.n_queue_small = len(q_small)
        a.queue_large = ct_ptr(q_large); a.n_queue_large = len(q_large)
        a.q_cursors = ct_ptr(self.t_cursors)
        a.occ = ct_ptr(self.t_occ); a.acc = ct_ptr(self.t_acc)
        a.state_base = ct_ptr(self.t_state)
        a.small_cap = self.small_cap; a.large_cap = self.large_cap
        a.n_small_slots = self.n_small_slots; a.n_large_slots = self.n_large_slots
        a.frontier_base = ct_ptr(self.t_frontier)
        a.f_cap_small = self.f_cap_small; a.f_cap_large = self.f_cap_large
        a.touched_base = ct_ptr(self.t_touched)
        a.t_cap_small = self.t_cap_small; a.t_cap_large = self.t_cap_large
        a.fail_flags = ct_ptr(self.t_fail)
        a.stats = ct_ptr(self.t_stats)
        a.net_scans = ct_ptr(self.t_net_scans)
        # deterministic mode always uses the ping-pong kernel: its fixed
        # wave schedule + strict-termination semantics are the validated
        # bit-identical path (checkpoint-resume test)
        a.use_calendar = 1 if (not self.deterministic and
                               (getattr(self, "use_calendar", False) or
                                _os.environ.get("PNR_CALENDAR"))) else 0
        a.partial = 1 if getattr(self, "_partial", False) else 0
        a.inq_base = ct_ptr(self.t_inq)
        self._args_keepalive = (t_sink_rr, t_crit, t_sink_orig, q_small,
                                q_large)
        return a


def ct_ptr(t):
    return hip_api.ct.c_void_p(t.data_ptr())


def pathfinder_route_gpu(netlist, placement, g, arch, sta=None, max_iters=60,
                         pres_fac_init=0.5, pres_fac_mult=1.3, acc_fac=1.0,
                         astar_fac=1.2, verbose=False, device="cuda:0",
                         rip_up_always=False, deterministic=False,
                         bb_factor=4, crit_exp=1.0, max_criticality=0.99,
                         incremental=False):
    """GPU PathFinder outer loop — mirrors route.router.pathfinder_route.

    bb_factor: initial per-net bounding-box margin in tiles (reference:
    -bb_factor route option); grown automatically on route failure."""
    net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(
        netlist, placement, g, arch)
    from .router import ConnMap
    router = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr,
                       device=device, astar_fac=astar_fac,
                       deterministic=deterministic, bb_margin=bb_factor)
    n_rsinks = len(sink_rr)
    cmap = ConnMap(conn_index, sink_ptr, netlist.num_conns, n_rsinks)
    crit = np.zeros(n_rsinks, dtype=np.float32)
    conn_delay = np.zeros(netlist.num_conns, dtype=np.float32)
    intra_delay = float(arch.T_opin + arch.T_ipin)
    pres_fac = 0.0   # first iteration: congestion-blind (VPR style)
    cpd = 0.0
    history = []
    it = 0
    overused = -1
    prev_overused = 1 << 30
    for it in range(1, max_iters + 1):
        # selective reroute (reference: build_phase_two congested-only):
        # iteration 1 routes everything; later iterations re-route only
        # nets whose trees touch overused nodes
        subset = None
        # incremental mode: partial-rip selective iterations with a FULL
        # rip of the same active set every 2nd iteration (timing refresh).
        # After iteration 1 only congested nets are ever rerouted — a
        # concurrent rip-all reroute at high pres_fac is a limit cycle
        # (see parallel.dist.pathfinder_route_dist).
        resync = incremental and it > 2 and (it - 2) % 2 == 0
        if it > 1 and not rip_up_always:
            subset = router.congested_nets()
            inc = router.incomplete_nets()
            if len(inc):
                subset = np.union1d(subset, inc)
            if len(subset) == 0:
                subset = None
        router.reset_search_stats()
        # Stall breaker: concurrent selective reroute can limit-cycle on
        # the last contested nodes (snapshot collisions — nets re-divert
        # onto each other forever; seen at LU32: 22 overused for 60
        # iterations). When overused stops improving and the active set
        # is small, force the bb-disjoint wave schedule, whose in-order
        # congestion views resolve the tail.
        force_waves = (it >= 4 and 0 < overused and
                       overused > 0.85 * prev_overused and
                       subset is not None and len(subset) <= 2048)
        prev_overused = overused if overused > 0 else prev_overused
        overused, sink_delays = router.route_iteration(
            crit, pres_fac, net_subset=subset,
            partial=incremental and subset is not None and not resync,
            force_waves=force_waves)
        st = router.search_stats()
        history.append(dict(iter=it, overused=int(overused), cpd=cpd,
                            rounds=st["rounds"], scanned=st["scanned"],
                            rerouted=len(subset) if subset is not None
                            else router.num_nets))
        if verbose:
            print(f"[gpu] iter {it}: overused={overused} cpd={cpd*1e9:.2f}ns")
        import os as _os
        if _os.environ.get("PNR_CHECK_OCC"):
            # reference: occ==recalc cross-check EVERY iteration
            # (partitioning_multi_sink...:6194) — catches lost congestion
            # updates; off by default (costs one kernel + compare)
            if not router.check_occ_recount():
                raise RuntimeError(f"occ recount mismatch at iteration {it}")
        if sta is not None:
            cmap.conn_delays(sink_delays, out=conn_delay, fill=intra_delay)
            cpd, slack, c = sta.analyze(conn_delay)
            crit = cmap.sink_crit(c, max_crit=max_criticality,
                                  crit_exp=crit_exp)
        if overused == 0:
            break
        pres_fac = pres_fac_init if pres_fac == 0.0 else pres_fac * pres_fac_mult
        router.update_acc(acc_fac)

    ok = overused == 0 and not router.incomplete.any()
    if ok and not router.check_occ_recount():
        raise RuntimeError("GPU route: occ recount mismatch")
    return RouteResult(success=ok, iterations=it, overused=int(overused),
                       wirelength=router.wirelength(), crit_path_delay=cpd,
                       stats={"history": history,
                              "num_routed_nets": len(net_ids)},
                       router=router)
