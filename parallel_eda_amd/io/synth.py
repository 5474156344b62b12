"""Synthetic netlist generation.

There is no network access for MCNC/VTR/Titan benchmark files, so the named
BASELINE configs run on synthetic netlists of the named size: random
Rent's-rule-flavored netlists with locality, acyclic combinational structure
(sequential blocks break cycles), generated deterministically from a seed.

The generated object is the post-packing netlist the reference reads back
with read_netlist.c — block-level: IO and CLB blocks, nets with one driver
and >=1 sinks.
"""
from dataclasses import dataclass

import numpy as np

from ..arch.archdef import ArchDef, BLK_IO, BLK_CLB, BLK_RAM, BLK_DSP
from .. import ops


@dataclass
class SynthSpec:
    n_clb: int
    n_in: int            # input pads
    n_out: int           # output pads
    avg_fanout: float = 3.0
    seq_frac: float = 0.35   # fraction of CLBs that are sequential
    max_fanin: int = 16      # block input-pin budget (arch.clb_in)
    seed: int = 1
    # heterogeneous blocks (RAM columns / DSP columns); RAM blocks are
    # sequential (registered outputs), DSP blocks combinational.
    n_ram: int = 0
    n_dsp: int = 0
    ram_fanin: int = 32
    dsp_fanin: int = 36


def spec_for_arch(arch: ArchDef, fill: float = 0.85, seed: int = 1) -> SynthSpec:
    n_clb = int(arch.num_tiles_of_type(BLK_CLB) * fill)
    n_ram = int(arch.num_tiles_of_type(BLK_RAM) * fill)
    n_dsp = int(arch.num_tiles_of_type(BLK_DSP) * fill)
    n_io = max(4, int(0.12 * n_clb))
    n_io = min(n_io, arch.num_io_slots() // 2 - 2)
    return SynthSpec(n_clb=n_clb, n_in=max(2, n_io // 2),
                     n_out=max(2, n_io // 2), max_fanin=arch.clb_in,
                     seed=seed, n_ram=n_ram, n_dsp=n_dsp,
                     ram_fanin=arch.ram_in, dsp_fanin=arch.dsp_in)


class NetlistPy:
    """Python-side netlist holding numpy arrays + the C++ handle."""

    def __init__(self, block_type, block_is_seq, net_driver, net_sink_ptr,
                 net_sinks, names=None):
        self.block_type = np.asarray(block_type, dtype=np.int8)
        self.block_is_seq = np.asarray(block_is_seq, dtype=np.uint8)
        self.net_driver = np.asarray(net_driver, dtype=np.int32)
        self.net_sink_ptr = np.asarray(net_sink_ptr, dtype=np.int64)
        self.net_sinks = np.asarray(net_sinks, dtype=np.int32)
        self.names = names
        self.block_clock = None   # optional per-block clock domain ids
        self.clock_names = []     # domain id -> clock name
        self.macros = []          # optional carry-chain macro groups
        self._cpp = None

    @property
    def num_blocks(self):
        return len(self.block_type)

    @property
    def num_nets(self):
        return len(self.net_driver)

    @property
    def num_conns(self):
        return len(self.net_sinks)

    def cpp(self):
        if self._cpp is None:
            self._cpp = ops.cpu().Netlist(
                self.block_type, self.block_is_seq, self.net_driver,
                self.net_sink_ptr, self.net_sinks)
        return self._cpp

    def fanout(self, n):
        return self.net_sink_ptr[n + 1] - self.net_sink_ptr[n]


def synth_placed_netlist(arch: ArchDef, fill: float = 0.85,
                         avg_fanout: float = 3.0, locality: float = 6.0,
                         seq_frac: float = 0.35, seed: int = 1):
    """Generate a netlist WITH a placement, nets geometrically local.

    For large configs (LU32PEEng/bgm/bitcoin_miner scale) a full SA anneal
    as benchmark *setup* would dominate; real routing benchmarks start from
    a quality placement, so we synthesize netlist+placement jointly: blocks
    sit on the grid and each net connects a driver to sinks drawn from a
    local neighborhood (geometric decay, Rent-flavored long tail).
    Returns (NetlistPy, Placement-like object with x/y/slot arrays).
    """
    from ..place.placer import Placement
    rng = np.random.default_rng(seed)
    nx, ny, io_cap = arch.nx, arch.ny, arch.io_cap
    # per-type logic tiles (x-major flat index x_idx*ny + y_idx, matching
    # the homogeneous generator's tile ordering)
    col_t = np.asarray([arch.col_block_type(x) for x in range(1, nx + 1)])
    type_tiles = {}
    for t in (BLK_CLB, BLK_RAM, BLK_DSP):
        cols = np.nonzero(col_t == t)[0]          # 0-based column indices
        if len(cols):
            type_tiles[t] = (cols[:, None] * ny + np.arange(ny)[None, :]).ravel()
    n_of = {t: int(len(v) * fill) for t, v in type_tiles.items()}
    n_clb = n_of.get(BLK_CLB, 0)
    n_ram = n_of.get(BLK_RAM, 0)
    n_dsp = n_of.get(BLK_DSP, 0)
    n_logic = n_clb + n_ram + n_dsp
    n_io_pairs = max(2, min(int(0.06 * n_clb), (nx + ny) * io_cap // 2 - 2))
    n_in = n_out = n_io_pairs
    n_io = n_in + n_out
    nb = n_io + n_logic

    block_type = np.full(nb, BLK_CLB, dtype=np.int8)
    block_type[:n_io] = BLK_IO
    block_is_seq = np.zeros(nb, dtype=np.uint8)
    block_is_seq[:n_io] = 1
    clb0 = n_io
    ram0 = clb0 + n_clb
    dsp0 = ram0 + n_ram
    block_type[ram0:dsp0] = BLK_RAM
    block_type[dsp0:] = BLK_DSP
    block_is_seq[clb0:ram0] = (rng.random(n_clb) < seq_frac).astype(np.uint8)
    block_is_seq[ram0:dsp0] = 1   # RAM: registered outputs
    # DSP blocks combinational (rank-ordered below)

    # placement: logic blocks into random distinct tiles of their type;
    # IOs into perimeter slots
    parts = []
    for t in (BLK_CLB, BLK_RAM, BLK_DSP):
        if n_of.get(t, 0):
            pick = rng.choice(len(type_tiles[t]), size=n_of[t], replace=False)
            parts.append(type_tiles[t][pick])
    logic_tiles = np.concatenate(parts) if parts else np.empty(0, np.int64)
    cx = (logic_tiles // ny + 1).astype(np.int32)
    cy = (logic_tiles % ny + 1).astype(np.int32)
    io_locs = ([(0, y) for y in range(1, ny + 1)] +
               [(nx + 1, y) for y in range(1, ny + 1)] +
               [(x, 0) for x in range(1, nx + 1)] +
               [(x, ny + 1) for x in range(1, nx + 1)])
    io_slots = [(x, y, s) for (x, y) in io_locs for s in range(io_cap)]
    sel = rng.choice(len(io_slots), size=n_io, replace=False)
    bx = np.zeros(nb, dtype=np.int32)
    by = np.zeros(nb, dtype=np.int32)
    bslot = np.zeros(nb, dtype=np.int32)
    for i, k in enumerate(sel):
        x, y, s = io_slots[k]
        bx[i], by[i], bslot[i] = x, y, s
    bx[clb0:], by[clb0:] = cx, cy

    # nets: every input pad + every logic block drives one net
    drivers = np.concatenate([
        np.arange(n_in, dtype=np.int32),
        np.arange(clb0, nb, dtype=np.int32)])
    n_nets = len(drivers)
    rank = rng.permutation(n_logic)

    fanin = np.zeros(nb, dtype=np.int64)
    max_fanin = np.full(nb, arch.clb_in, dtype=np.int64)
    max_fanin[:n_io] = 1
    max_fanin[ram0:dsp0] = arch.ram_in
    max_fanin[dsp0:] = arch.dsp_in
    # KD-free locality: bucket logic blocks per tile-cell grid
    cell = max(2, int(locality))
    ncx = (nx + cell - 1) // cell
    ncy = (ny + cell - 1) // cell
    buckets = [[] for _ in range(ncx * ncy)]
    for i in range(n_logic):
        bxi = min((cx[i] - 1) // cell, ncx - 1)
        byi = min((cy[i] - 1) // cell, ncy - 1)
        buckets[bxi * ncy + byi].append(i)

    sink_lists = []
    out_pad_driven = np.zeros(n_out, dtype=bool)
    for drv in drivers:
        fanout = max(1, int(rng.poisson(avg_fanout)))
        if drv < n_in:
            dxx, dyy = 1 + rng.integers(nx), 1 + rng.integers(ny)
        else:
            dxx, dyy = bx[drv], by[drv]
        drv_is_comb = drv >= clb0 and not block_is_seq[drv]
        drv_rank = rank[drv - clb0] if drv_is_comb else -1
        sinks = set()
        for _ in range(fanout * 6):
            if len(sinks) >= fanout:
                break
            if rng.random() < 0.03 and n_out > 0:
                o = int(rng.integers(n_out))
                blk = n_in + o
                if out_pad_driven[o] or blk in sinks:
                    continue
                sinks.add(blk)
                out_pad_driven[o] = True
                fanin[blk] += 1
                continue
            # locality: pick a cell near the driver (geometric radius),
            # occasionally a uniform long-range sink
            if rng.random() < 0.08:
                c = int(rng.integers(n_logic))
            else:
                r = rng.geometric(1.0 / max(1.0, locality / cell))
                ang = rng.random() * 2 * np.pi
                tx = int(dxx + np.cos(ang) * r * cell)
                ty = int(dyy + np.sin(ang) * r * cell)
                bxi = min(max((tx - 1) // cell, 0), ncx - 1)
                byi = min(max((ty - 1) // cell, 0), ncy - 1)
                bl = buckets[bxi * ncy + byi]
                if not bl:
                    continue
                c = bl[int(rng.integers(len(bl)))]
            blk = clb0 + c
            if blk == drv or blk in sinks:
                continue
            if fanin[blk] >= max_fanin[blk]:
                continue
            if not block_is_seq[blk] and drv_rank >= 0 and rank[c] <= drv_rank:
                continue
            sinks.add(blk)
            fanin[blk] += 1
        if not sinks:
            for c in rng.permutation(n_logic)[:64]:
                blk = clb0 + int(c)
                if blk != drv and (block_is_seq[blk] or drv_rank < 0 or
                                   rank[int(c)] > drv_rank) \
                        and fanin[blk] < max_fanin[blk]:
                    sinks.add(blk)
                    fanin[blk] += 1
                    break
        if not sinks:
            o = int(np.argmin(out_pad_driven))
            sinks.add(n_in + o)
            out_pad_driven[o] = True
        sink_lists.append(sorted(sinks))
    for o in range(n_out):
        if not out_pad_driven[o]:
            n = int(rng.integers(n_nets))
            sink_lists[n].append(n_in + o)

    sink_ptr = np.zeros(n_nets + 1, dtype=np.int64)
    for i, s in enumerate(sink_lists):
        sink_ptr[i + 1] = sink_ptr[i] + len(s)
    net_sinks = np.concatenate([np.asarray(s, dtype=np.int32) for s in sink_lists])
    nl = NetlistPy(block_type, block_is_seq, drivers, sink_ptr, net_sinks)
    pl = Placement(bx, by, bslot)
    return nl, pl


def synth_netlist(spec: SynthSpec) -> NetlistPy:
    """Generate a deterministic synthetic netlist.

    Block ids: [0, n_in) input pads; [n_in, n_in+n_out) output pads;
    [n_in+n_out, ...) CLBs. Combinational edges always go from
    lower "rank" to higher rank among comb CLBs => acyclic.
    """
    rng = np.random.default_rng(spec.seed)
    n_io = spec.n_in + spec.n_out
    # logic blocks: [clb0, clb0+n_clb) CLB, then RAM, then DSP
    n_logic = spec.n_clb + spec.n_ram + spec.n_dsp
    nb = n_io + n_logic
    block_type = np.full(nb, BLK_CLB, dtype=np.int8)
    block_type[:n_io] = BLK_IO
    block_is_seq = np.zeros(nb, dtype=np.uint8)
    block_is_seq[:n_io] = 1  # pads are timing endpoints
    clb0 = n_io
    ram0 = clb0 + spec.n_clb
    dsp0 = ram0 + spec.n_ram
    block_type[ram0:dsp0] = BLK_RAM
    block_type[dsp0:] = BLK_DSP
    seq_mask = rng.random(spec.n_clb) < spec.seq_frac
    block_is_seq[clb0:ram0] = seq_mask.astype(np.uint8)
    block_is_seq[ram0:dsp0] = 1  # RAM: registered outputs
    # DSP blocks stay combinational (rank-ordered like comb CLBs)

    # Drivers: every input pad and every CLB drives exactly one net.
    drivers = np.concatenate([
        np.arange(spec.n_in, dtype=np.int32),                      # input pads
        np.arange(clb0, nb, dtype=np.int32),                       # CLBs
    ])
    n_nets = len(drivers)

    # Comb rank: random permutation of logic blocks; an edge into a COMB
    # block must come from a driver with lower rank (or a seq block / pad).
    rank = np.empty(n_logic, dtype=np.int64)
    rank[:] = rng.permutation(n_logic)

    # Sink candidate pools with block fan-in budgets: a CLB accepts at most
    # max_fanin incoming connections (it has that many input pins); an
    # output pad accepts exactly one driver.
    fanin = np.zeros(nb, dtype=np.int64)
    max_fanin = np.full(nb, spec.max_fanin, dtype=np.int64)
    max_fanin[:n_io] = 1  # pads: single pin
    max_fanin[ram0:dsp0] = spec.ram_fanin
    max_fanin[dsp0:] = spec.dsp_fanin
    sink_lists = []
    out_pad_driven = np.zeros(spec.n_out, dtype=bool)
    for i, drv in enumerate(drivers):
        fanout = max(1, int(rng.poisson(spec.avg_fanout)))
        sinks = set()
        drv_is_clb = drv >= clb0
        drv_rank = rank[drv - clb0] if (drv_is_clb and not block_is_seq[drv]) else -1
        for _ in range(fanout * 4):
            if len(sinks) >= fanout:
                break
            if rng.random() < 0.04 and spec.n_out > 0:
                # drive an output pad (one driver each)
                o = int(rng.integers(spec.n_out))
                blk = spec.n_in + o
                if out_pad_driven[o] or blk in sinks:
                    continue
                sinks.add(blk)
                out_pad_driven[o] = True
                fanin[blk] += 1
                continue
            c = int(rng.integers(n_logic))
            blk = clb0 + c
            if blk == drv or blk in sinks:
                continue
            if fanin[blk] >= max_fanin[blk]:
                continue
            if not block_is_seq[blk]:
                # comb sink: need drv_rank < rank[c] for acyclicity
                if drv_rank >= 0 and rank[c] <= drv_rank:
                    continue
            sinks.add(blk)
            fanin[blk] += 1
        if not sinks:
            # fall back: an undriven output pad, else any block with room
            # (seq, or comb with a legal rank)
            placed = False
            for o in range(spec.n_out):
                if not out_pad_driven[o]:
                    sinks.add(spec.n_in + o)
                    out_pad_driven[o] = True
                    fanin[spec.n_in + o] += 1
                    placed = True
                    break
            if not placed:
                for c in rng.permutation(n_logic):
                    blk = clb0 + int(c)
                    if blk == drv or fanin[blk] >= max_fanin[blk]:
                        continue
                    if not block_is_seq[blk] and drv_rank >= 0                             and rank[int(c)] <= drv_rank:
                        continue
                    sinks.add(blk)
                    fanin[blk] += 1
                    placed = True
                    break
            if not placed:
                # every fan-in budget saturated: the driver's output is
                # unused — drop the net (the reference's sweep removes
                # dangling nets the same way)
                sink_lists.append([])
                continue
        sink_lists.append(sorted(sinks))

    # drop empty (swept) nets
    keep = [i for i, sl in enumerate(sink_lists) if sl]
    if len(keep) != len(drivers):
        drivers = drivers[np.asarray(keep, dtype=np.int64)]
        sink_lists = [sink_lists[i] for i in keep]
        n_nets = len(drivers)

    # ensure every output pad is driven by exactly one net
    for o in range(spec.n_out):
        if not out_pad_driven[o]:
            n = int(rng.integers(n_nets))
            sink_lists[n].append(spec.n_in + o)
            out_pad_driven[o] = True
            fanin[spec.n_in + o] += 1

    sink_ptr = np.zeros(n_nets + 1, dtype=np.int64)
    for i, s in enumerate(sink_lists):
        sink_ptr[i + 1] = sink_ptr[i] + len(s)
    net_sinks = np.concatenate([np.asarray(s, dtype=np.int32) for s in sink_lists])

    # Every comb CLB must have at least one input (else it's dangling logic);
    # this holds with high probability — patch the stragglers.
    has_in = np.zeros(nb, dtype=bool)
    has_in[net_sinks] = True
    stragglers = [b for b in range(clb0, nb) if not has_in[b]]
    if stragglers:
        extra = []
        for b in stragglers:
            # drive from a random input pad's net
            n = int(rng.integers(spec.n_in)) if spec.n_in else 0
            extra.append((n, b))
        # rebuild CSR with extras appended
        per_net = [list(s) for s in sink_lists]
        for n, b in extra:
            per_net[n].append(b)
        sink_ptr = np.zeros(n_nets + 1, dtype=np.int64)
        for i, s in enumerate(per_net):
            sink_ptr[i + 1] = sink_ptr[i] + len(s)
        net_sinks = np.concatenate([np.asarray(s, dtype=np.int32) for s in per_net])

    return NetlistPy(block_type, block_is_seq, drivers, sink_ptr, net_sinks)
