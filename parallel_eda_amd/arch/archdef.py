"""Architecture / device model for the island-style FPGA fabric.

Semantics modeled on the reference's libarchfpga types
(libarchfpga/include/physical_types.h:496-805, read_xml_arch_file.c:2528
XmlReadArch) but re-designed as a flat parameter set: the engine targets
segmented unidirectional island-style fabrics, which is what every BASELINE
config (k4_N4_90nm, k6_frac_N10_mem32K_40nm, stratixiv-like) reduces to for
place-and-route purposes.

Grid convention:
  Logic tiles at (x,y) for x in 1..nx, y in 1..ny.
  IO tiles on the perimeter: x=0 / x=nx+1 (y in 1..ny) and y=0 / y=ny+1
  (x in 1..nx), each holding `io_cap` IO slots. Corners unused.
  CHANX channel y (0..ny) spans x in 1..nx, between logic rows y and y+1.
  CHANY channel x (0..nx) spans y in 1..ny.
  Switch block SB(i,j) (i in 0..nx, j in 0..ny) is the crossing of
  CHANX y=j and CHANY x=i.
"""
from dataclasses import dataclass, asdict


# Block type ids (placement + netlist)
BLK_IO = 0
BLK_CLB = 1
BLK_RAM = 2   # memory-block column tiles (stratixiv M9K / mem32K class)
BLK_DSP = 3   # DSP/multiplier column tiles

BLK_NAMES = ["io", "clb", "ram", "dsp"]

# RR node types (must match csrc/cpu/pnr_types.h)
RR_SOURCE = 0
RR_SINK = 1
RR_OPIN = 2
RR_IPIN = 3
RR_CHANX = 4
RR_CHANY = 5

RR_TYPE_NAMES = ["SOURCE", "SINK", "OPIN", "IPIN", "CHANX", "CHANY"]


@dataclass
class ArchDef:
    """Flat device description consumed by the rr-graph builder."""
    name: str = "k6_n10"
    nx: int = 30
    ny: int = 30
    W: int = 64           # channel width (tracks per channel); even
    L: int = 4            # wire segment length in tiles
    fc_in: int = 8        # tracks each IPIN connects to (absolute count)
    fc_out: int = 8       # wire-starts each OPIN can drive (absolute count)
    sb_turn_fanin: int = 1  # in-wires per TURN side at each wire's driver
                            # mux (1 reproduces a pure permutation network,
                            # whose orbit structure leaves ~11% of nets
                            # locally unreachable — see rr_build.cpp)
    w_l1: int = -1        # tracks that are LENGTH-1 wires (-1 = auto W/8).
                          # A single-length unidir fabric moves in strides
                          # of exactly L, confining every route to one
                          # (mod L, mod L) SB sublattice; the length mix
                          # restores bb-local reachability (rr_build.cpp)
    clb_in: int = 40      # CLB input pins (≡ SINK capacity)
    clb_out: int = 10     # CLB output pins (≡ SOURCE capacity)
    clb_n_ble: int = 10   # BLEs per CLB (packer cluster size, arch <pb_type num_pb>)
    io_cap: int = 8       # IO slots per perimeter tile

    # Heterogeneous column tiles (reference: libarchfpga grid types with
    # column-repeat fill patterns, physical_types.h grid_loc_def; stratixiv /
    # k6_frac_N10_mem32K arches place RAM / DSP hard blocks in dedicated
    # columns). 0 = no such columns (homogeneous CLB fabric).
    # Column assignment (mirrored in csrc/cpu/rr_build.cpp col_btype):
    #   RAM  columns: x % ram_col_every == min(2, ram_col_every - 1)
    #   DSP  columns: x % dsp_col_every == min(5, dsp_col_every - 1),
    #                 RAM takes precedence on collision.
    ram_col_every: int = 0
    dsp_col_every: int = 0
    ram_in: int = 32      # RAM block input pins
    ram_out: int = 32     # RAM block output pins
    dsp_in: int = 36
    dsp_out: int = 18

    # Timing (seconds / ohms / farads); values in the range of 40nm arches.
    R_wire: float = 101.0      # ohm per tile of wire
    C_wire: float = 22.5e-15   # farad per tile of wire
    R_sw: float = 551.0        # switch-block mux output buffer R
    C_sw_in: float = 0.77e-15  # switch input C
    T_sw: float = 58e-12       # switch-block switch intrinsic delay
    T_opin: float = 60e-12     # OPIN output buffer delay
    T_ipin: float = 95e-12     # IPIN connection-block mux delay
    T_clb: float = 261e-12     # CLB combinational (in->out) delay
    T_seq_out: float = 124e-12 # clock-to-Q
    T_seq_in: float = 66e-12   # setup
    T_ram: float = 1.5e-9      # RAM block delay (seq: informational)
    T_dsp: float = 1.2e-9      # DSP block combinational delay

    # Congestion base costs by rr type (SOURCE,SINK,OPIN,IPIN,CHANX,CHANY)
    # (reference: rr_graph_indexed_data.c base costs)
    def base_costs(self):
        return [0.0, 0.0, 1.0, 0.95, 1.0, 1.0]

    def num_clb_tiles(self):
        return self.nx * self.ny

    def num_io_slots(self):
        return 2 * (self.nx + self.ny) * self.io_cap

    # ---- heterogeneous-tile helpers ----
    def is_heterogeneous(self):
        return self.ram_col_every > 0 or self.dsp_col_every > 0

    def col_block_type(self, x: int) -> int:
        """Block type of logic column x (1..nx). Mirrors rr_build.cpp."""
        if self.ram_col_every > 0 and \
                x % self.ram_col_every == min(2, self.ram_col_every - 1):
            return BLK_RAM
        if self.dsp_col_every > 0 and \
                x % self.dsp_col_every == min(5, self.dsp_col_every - 1):
            return BLK_DSP
        return BLK_CLB

    def pins_of(self, btype: int):
        """(n_in, n_out) pin counts of a block type."""
        if btype == BLK_CLB:
            return self.clb_in, self.clb_out
        if btype == BLK_RAM:
            return self.ram_in, self.ram_out
        if btype == BLK_DSP:
            return self.dsp_in, self.dsp_out
        return self.io_cap, self.io_cap  # IO tile: io_cap 1-pin slots

    def block_delay_of(self, btype: int) -> float:
        """Combinational propagation delay of a block type (STA)."""
        if btype == BLK_RAM:
            return self.T_ram
        if btype == BLK_DSP:
            return self.T_dsp
        return self.T_clb

    def tile_btype_grid(self):
        """(nx+2)*(ny+2) int8 grid of tile block types, x-major
        (index x*(ny+2)+y, matching RRGraph.tile_id). -1 = unusable corner."""
        import numpy as np
        gx, gy = self.nx + 2, self.ny + 2
        g = np.full(gx * gy, -1, dtype=np.int8)
        for x in range(1, self.nx + 1):
            t = self.col_block_type(x)
            g[x * gy + 1: x * gy + self.ny + 1] = t
            g[x * gy + 0] = BLK_IO
            g[x * gy + self.ny + 1] = BLK_IO
        for y in range(1, self.ny + 1):
            g[0 * gy + y] = BLK_IO
            g[(self.nx + 1) * gy + y] = BLK_IO
        return g

    def num_tiles_of_type(self, btype: int) -> int:
        if btype == BLK_IO:
            return 2 * (self.nx + self.ny)
        return sum(1 for x in range(1, self.nx + 1)
                   if self.col_block_type(x) == btype) * self.ny

    def to_dict(self):
        return asdict(self)


def _scaled(name, nx, ny, W, **kw):
    d = dict(name=name, nx=nx, ny=ny, W=W)
    d.update(kw)
    return ArchDef(**d)


# Named config scale ladder (BASELINE.json configs). Sizes approximate the
# named circuits' device footprints; netlists are synthetic (no network).
BUILTIN_ARCHES = {
    # tseng on k4_N4_90nm: ~1k CLBs
    "tseng": _scaled("tseng_k4n4", 12, 12, 24, L=1, clb_in=16, clb_out=4,
                     fc_in=6, fc_out=6, io_cap=2),
    # unit-test tiny arch
    "tiny": _scaled("tiny", 4, 4, 12, L=2, clb_in=6, clb_out=2,
                    fc_in=4, fc_out=4, io_cap=2),
    # stereovision2 on k6_frac_N10: ~30x30 ... real is ~90x90
    "stereovision2": _scaled("sv2_k6n10", 90, 90, 100),
    # LU32PEEng: ~120x120 grid class
    "LU32PEEng": _scaled("lu32_k6n10", 120, 120, 160),
    # bgm: ~90x90
    "bgm": _scaled("bgm_k6n10", 100, 100, 120),
    # Titan bitcoin_miner on stratixiv-like fabric: big grid, wide channels
    "bitcoin_miner": _scaled("bitcoin_stratixiv", 280, 200, 300, L=4,
                             clb_in=52, clb_out=20, fc_in=12, fc_out=12,
                             io_cap=16),
    # --- heterogeneous fabrics (RAM/DSP column tiles) ---
    # unit-test het arch: RAM cols x=2,6; DSP col x=5 (nx=8)
    "tiny_het": _scaled("tiny_het", 8, 6, 16, L=2, clb_in=6, clb_out=2,
                        fc_in=4, fc_out=4, io_cap=2,
                        ram_col_every=4, dsp_col_every=8,
                        ram_in=8, ram_out=4, dsp_in=8, dsp_out=4),
    # k6_frac_N10_mem32K_40nm-like: memory column every 8
    "mem32K": _scaled("k6n10_mem32K", 40, 40, 80, ram_col_every=8,
                      ram_in=40, ram_out=32),
    # stratixiv-like with M9K RAM columns + DSP columns at bitcoin scale
    "bitcoin_miner_het": _scaled("bitcoin_stratixiv_het", 280, 200, 300, L=4,
                                 clb_in=52, clb_out=20, fc_in=12, fc_out=12,
                                 io_cap=16, ram_col_every=8, dsp_col_every=16,
                                 ram_in=64, ram_out=32, dsp_in=72, dsp_out=36),
}


def get_arch(name: str) -> ArchDef:
    if name in BUILTIN_ARCHES:
        return BUILTIN_ARCHES[name]
    raise KeyError(f"unknown builtin arch {name!r}; have {list(BUILTIN_ARCHES)}")
