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
    clb_in: int = 40      # CLB input pins (≡ SINK capacity)
    clb_out: int = 10     # CLB output pins (≡ SOURCE capacity)
    io_cap: int = 8       # IO slots per perimeter tile

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

    # Congestion base costs by rr type (SOURCE,SINK,OPIN,IPIN,CHANX,CHANY)
    # (reference: rr_graph_indexed_data.c base costs)
    def base_costs(self):
        return [0.0, 0.0, 1.0, 0.95, 1.0, 1.0]

    def num_clb_tiles(self):
        return self.nx * self.ny

    def num_io_slots(self):
        return 2 * (self.nx + self.ny) * self.io_cap

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
}


def get_arch(name: str) -> ArchDef:
    if name in BUILTIN_ARCHES:
        return BUILTIN_ARCHES[name]
    raise KeyError(f"unknown builtin arch {name!r}; have {list(BUILTIN_ARCHES)}")
