// parallel_eda_amd — shared host-side types for the MI355X-native FPGA P&R engine.
//
// Everything is struct-of-arrays from the start: the same arrays that the CPU
// oracles walk are uploaded verbatim to HBM for the HIP kernels (coalesced
// per-lane access over node ids). This replaces the reference's
// pointer-per-node graph (vpr/SRC/parallel_route/new_rr_graph.h:10-63,
// graph.h) with a flat CSR layout.
#pragma once
#include <algorithm>
#include <cstdint>
#include <memory>
#include <vector>
#include <cmath>
#include <string>
#include <stdexcept>

namespace pnr {

// RR node types — must match parallel_eda_amd/arch/archdef.py
enum RRType : int8_t {
  SOURCE = 0,
  SINK = 1,
  OPIN = 2,
  IPIN = 3,
  CHANX = 4,
  CHANY = 5,
};

// Switch ids
enum SwitchId : int8_t {
  SW_ZERO = 0,   // SOURCE->OPIN, IPIN->SINK (tiny delay, no R)
  SW_SB = 1,     // switch-block buffered mux
  SW_OPIN = 2,   // OPIN output buffer into wire start mux
  SW_IPIN = 3,   // connection-block mux into IPIN
};
constexpr int NUM_SWITCHES = 4;

struct ArchParams {
  int nx, ny, W, L;
  int fc_in, fc_out;
  int sb_turn_fanin = 1;  // in-wires per turn side at each wire's driver mux
  int w_l1 = -1;          // tracks (of W) that are LENGTH-1 wires; -1 = auto
                          // (W/8 rounded to a pair). A pure single-length
                          // unidir fabric confines routing to a
                          // (mod L x mod L) switch-block sublattice — the
                          // length mix is what makes bb-local routing
                          // possible (real fabrics mix L1/L4/L16).
  int clb_in, clb_out;
  int io_cap;
  // heterogeneous column tiles (0 = none); column phases mirror
  // parallel_eda_amd/arch/archdef.py col_block_type
  int ram_col_every = 0, dsp_col_every = 0;
  int ram_in = 0, ram_out = 0, dsp_in = 0, dsp_out = 0;
  float R_wire, C_wire, R_sw, C_sw_in, T_sw, T_opin, T_ipin;
  float base_cost[6];
};

// Block type of logic column x (1..nx): 1=CLB 2=RAM 3=DSP.
inline int col_btype(const ArchParams& ap, int x) {
  if (ap.ram_col_every > 0 &&
      x % ap.ram_col_every == std::min(2, ap.ram_col_every - 1))
    return 2;
  if (ap.dsp_col_every > 0 &&
      x % ap.dsp_col_every == std::min(5, ap.dsp_col_every - 1))
    return 3;
  return 1;
}

struct RRGraph {
  int nx = 0, ny = 0, W = 0, L = 0;
  int num_nodes = 0;
  int64_t num_edges = 0;

  // Node SoA
  std::vector<int8_t> type;
  std::vector<int16_t> xlow, ylow, xhigh, yhigh;
  std::vector<int16_t> ptc;       // track index / pin index
  std::vector<int16_t> capacity;
  std::vector<float> R, C;

  // CSR out-edges
  std::vector<int64_t> row_ptr;   // num_nodes+1
  std::vector<int32_t> edge_dst;
  std::vector<int8_t> edge_sw;

  // Switch table
  float sw_R[NUM_SWITCHES], sw_Cin[NUM_SWITCHES], sw_Tdel[NUM_SWITCHES];
  uint8_t sw_buffered[NUM_SWITCHES];

  float base_cost[6];

  // tile id = x*(ny+2)+y over the (nx+2)x(ny+2) grid
  std::vector<int32_t> tile_source;  // -1 if none
  std::vector<int32_t> tile_sink;

  int tile_id(int x, int y) const { return x * (ny + 2) + y; }
  int degree_max = 0;
};

// Built by rr_build.cpp
RRGraph build_rr_graph(const ArchParams& ap);

// ---------- Netlist (post-packing) ----------
// Blocks have a type (0=IO,1=CLB) and placement (x,y,slot).
// Nets: driver block + sink blocks. CSR over sinks.
struct Netlist {
  int num_blocks = 0;
  int num_nets = 0;
  std::vector<int8_t> block_type;
  std::vector<uint8_t> block_is_seq;   // sequential (FF/IO) endpoint?
  std::vector<int32_t> net_driver;     // block id per net
  std::vector<int64_t> net_sink_ptr;   // num_nets+1
  std::vector<int32_t> net_sinks;      // block ids
};

}  // namespace pnr
