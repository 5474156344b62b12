// Shared device-side types for the CDNA4 kernels. C ABI: every launcher is
// extern "C", takes raw device pointers (uploaded torch tensors) + a
// hipStream_t, and is called from Python via ctypes.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace pnrh {

// RR graph, SoA, device-resident (mirrors csrc/cpu/pnr.h RRGraph).
struct RRDev {
  const int8_t* type;
  const int16_t* xlow;
  const int16_t* ylow;
  const int16_t* xhigh;
  const int16_t* yhigh;
  const int16_t* capacity;
  const float* R;
  const float* C;
  const int32_t* row_ptr;    // int32: num_edges < 2^31 enforced on host
  const int32_t* edge_dst;
  const int8_t* edge_sw;
  const float* sw_R;         // [4]
  const float* sw_Tdel;      // [4]
  const float* base_cost;    // [6]
  const int32_t* idx_in_tile;  // dense per-anchor-tile node index
  int32_t num_nodes;
  int32_t nx, ny, L;
  int32_t npt;               // max nodes anchored per tile
};

// Net tables (device)
struct NetsDev {
  const int32_t* src;        // SOURCE rr node per net
  const int32_t* sink_ptr;   // [num_nets+1]
  const int32_t* sink_rr;    // SINK rr nodes, ordered by routing priority
  const float* crit;         // per sink (aligned with sink_rr)
  const int32_t* sink_orig;  // original sink index per ordered sink (stable
                             // across per-iteration criticality reorders)
  const int16_t* bb;         // [num_nets][4]: x0,y0,x1,y1 (tile bb incl. margin)
  int32_t num_nets;
};

// Persistent route trees (device, rebuilt every iteration)
struct TreesDev {
  const int64_t* off;        // [num_nets+1] capacity offsets
  int32_t* node;             // tree node ids
  int32_t* parent;           // index into the net's tree slice
  int8_t* sw;
  float* delay;              // source->node delay
  int32_t* len;              // [num_nets]
  float* sink_delay;         // per sink (aligned with sink_rr)
};

// Per-slot search scratch
struct SlotsDev {
  uint64_t* state;           // packed (back_cost_bits<<32 | prev) per local idx
  uint8_t* slot_class;       // unused for now
  int32_t n_small, n_large;
  int64_t small_cap;         // state entries per small slot
  int64_t large_cap;         // state entries per large slot (full chip)
  // frontier ping-pong buffers, per slot
  float4* frontier;          // 2 buffers per slot, each f_cap entries
  int64_t f_cap_small;
  int64_t f_cap_large;
};

struct RouteParams {
  float astar_fac;
  float pres_fac;
  float seg_delay, ipin_delay, seg_base, ipin_base;
  float delta_fac;           // bucket width in edge-step cost units
  float cong_mult;           // lookahead congestion-cost scale (mean used-
                             // wire cost); keeps A* focused when pres/acc
                             // inflate real edge costs far above base
  int32_t max_rounds;        // safety bound on delta-stepping rounds
  int32_t strict_term;       // deterministic mode: process the == bucket
};

}  // namespace pnrh
