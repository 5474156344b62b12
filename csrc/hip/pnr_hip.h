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
  int32_t partial;           // partial rip-up: keep clean subtrees, skip
                             // still-connected sinks (reference:
                             // route_tree_mark_congested_...; CPU oracle
                             // route_net_incremental)
};

// ---- shared device helpers (router kernels) ----
#if defined(__HIP_DEVICE_COMPILE__) || defined(__HIPCC__)
__device__ __forceinline__ uint32_t f32_bits(float f) {
  // order-preserving bits for non-negative floats
  return __float_as_uint(f);
}
__device__ __forceinline__ float bits_f32(uint32_t u) { return __uint_as_float(u); }

__device__ __forceinline__ uint64_t pack_state(float back, int32_t prev) {
  return ((uint64_t)f32_bits(back) << 32) | (uint32_t)prev;
}

// state[] is updated with device-scope atomicMin, which executes at L2 and
// BYPASSES the CU's vector L1. A plain load can hit a stale L1 line (e.g.
// the INF written by the touched-list reset) and miss the atomic's value —
// the backtrack then extracts prev = -1 and wild-walks (the intermittent
// memory faults of profiles/README.md's fault-hunt log). Every read of
// state[] therefore goes through an agent-scope atomic load (L1-bypassing).
__device__ __forceinline__ uint64_t load_state(const uint64_t* p) {
  return __hip_atomic_load((const unsigned long long*)p, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
}

struct SinkCtx {
  int32_t sink_node;
  int16_t sx, sy;
  float crit;
  float astar_fac;
};

__device__ __forceinline__ float expected_cost(const RRDev& g, const RouteParams& P,
                                               int32_t v, const SinkCtx& S) {
  int8_t ty = g.type[v];
  if (ty == 1 /*SINK*/) return 0.0f;
  int tx = S.sx, ty2 = S.sy;
  int dx = 0, dy = 0;
  int xl = g.xlow[v], xh = g.xhigh[v], yl = g.ylow[v], yh = g.yhigh[v];
  if (xl > tx) dx = xl - tx; else if (xh < tx) dx = tx - xh;
  if (yl > ty2) dy = yl - ty2; else if (yh < ty2) dy = ty2 - yh;
  int dist = dx + dy;
  int nseg = (dist + g.L - 1) / g.L;
  float cong = nseg * P.seg_base * P.cong_mult + P.ipin_base;
  float del = nseg * P.seg_delay + P.ipin_delay;
  return S.crit * del + (1.0f - S.crit) * cong;
}

// congestion cost of entering node v (pres computed from occ on the fly;
// semantics of congestion.cxx:296 update_one_cost_internal's pres formula)
__device__ __forceinline__ float cong_cost(const RRDev& g, const RouteParams& P,
                                           const int32_t* occ, const float* acc,
                                           int32_t v) {
  int over = occ[v] + 1 - g.capacity[v];
  float pres = over > 0 ? 1.0f + over * P.pres_fac : 1.0f;
  return g.base_cost[g.type[v]] * acc[v] * pres;
}

__device__ __forceinline__ float hop_delay(const RRDev& g, int8_t sw, int32_t v) {
  return g.sw_Tdel[sw] + g.C[v] * (g.sw_R[sw] + 0.5f * g.R[v]);
}

struct LocalIdx {
  // small class: dense bb-local index; large class: global node id
  int bx0, by0, bw, bh, npt;
  bool dense;
  __device__ __forceinline__ int64_t operator()(const RRDev& g, int32_t v) const {
    if (!dense) return v;
    int tx = g.xlow[v] - bx0;
    int ty = g.ylow[v] - by0;
    if (tx < 0 || ty < 0 || tx >= bw || ty >= bh) return -1;
    return ((int64_t)tx * bh + ty) * npt + g.idx_in_tile[v];
  }
  __device__ __forceinline__ bool in_bb(const RRDev& g, int32_t v) const {
    int tx = g.xlow[v] - bx0;
    int ty = g.ylow[v] - by0;
    return tx >= 0 && ty >= 0 && tx < bw && ty < bh;
  }
};

#endif  // device helpers

}  // namespace pnrh

// C-ABI launch-argument block shared by the router kernel TUs and the
// ctypes mirror in parallel_eda_amd/ops/hip_api.py (layouts must match).
struct RouteLaunchArgs {
  // RRDev
  const int8_t* type; const int16_t* xlow; const int16_t* ylow;
  const int16_t* xhigh; const int16_t* yhigh; const int16_t* capacity;
  const float* R; const float* C;
  const int32_t* row_ptr; const int32_t* edge_dst; const int8_t* edge_sw;
  const float* sw_R; const float* sw_Tdel; const float* base_cost;
  const int32_t* idx_in_tile;
  int32_t num_nodes, nx, ny, L, npt;
  // NetsDev
  const int32_t* net_src; const int32_t* sink_ptr; const int32_t* sink_rr;
  const float* crit; const int32_t* sink_orig; const int16_t* bb;
  int32_t num_nets;
  // TreesDev
  const int64_t* tree_off; int32_t* tree_node; int32_t* tree_parent;
  int8_t* tree_sw; float* tree_delay; int32_t* tree_len; float* sink_delay;
  // params
  float astar_fac, pres_fac, seg_delay, ipin_delay, seg_base, ipin_base;
  float delta_fac;
  float cong_mult;
  int32_t max_rounds;
  int32_t strict_term;
  // queues
  const int32_t* queue_small; int32_t n_queue_small;
  const int32_t* queue_large; int32_t n_queue_large;
  int32_t* q_cursors;
  int32_t* occ; const float* acc;
  uint64_t* state_base; int64_t small_cap; int64_t large_cap;
  int32_t n_small_slots; int32_t n_large_slots;
  float4* frontier_base; int64_t f_cap_small; int64_t f_cap_large;
  int32_t* touched_base; int64_t t_cap_small; int64_t t_cap_large;
  int32_t* fail_flags;
  unsigned long long* stats;   // [8] search counters or null
  unsigned long long* net_scans;  // per-net scan counters or null
  int32_t use_calendar;        // EXPERIMENTAL: calendar-queue frontier
  int32_t partial;             // partial rip-up (see RouteParams)
  uint8_t* inq_base;           // per-state-entry in-queue flag (same slot
                               // layout as state_base; frontier dedup)
};
