// CDNA4 batched simulated-annealing placer.
//
// Re-designs the reference's serial try_swap loop (vpr/SRC/place/place.c:1252
// try_swap, update_bb:2292, get_net_cost:2204, comp_delta_td_cost) as
// batched parallel moves: each thread proposes one range-limited move/swap,
// evaluates the EXACT bb+timing delta against the frozen pre-batch state,
// runs Metropolis, then claims its touched nets + grid locations with
// atomicMin (move-index priority => deterministic winner set); a second
// kernel applies the conflict-free winners, whose deltas are exact and
// additive (disjoint nets/locations => sequential-equivalent batch).
#include "pnr_hip.h"

namespace pnrh {

struct PlaceDev {
  // netlist
  const int32_t* net_blk_ptr;   // [num_nets+1] CSR: net -> member blocks
  const int32_t* net_blks;      //   (driver first, then sinks; may repeat)
  const int32_t* blk_net_ptr;   // [num_blocks+1] CSR: block -> nets (deduped)
  const int32_t* blk_nets;
  const int8_t* blk_type;       // 0=IO 1=CLB 2=RAM 3=DSP
  const int8_t* tile_btype;     // [gx*gy] tile block type, -1 corner;
                                // nullptr => homogeneous (perimeter IO)
  const int32_t* type_cols;     // sorted column lists: RAM cols then DSP
  const int32_t* type_col_ptr;  // [3] bounds into type_cols (RAM, DSP)
  const uint8_t* fixed;         // per-block pin-down mask; nullptr = none
  const float* net_q;           // crossing factor per net
  // timing
  const int32_t* net_sink_ptr;  // [num_nets+1] conn ranges
  const float* conn_crit;       // per conn
  const float* delay_mat;       // [gx*gy] |dx|*gy+|dy|
  // placement state
  int32_t* bx;
  int32_t* by;
  int32_t* bslot;
  int32_t* grid;                // [(x*gy+y)*cap + slot] -> block or -1
  float* net_cost;              // current bb cost per net
  float* net_tcost;             // current timing cost per net
  int32_t num_blocks, num_nets, gx, gy, cap, nx, ny, io_cap;
  int32_t rx0, rx1;             // move region (column strip) for the
                                // distributed strip-sharded anneal
                                // (parallel/dist_place.py); rx0 < 0 = off
  // carry-chain macros (reference: place_macro.c): rigid block groups
  // moved as one. macro_of: [num_blocks] macro id or -1 (nullptr = no
  // macros); macro_ptr/macro_blk: CSR of member blocks.
  const int32_t* macro_of;
  const int32_t* macro_ptr;
  const int32_t* macro_blk;
};

struct MovesDev {
  int32_t* mv_blk;      // proposer's block
  int32_t* mv_to;       // encoded dest (x*gy+y)*cap+slot
  int32_t* mv_other;    // block swapped with (or -1)
  float* mv_dbb;        // exact bb delta
  float* mv_dtd;        // exact td delta
  uint8_t* mv_flags;    // 1 = Metropolis-accepted candidate; 2 = winner
  int32_t* net_claim;   // [num_nets] atomicMin move index
  int32_t* loc_claim;   // [gx*gy] atomicMin move index (tile granularity)
  int32_t* counters;    // [4]: attempts, accepted, winners, conflicts
  int32_t n_moves;
};

__device__ __forceinline__ uint32_t rng_hash(uint32_t a, uint32_t b, uint32_t c) {
  uint32_t h = a * 0x9E3779B9u ^ b * 0x85EBCA6Bu ^ c * 0xC2B2AE35u;
  h ^= h >> 16; h *= 0x7FEB352Du; h ^= h >> 15; h *= 0x846CA68Bu; h ^= h >> 16;
  return h;
}

__device__ __forceinline__ float cross_count_dev(int n) {
  const float q3 = 1.0f, q50 = 2.79f;
  if (n <= 3) return q3;
  if (n >= 50) return q50 + 0.02616f * (n - 50);
  return q3 + (q50 - q3) * (n - 3) / 47.0f;
}

__device__ __forceinline__ bool is_io_loc(const PlaceDev& p, int x, int y) {
  return x == 0 || x == p.gx - 1 || y == 0 || y == p.gy - 1;
}
__device__ __forceinline__ int cap_at(const PlaceDev& p, int x, int y) {
  bool io = is_io_loc(p, x, y);
  if (!io && x >= 1 && x <= p.nx && y >= 1 && y <= p.ny) return 1;
  if (io && ((x >= 1 && x <= p.nx) != (y >= 1 && y <= p.ny))) return p.io_cap;
  return 0;
}

// bb cost of net n with up to 2 positional overrides
__device__ float net_bb_cost(const PlaceDev& p, int n, int32_t b1, int x1,
                             int y1, int32_t b2, int x2, int y2) {
  int xmin = 1 << 28, xmax = -1, ymin = 1 << 28, ymax = -1;
  int32_t e0 = p.net_blk_ptr[n], e1 = p.net_blk_ptr[n + 1];
  for (int32_t e = e0; e < e1; ++e) {
    int32_t b = p.net_blks[e];
    int x = p.bx[b], y = p.by[b];
    if (b == b1) { x = x1; y = y1; }
    else if (b == b2) { x = x2; y = y2; }
    xmin = min(xmin, x); xmax = max(xmax, x);
    ymin = min(ymin, y); ymax = max(ymax, y);
  }
  return p.net_q[n] * ((xmax - xmin + 1) + (ymax - ymin + 1));
}

// bb cost of net n with every member of macro `mid` displaced (ddx,ddy)
__device__ float net_bb_cost_macro(const PlaceDev& p, int n, int mid,
                                   int ddx, int ddy) {
  int xmin = 1 << 28, xmax = -1, ymin = 1 << 28, ymax = -1;
  int32_t e0 = p.net_blk_ptr[n], e1 = p.net_blk_ptr[n + 1];
  for (int32_t e = e0; e < e1; ++e) {
    int32_t b = p.net_blks[e];
    int x = p.bx[b], y = p.by[b];
    if (p.macro_of[b] == mid) { x += ddx; y += ddy; }
    xmin = min(xmin, x); xmax = max(xmax, x);
    ymin = min(ymin, y); ymax = max(ymax, y);
  }
  return p.net_q[n] * ((xmax - xmin + 1) + (ymax - ymin + 1));
}

__device__ float net_td_cost_macro(const PlaceDev& p, int n, int mid,
                                   int ddx, int ddy) {
  if (p.delay_mat == nullptr) return 0.0f;
  int32_t e0 = p.net_blk_ptr[n], e1 = p.net_blk_ptr[n + 1];
  int32_t drv = p.net_blks[e0];
  int dx0 = p.bx[drv], dy0 = p.by[drv];
  if (p.macro_of[drv] == mid) { dx0 += ddx; dy0 += ddy; }
  float t = 0.0f;
  int32_t c0 = p.net_sink_ptr[n];
  for (int32_t e = e0 + 1; e < e1; ++e) {
    int32_t b = p.net_blks[e];
    int x = p.bx[b], y = p.by[b];
    if (p.macro_of[b] == mid) { x += ddx; y += ddy; }
    int dx = abs(x - dx0), dy = abs(y - dy0);
    t += p.conn_crit[c0 + (e - e0 - 1)] * p.delay_mat[dx * p.gy + dy];
  }
  return t;
}

// timing cost of net n (sum over conns of crit * delay) with overrides
__device__ float net_td_cost(const PlaceDev& p, int n, int32_t b1, int x1,
                             int y1, int32_t b2, int x2, int y2) {
  if (p.delay_mat == nullptr) return 0.0f;
  int32_t e0 = p.net_blk_ptr[n], e1 = p.net_blk_ptr[n + 1];
  int32_t drv = p.net_blks[e0];
  int dx0 = p.bx[drv], dy0 = p.by[drv];
  if (drv == b1) { dx0 = x1; dy0 = y1; }
  else if (drv == b2) { dx0 = x2; dy0 = y2; }
  float t = 0.0f;
  int32_t c0 = p.net_sink_ptr[n];
  for (int32_t e = e0 + 1; e < e1; ++e) {
    int32_t b = p.net_blks[e];
    int x = p.bx[b], y = p.by[b];
    if (b == b1) { x = x1; y = y1; }
    else if (b == b2) { x = x2; y = y2; }
    int dx = abs(x - dx0), dy = abs(y - dy0);
    t += p.conn_crit[c0 + (e - e0 - 1)] * p.delay_mat[dx * p.gy + dy];
  }
  return t;
}

#define MAX_MOVE_NETS 160
#define PROP_WAVES 4

__device__ __forceinline__ float wave_sum_f32(float v) {
  for (int off = 32; off; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

// binary search in block b's sorted net list
__device__ __forceinline__ bool blk_has_net(const PlaceDev& p, int32_t b,
                                            int32_t n) {
  int32_t lo = p.blk_net_ptr[b], hi = p.blk_net_ptr[b + 1];
  while (lo < hi) {
    int32_t mid2 = (lo + hi) >> 1;
    if (p.blk_nets[mid2] < n) lo = mid2 + 1; else hi = mid2;
  }
  return lo < p.blk_net_ptr[b + 1] && p.blk_nets[lo] == n;
}

__device__ int propose_select_impl(const PlaceDev& p, int rlim,
                                   uint32_t seed, uint32_t batch, int i,
                                   int& x1o, int& y1o, int& slot1o);

// One WAVEFRONT per proposal (the per-thread version left ~1 live
// thread per CU and propose was 52% of LU32 kernel time at 158 us per
// 554-move batch): lane 0 selects the candidate, the wave evaluates the
// affected nets' exact deltas in parallel and reduces them, and claims
// go out lane-strided (atomicMin is idempotent, so the two block lists
// need no dedup — only the delta sum skips shared nets).
__launch_bounds__(64 * PROP_WAVES, 2)
__global__ void place_propose_kernel(PlaceDev p, MovesDev m, float T,
                                     int rlim, float timing_tradeoff,
                                     float inv_bb_norm, float inv_td_norm,
                                     uint32_t seed, uint32_t batch) {
  const int lane = threadIdx.x & 63;
  const int i = blockIdx.x * PROP_WAVES + (threadIdx.x >> 6);
  if (i >= m.n_moves) return;
  int sel = -1, sx1 = -1, sy1 = -1, sslot = 0;
  if (lane == 0) {
    m.mv_flags[i] = 0;
    sel = propose_select_impl(p, rlim, seed, batch, i, sx1, sy1, sslot);
  }
  sel = __shfl(sel, 0);
  if (sel < 0) return;
  const int x1 = __shfl(sx1, 0);
  const int y1 = __shfl(sy1, 0);
  const int slot1 = __shfl(sslot, 0);
  const int32_t blk = sel;
  const int x0 = p.bx[blk], y0 = p.by[blk];
  const int mid = p.macro_of ? p.macro_of[blk] : -1;
  if (mid >= 0) {
    // -------- macro move (reference: place_macro.c; free-target) -----
    const int ddx = x1 - x0, ddy = y1 - y0;
    const int32_t mm0 = p.macro_ptr[mid], mm1 = p.macro_ptr[mid + 1];
    if (mm1 - mm0 > 16) return;
    int ok = 0;
    if (lane == 0) {
      ok = 1;
      for (int32_t j = mm0; j < mm1 && ok; ++j) {
        int32_t b = p.macro_blk[j];
        int sx = p.bx[b], sy = p.by[b];
        int tx = sx + ddx, ty = sy + ddy;
        if (p.fixed && p.fixed[b]) ok = 0;
        else if (tx < 0 || tx >= p.gx || ty < 0 || ty >= p.gy) ok = 0;
        else if (p.rx0 >= 0 && (tx < p.rx0 || tx > p.rx1 ||
                                sx < p.rx0 || sx > p.rx1)) ok = 0;
        else if (p.tile_btype
                     ? (p.tile_btype[tx * p.gy + ty] != p.blk_type[b])
                     : (is_io_loc(p, tx, ty) != (p.blk_type[b] == 0))) ok = 0;
        else if (cap_at(p, tx, ty) != 1) ok = 0;
        else {
          int32_t occ = p.grid[((int64_t)tx * p.gy + ty) * p.cap];
          if (occ >= 0 && p.macro_of[occ] != mid) ok = 0;
        }
      }
      if (ok) atomicAdd(&m.counters[0], 1);
    }
    ok = __shfl(ok, 0);
    if (!ok) return;
    float dbb = 0.0f, dtd = 0.0f;
    for (int32_t j = mm0; j < mm1; ++j) {
      int32_t b = p.macro_blk[j];
      for (int32_t k = p.blk_net_ptr[b] + lane; k < p.blk_net_ptr[b + 1];
           k += 64) {
        int32_t n = p.blk_nets[k];
        bool dup = false;
        for (int32_t j2 = mm0; j2 < j && !dup; ++j2)
          dup = blk_has_net(p, p.macro_blk[j2], n);
        if (dup) continue;
        dbb += net_bb_cost_macro(p, n, mid, ddx, ddy) - p.net_cost[n];
        if (timing_tradeoff > 0.0f)
          dtd += net_td_cost_macro(p, n, mid, ddx, ddy) - p.net_tcost[n];
      }
    }
    dbb = wave_sum_f32(dbb);
    dtd = wave_sum_f32(dtd);
    float delta = (1.0f - timing_tradeoff) * dbb * inv_bb_norm +
                  timing_tradeoff * dtd * inv_td_norm;
    bool accept;
    if (delta <= 0.0f) accept = true;
    else if (T <= 0.0f) accept = false;
    else {
      float u = (rng_hash(seed, batch, i * 131 + 5) & 0xFFFFFF) *
                (1.0f / 16777216.0f);
      accept = u < __expf(-delta / T);
    }
    if (!accept) return;
    if (lane == 0) {
      atomicAdd(&m.counters[1], 1);
      m.mv_blk[i] = blk;
      m.mv_to[i] = ((ddx + 4096) << 13) | (ddy + 4096);
      m.mv_other[i] = -2 - mid;
      m.mv_dbb[i] = dbb;
      m.mv_dtd[i] = dtd;
      m.mv_flags[i] = 1;
    }
    for (int32_t j = mm0; j < mm1; ++j) {
      int32_t b = p.macro_blk[j];
      for (int32_t k = p.blk_net_ptr[b] + lane; k < p.blk_net_ptr[b + 1];
           k += 64)
        atomicMin(&m.net_claim[p.blk_nets[k]], i);
    }
    for (int32_t j = mm0 + lane; j < mm1; j += 64) {
      int32_t b = p.macro_blk[j];
      atomicMin(&m.loc_claim[p.bx[b] * p.gy + p.by[b]], i);
      atomicMin(&m.loc_claim[(p.bx[b] + ddx) * p.gy + p.by[b] + ddy], i);
    }
    return;
  }
  // -------- single-block move / swap (wave-uniform checks) --------
  int32_t other = p.grid[((int64_t)x1 * p.gy + y1) * p.cap + slot1];
  if (other == blk) return;
  if (other >= 0 && p.fixed && p.fixed[other]) return;
  if (other >= 0 && p.macro_of && p.macro_of[other] >= 0)
    return;   // never swap a chain member out from under its macro
  if (lane == 0) atomicAdd(&m.counters[0], 1);  // valid proposals only
  // exact deltas, lane-strided over the two blocks' net lists; nets
  // shared by both blocks are counted once (skip in other's pass)
  float dbb = 0.0f, dtd = 0.0f;
  const int ox = other >= 0 ? x0 : -1000, oy = other >= 0 ? y0 : -1000;
  for (int32_t k = p.blk_net_ptr[blk] + lane; k < p.blk_net_ptr[blk + 1];
       k += 64) {
    int32_t n = p.blk_nets[k];
    dbb += net_bb_cost(p, n, blk, x1, y1, other, ox, oy) - p.net_cost[n];
    if (timing_tradeoff > 0.0f)
      dtd += net_td_cost(p, n, blk, x1, y1, other, ox, oy) - p.net_tcost[n];
  }
  if (other >= 0) {
    for (int32_t k = p.blk_net_ptr[other] + lane;
         k < p.blk_net_ptr[other + 1]; k += 64) {
      int32_t n = p.blk_nets[k];
      if (blk_has_net(p, blk, n)) continue;
      dbb += net_bb_cost(p, n, blk, x1, y1, other, ox, oy) - p.net_cost[n];
      if (timing_tradeoff > 0.0f)
        dtd += net_td_cost(p, n, blk, x1, y1, other, ox, oy) - p.net_tcost[n];
    }
  }
  dbb = wave_sum_f32(dbb);
  dtd = wave_sum_f32(dtd);
  float delta = (1.0f - timing_tradeoff) * dbb * inv_bb_norm +
                timing_tradeoff * dtd * inv_td_norm;
  bool accept;
  if (delta <= 0.0f) accept = true;
  else if (T <= 0.0f) accept = false;
  else {
    float u = (rng_hash(seed, batch, i * 131 + 5) & 0xFFFFFF) *
              (1.0f / 16777216.0f);
    accept = u < __expf(-delta / T);
  }
  if (!accept) return;
  if (lane == 0) {
    atomicAdd(&m.counters[1], 1);
    m.mv_blk[i] = blk;
    m.mv_to[i] = ((x1 * p.gy + y1) * p.cap + slot1);
    m.mv_other[i] = other;
    m.mv_dbb[i] = dbb;
    m.mv_dtd[i] = dtd;
    m.mv_flags[i] = 1;
    atomicMin(&m.loc_claim[x0 * p.gy + y0], i);
    atomicMin(&m.loc_claim[x1 * p.gy + y1], i);
  }
  for (int32_t k = p.blk_net_ptr[blk] + lane; k < p.blk_net_ptr[blk + 1];
       k += 64)
    atomicMin(&m.net_claim[p.blk_nets[k]], i);
  if (other >= 0)
    for (int32_t k = p.blk_net_ptr[other] + lane;
         k < p.blk_net_ptr[other + 1]; k += 64)
      atomicMin(&m.net_claim[p.blk_nets[k]], i);
}

// lane-0 candidate selection (same RNG stream as the round-1 per-thread
// kernel); returns the block or -1, target via out-params
__device__ int propose_select_impl(const PlaceDev& p, int rlim,
                                   uint32_t seed, uint32_t batch, int i,
                                   int& x1o, int& y1o, int& slot1o) {
  uint32_t r0 = rng_hash(seed ^ 0x5BD1E995u, batch, i);
  int32_t blk = r0 % p.num_blocks;
  if (p.fixed && p.fixed[blk]) return -1;
  if (p.rx0 >= 0 && (p.bx[blk] < p.rx0 || p.bx[blk] > p.rx1))
    return -1;
  bool io = p.blk_type[blk] == 0;
  int x0 = p.bx[blk], y0 = p.by[blk];
  int x1 = -1, y1 = -1, slot1 = 0;
  int bt = p.blk_type[blk];
  if (p.tile_btype && bt >= 2 && p.type_cols) {
    // sparse column type (RAM/DSP): draw the target column from this
    // type's sorted column list clipped to the range window — rejection
    // over the square window would nearly always miss sparse columns
    // (mirrors the CPU placer's find_to for column types).
    int c0 = p.type_col_ptr[bt - 2], c1 = p.type_col_ptr[bt - 1];
    int lo = c0, hi = c1;              // first col >= x0 - rlim
    while (lo < hi) { int m = (lo + hi) >> 1;
                      if (p.type_cols[m] < x0 - rlim) lo = m + 1; else hi = m; }
    int lo2 = lo, hi2 = c1;            // first col > x0 + rlim
    while (lo2 < hi2) { int m = (lo2 + hi2) >> 1;
                        if (p.type_cols[m] <= x0 + rlim) lo2 = m + 1; else hi2 = m; }
    int ncol = lo2 - lo;
    if (ncol > 0) {
      int ylo = max(1, y0 - rlim), yhi = min(p.ny, y0 + rlim);
      for (int att = 0; att < 8; ++att) {
        uint32_t r1 = rng_hash(seed, batch, i * 131 + 7 * att + 1);
        uint32_t r2 = rng_hash(seed, batch, i * 131 + 7 * att + 2);
        int tx = p.type_cols[lo + (int)(r1 % ncol)];
        if (p.rx0 >= 0 && (tx < p.rx0 || tx > p.rx1)) continue;
        int ty = ylo + (int)(r2 % (yhi - ylo + 1));
        if (tx == x0 && ty == y0) continue;
        x1 = tx; y1 = ty; slot1 = 0;
        break;
      }
    }
  } else {
    for (int att = 0; att < 8; ++att) {
      uint32_t r1 = rng_hash(seed, batch, i * 131 + 7 * att + 1);
      uint32_t r2 = rng_hash(seed, batch, i * 131 + 7 * att + 2);
      int tx = x0 + (int)(r1 % (2 * rlim + 1)) - rlim;
      int ty = y0 + (int)(r2 % (2 * rlim + 1)) - rlim;
      if (tx < 0 || tx >= p.gx || ty < 0 || ty >= p.gy) continue;
      if (p.rx0 >= 0 && (tx < p.rx0 || tx > p.rx1)) continue;
      if (p.tile_btype) {
        // heterogeneous fabric: destination tile must match the block type
        if (p.tile_btype[tx * p.gy + ty] != p.blk_type[blk]) continue;
      } else if (is_io_loc(p, tx, ty) != io) continue;
      int c = cap_at(p, tx, ty);
      if (c <= 0) continue;
      if (tx == x0 && ty == y0) continue;
      x1 = tx; y1 = ty;
      slot1 = (int)(rng_hash(seed, batch, i * 131 + 7 * att + 3) % c);
      break;
    }
  }
  if (x1 < 0) return -1;
  x1o = x1; y1o = y1; slot1o = slot1;
  return blk;
}

__launch_bounds__(64 * PROP_WAVES, 2)
__global__ void place_resolve_kernel(PlaceDev p, MovesDev m) {
  const int lane = threadIdx.x & 63;
  const int i = blockIdx.x * PROP_WAVES + (threadIdx.x >> 6);
  if (i >= m.n_moves || m.mv_flags[i] != 1) return;
  int32_t blk = m.mv_blk[i];
  int32_t other = m.mv_other[i];
  int32_t to = m.mv_to[i];
  bool bad = false;
  if (other <= -2) {
    // macro move: every member's src+dst tile and every member net
    const int mid = -2 - other;
    const int ddx = (to >> 13) - 4096, ddy = (to & 0x1FFF) - 4096;
    const int32_t mm0 = p.macro_ptr[mid], mm1 = p.macro_ptr[mid + 1];
    for (int32_t j = mm0 + lane; j < mm1; j += 64) {
      int32_t b = p.macro_blk[j];
      int sx = p.bx[b], sy = p.by[b];
      if (m.loc_claim[sx * p.gy + sy] != i ||
          m.loc_claim[(sx + ddx) * p.gy + sy + ddy] != i) bad = true;
    }
    for (int32_t j = mm0; j < mm1; ++j) {
      int32_t b = p.macro_blk[j];
      for (int32_t k = p.blk_net_ptr[b] + lane; k < p.blk_net_ptr[b + 1];
           k += 64)
        if (m.net_claim[p.blk_nets[k]] != i) bad = true;
    }
  } else {
    int x1 = to / p.cap / p.gy, y1 = (to / p.cap) % p.gy;
    int x0 = p.bx[blk], y0 = p.by[blk];
    if (lane == 0 && (m.loc_claim[x0 * p.gy + y0] != i ||
                      m.loc_claim[x1 * p.gy + y1] != i)) bad = true;
    for (int32_t k = p.blk_net_ptr[blk] + lane; k < p.blk_net_ptr[blk + 1];
         k += 64)
      if (m.net_claim[p.blk_nets[k]] != i) bad = true;
    if (other >= 0)
      for (int32_t k = p.blk_net_ptr[other] + lane;
           k < p.blk_net_ptr[other + 1]; k += 64)
        if (m.net_claim[p.blk_nets[k]] != i) bad = true;
  }
  const bool win = !__any(bad);
  if (lane == 0) {
    if (!win) { m.mv_flags[i] = 0; atomicAdd(&m.counters[3], 1); }
    else { m.mv_flags[i] = 2; atomicAdd(&m.counters[2], 1); }
  }
}

__launch_bounds__(64 * PROP_WAVES, 2)
__global__ void place_apply_kernel(PlaceDev p, MovesDev m, float timing_tradeoff,
                                   double* cost_acc /*[2]: dbb, dtd*/) {
  const int lane = threadIdx.x & 63;
  const int i = blockIdx.x * PROP_WAVES + (threadIdx.x >> 6);
  if (i >= m.n_moves || m.mv_flags[i] != 2) return;
  int32_t blk = m.mv_blk[i];
  int32_t other = m.mv_other[i];
  int32_t to = m.mv_to[i];
  // Net-cost refresh runs FIRST, lane-strided, using the override cost
  // functions on the PRE-move positions (so lanes never read positions
  // another lane just wrote); duplicate refreshes of nets shared by both
  // blocks are idempotent. Then lane 0 commits the grid/position update.
  if (other <= -2) {
    const int mid = -2 - other;
    const int ddx = (to >> 13) - 4096, ddy = (to & 0x1FFF) - 4096;
    const int32_t mm0 = p.macro_ptr[mid], mm1 = p.macro_ptr[mid + 1];
    for (int32_t j = mm0; j < mm1; ++j) {
      int32_t b = p.macro_blk[j];
      for (int32_t k = p.blk_net_ptr[b] + lane; k < p.blk_net_ptr[b + 1];
           k += 64) {
        int32_t n = p.blk_nets[k];
        p.net_cost[n] = net_bb_cost_macro(p, n, mid, ddx, ddy);
        if (timing_tradeoff > 0.0f)
          p.net_tcost[n] = net_td_cost_macro(p, n, mid, ddx, ddy);
      }
    }
    if (lane == 0) {
      for (int32_t j = mm0; j < mm1; ++j) {
        int32_t b = p.macro_blk[j];
        p.grid[((int64_t)p.bx[b] * p.gy + p.by[b]) * p.cap + p.bslot[b]] = -1;
      }
      for (int32_t j = mm0; j < mm1; ++j) {
        int32_t b = p.macro_blk[j];
        int tx = p.bx[b] + ddx, ty = p.by[b] + ddy;
        p.grid[((int64_t)tx * p.gy + ty) * p.cap] = b;
        p.bx[b] = tx; p.by[b] = ty; p.bslot[b] = 0;
      }
      unsafeAtomicAdd(&cost_acc[0], (double)m.mv_dbb[i]);
      unsafeAtomicAdd(&cost_acc[1], (double)m.mv_dtd[i]);
    }
    return;
  }
  int slot1 = to % p.cap;
  int x1 = to / p.cap / p.gy, y1 = (to / p.cap) % p.gy;
  int x0 = p.bx[blk], y0 = p.by[blk], s0 = p.bslot[blk];
  const int ox = other >= 0 ? x0 : -1000, oy = other >= 0 ? y0 : -1000;
  for (int32_t k = p.blk_net_ptr[blk] + lane; k < p.blk_net_ptr[blk + 1];
       k += 64) {
    int32_t n = p.blk_nets[k];
    p.net_cost[n] = net_bb_cost(p, n, blk, x1, y1, other, ox, oy);
    if (timing_tradeoff > 0.0f)
      p.net_tcost[n] = net_td_cost(p, n, blk, x1, y1, other, ox, oy);
  }
  if (other >= 0)
    for (int32_t k = p.blk_net_ptr[other] + lane;
         k < p.blk_net_ptr[other + 1]; k += 64) {
      int32_t n = p.blk_nets[k];
      p.net_cost[n] = net_bb_cost(p, n, blk, x1, y1, other, ox, oy);
      if (timing_tradeoff > 0.0f)
        p.net_tcost[n] = net_td_cost(p, n, blk, x1, y1, other, ox, oy);
    }
  if (lane == 0) {
    p.grid[((int64_t)x0 * p.gy + y0) * p.cap + s0] = other >= 0 ? other : -1;
    p.grid[((int64_t)x1 * p.gy + y1) * p.cap + slot1] = blk;
    p.bx[blk] = x1; p.by[blk] = y1; p.bslot[blk] = slot1;
    if (other >= 0) { p.bx[other] = x0; p.by[other] = y0; p.bslot[other] = s0; }
    unsafeAtomicAdd(&cost_acc[0], (double)m.mv_dbb[i]);
    unsafeAtomicAdd(&cost_acc[1], (double)m.mv_dtd[i]);
  }
}

// claims reset per batch; counters accumulate across batches (host zeroes
// them per run_batches call) so the schedule sees the TRUE accept rate.
__global__ void place_reset_claims_kernel(int32_t* net_claim, int32_t nn,
                                          int32_t* loc_claim, int32_t nl,
                                          int32_t* counters) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int total = max(nn, nl);
  (void)counters;
  for (; i < total; i += gridDim.x * blockDim.x) {
    if (i < nn) net_claim[i] = 0x7FFFFFFF;
    if (i < nl) loc_claim[i] = 0x7FFFFFFF;
  }
}

// full net-cost refresh (init + periodic drift resync).
// Host must zero out[0..1] before the launch.
__global__ void place_refresh_costs_kernel(PlaceDev p, float timing_tradeoff,
                                           double* out /*[2]*/) {
  int n = blockIdx.x * blockDim.x + threadIdx.x;
  for (; n < p.num_nets; n += gridDim.x * blockDim.x) {
    float c = net_bb_cost(p, n, -1, 0, 0, -1, 0, 0);
    p.net_cost[n] = c;
    unsafeAtomicAdd(&out[0], (double)c);
    float t = (timing_tradeoff > 0.0f)
                  ? net_td_cost(p, n, -1, 0, 0, -1, 0, 0) : 0.0f;
    p.net_tcost[n] = t;
    unsafeAtomicAdd(&out[1], (double)t);
  }
}

}  // namespace pnrh

using namespace pnrh;

extern "C" {

struct PlaceLaunchArgs {
  const int32_t* net_blk_ptr; const int32_t* net_blks;
  const int32_t* blk_net_ptr; const int32_t* blk_nets;
  const int8_t* blk_type; const int8_t* tile_btype;
  const int32_t* type_cols; const int32_t* type_col_ptr;
  const uint8_t* fixed;
  const float* net_q;
  const int32_t* net_sink_ptr; const float* conn_crit; const float* delay_mat;
  int32_t* bx; int32_t* by; int32_t* bslot; int32_t* grid;
  float* net_cost; float* net_tcost;
  int32_t num_blocks, num_nets, gx, gy, cap, nx, ny, io_cap;
  int32_t rx0, rx1;
  const int32_t* macro_of; const int32_t* macro_ptr;
  const int32_t* macro_blk;
  // moves
  int32_t* mv_blk; int32_t* mv_to; int32_t* mv_other;
  float* mv_dbb; float* mv_dtd; uint8_t* mv_flags;
  int32_t* net_claim; int32_t* loc_claim; int32_t* counters;
  int32_t n_moves;
  // schedule
  float T; int32_t rlim; float timing_tradeoff;
  float inv_bb_norm, inv_td_norm;
  uint32_t seed, batch;
  double* cost_acc;
};

static void unpack(const PlaceLaunchArgs* a, PlaceDev& p, MovesDev& m) {
  p.net_blk_ptr = a->net_blk_ptr; p.net_blks = a->net_blks;
  p.blk_net_ptr = a->blk_net_ptr; p.blk_nets = a->blk_nets;
  p.blk_type = a->blk_type; p.tile_btype = a->tile_btype;
  p.type_cols = a->type_cols; p.type_col_ptr = a->type_col_ptr;
  p.fixed = a->fixed;
  p.net_q = a->net_q;
  p.net_sink_ptr = a->net_sink_ptr; p.conn_crit = a->conn_crit;
  p.delay_mat = a->delay_mat;
  p.bx = a->bx; p.by = a->by; p.bslot = a->bslot; p.grid = a->grid;
  p.net_cost = a->net_cost; p.net_tcost = a->net_tcost;
  p.num_blocks = a->num_blocks; p.num_nets = a->num_nets;
  p.gx = a->gx; p.gy = a->gy; p.cap = a->cap;
  p.nx = a->nx; p.ny = a->ny; p.io_cap = a->io_cap;
  p.rx0 = a->rx0; p.rx1 = a->rx1;
  p.macro_of = a->macro_of; p.macro_ptr = a->macro_ptr;
  p.macro_blk = a->macro_blk;
  m.mv_blk = a->mv_blk; m.mv_to = a->mv_to; m.mv_other = a->mv_other;
  m.mv_dbb = a->mv_dbb; m.mv_dtd = a->mv_dtd; m.mv_flags = a->mv_flags;
  m.net_claim = a->net_claim; m.loc_claim = a->loc_claim;
  m.counters = a->counters; m.n_moves = a->n_moves;
}

int pnr_place_batch(const PlaceLaunchArgs* a, void* stream) {
  PlaceDev p; MovesDev m;
  unpack(a, p, m);
  hipStream_t s = (hipStream_t)stream;
  int rg = (max(max(a->num_nets, a->gx * a->gy), a->n_moves) + 255) / 256;
  rg = rg < 2048 ? rg : 2048;
  hipLaunchKernelGGL(place_reset_claims_kernel, dim3(rg), dim3(256), 0, s,
                     m.net_claim, p.num_nets, m.loc_claim, p.gx * p.gy,
                     m.counters);
  // one wave64 per proposal, PROP_WAVES waves per workgroup
  int mg = (m.n_moves + PROP_WAVES - 1) / PROP_WAVES;
  hipLaunchKernelGGL(place_propose_kernel, dim3(mg), dim3(64 * PROP_WAVES),
                     0, s, p, m, a->T, a->rlim, a->timing_tradeoff,
                     a->inv_bb_norm, a->inv_td_norm, a->seed, a->batch);
  hipLaunchKernelGGL(place_resolve_kernel, dim3(mg), dim3(64 * PROP_WAVES),
                     0, s, p, m);
  hipLaunchKernelGGL(place_apply_kernel, dim3(mg), dim3(64 * PROP_WAVES),
                     0, s, p, m, a->timing_tradeoff, a->cost_acc);
  return (int)hipGetLastError();
}

int pnr_place_refresh(const PlaceLaunchArgs* a, void* stream) {
  PlaceDev p; MovesDev m;
  unpack(a, p, m);
  hipStream_t s = (hipStream_t)stream;
  // single-block prologue zeroes out; then grid-stride accumulate
  hipLaunchKernelGGL(place_refresh_costs_kernel, dim3(1024), dim3(256), 0, s,
                     p, a->timing_tradeoff, a->cost_acc);
  return (int)hipGetLastError();
}

int64_t pnr_place_args_sizeof() { return (int64_t)sizeof(PlaceLaunchArgs); }

}  // extern "C"
