// EXPERIMENTAL (round-2): calendar-queue frontier for the wavefront router.
//
// Replaces the ping-pong frontier of route_nets_kernel (which rescans every
// kept entry each delta-stepping round — measured 4-6x scans/touched, and
// 339k scans/sink at bitcoin_miner scale) with a circular calendar of
// NB=64 buckets indexed by floor((tot - f0)/delta) plus a ping-pong
// overflow region. The algorithm was validated against exact Dijkstra and
// the ping-pong structure in tools/frontier_sim.py (6.7x fewer entry
// scans at identical results). Memory layout reuses the existing per-slot
// frontier allocation: buckets = buffer0 (64 x f_cap/64), overflow =
// buffer1 split in two halves.
//
// Launch via pnr_route_nets with use_calendar=1 (default 0 — the
// validated ping-pong kernel remains the production path until this one
// is measured on hardware).
#include "pnr_hip.h"

namespace pnrh {

#define CWG_THREADS 256
#define CAL_NB 64
#define CPATH_CAP 4096
#define CFAIL_FRONTIER 1
#define CFAIL_ROUNDS 2
#define CFAIL_NO_PATH 3
#define CFAIL_TREE_CAP 4
#define CFAIL_PATH_CAP 5
#define CFAIL_TOUCHED 6
#define CINF_STATE 0xffffffffffffffffull

struct CalShared {
  int fail;
  int bcnt[CAL_NB];      // entries in each bucket slot
  int bproc;             // processed cursor of the CURRENT bucket
  int ov_cnt[2];         // overflow ping-pong counts
  int ov_cur;
  unsigned best_sink_back;
  unsigned ov_min;       // min tot bits over overflow (for redistribution)
  int touched_cnt;
  int attach_idx;
  int path_len;
  unsigned attach_node;
  float f0;
  int base;              // absolute index of the current bucket
  int qi;
  int tree_len;
  int32_t path[CPATH_CAP];
};

__launch_bounds__(CWG_THREADS, 1)
__global__ void route_nets_cal_kernel(
    RRDev g, NetsDev nets, TreesDev trees, RouteParams P,
    const int32_t* __restrict__ queue_small, int32_t n_queue_small,
    const int32_t* __restrict__ queue_large, int32_t n_queue_large,
    int32_t* q_cursors,
    int32_t* occ, const float* __restrict__ acc,
    uint64_t* state_base, int64_t small_cap, int64_t large_cap,
    int32_t n_small_slots,
    float4* frontier_base, int64_t f_cap_small, int64_t f_cap_large,
    int32_t* touched_base, int64_t t_cap_small, int64_t t_cap_large,
    int32_t* fail_flags, unsigned long long* stats,
    unsigned long long* net_scans) {
  const int tid = threadIdx.x;
  const bool is_small = (int)blockIdx.x < n_small_slots;
  const int slot = blockIdx.x;

  __shared__ CalShared sh;

  uint64_t* state;
  float4* buckets;    // CAL_NB x bcap
  float4* ovf[2];     // 2 x ov_cap
  int32_t* touched;
  int64_t f_cap, t_cap;
  if (is_small) {
    state = state_base + (int64_t)slot * small_cap;
    float4* fb = frontier_base + (int64_t)slot * 2 * f_cap_small;
    f_cap = f_cap_small; t_cap = t_cap_small;
    buckets = fb;
    ovf[0] = fb + f_cap; ovf[1] = fb + f_cap + f_cap / 2;
    touched = touched_base + (int64_t)slot * t_cap_small;
  } else {
    int ls = slot - n_small_slots;
    state = state_base + (int64_t)n_small_slots * small_cap
          + (int64_t)ls * large_cap;
    float4* fb = frontier_base + (int64_t)n_small_slots * 2 * f_cap_small
               + (int64_t)ls * 2 * f_cap_large;
    f_cap = f_cap_large; t_cap = t_cap_large;
    buckets = fb;
    ovf[0] = fb + f_cap; ovf[1] = fb + f_cap + f_cap / 2;
    touched = touched_base + (int64_t)n_small_slots * t_cap_small
            + (int64_t)ls * t_cap_large;
  }
  const int64_t s_cap = is_small ? small_cap : large_cap;
  (void)s_cap;
  const int bcap = (int)(f_cap / CAL_NB);
  const int ov_cap = (int)(f_cap / 2);

  const int32_t* queue = is_small ? queue_small : queue_large;
  const int32_t n_queue = is_small ? n_queue_small : n_queue_large;
  int32_t* cursor = q_cursors + (is_small ? 0 : 1);

  for (;;) {
    if (tid == 0) sh.qi = atomicAdd(cursor, 1);
    __syncthreads();
    int qi = sh.qi;
    __syncthreads();
    if (qi >= n_queue) return;
    const int32_t inet = queue[qi];

    const int32_t src = nets.src[inet];
    const int32_t s0 = nets.sink_ptr[inet], s1 = nets.sink_ptr[inet + 1];
    const int64_t toff = trees.off[inet];
    const int32_t tcap = (int32_t)(trees.off[inet + 1] - toff);
    int32_t* t_node = trees.node + toff;
    int32_t* t_parent = trees.parent + toff;
    int8_t* t_sw = trees.sw + toff;
    float* t_delay = trees.delay + toff;

    LocalIdx L;
    L.dense = is_small;
    L.bx0 = nets.bb[4 * inet + 0];
    L.by0 = nets.bb[4 * inet + 1];
    L.bw = nets.bb[4 * inet + 2] - L.bx0 + 1;
    L.bh = nets.bb[4 * inet + 3] - L.by0 + 1;
    L.npt = g.npt;

    // rip-up + root (same as production kernel; partial mode mirrors
    // router_kernel.hip's P.partial branches)
    int32_t old_len = trees.len[inet];
    if (tid == 0) {
      int ok = 0;
      if (P.partial && old_len > 0 && old_len <= tcap &&
          old_len <= (int32_t)t_cap) {
        int32_t r = t_node[0];
        if (r >= 0 && r < g.num_nodes && occ[r] <= g.capacity[r]) ok = 1;
      }
      sh.qi = ok;
    }
    __syncthreads();
    const bool do_partial = sh.qi != 0;
    __syncthreads();
    int tree_len;
    if (!do_partial) {
      for (int k = tid; k < old_len && k < tcap; k += CWG_THREADS)
        atomicSub(&occ[t_node[k]], 1);
      __syncthreads();
      if (tid == 0) {
        t_node[0] = src; t_parent[0] = -1; t_sw[0] = -1; t_delay[0] = 0.0f;
        trees.len[inet] = 1;
        atomicAdd(&occ[src], 1);
        sh.fail = 0;
        sh.tree_len = 1;
      }
      __syncthreads();
      tree_len = 1;
    } else {
      if (tid == 0) {
        int32_t keep = 0;
        for (int32_t k = 0; k < old_len; ++k) {
          int32_t v = t_node[k];
          int32_t par = t_parent[k];
          int8_t swk = t_sw[k];
          float dl = t_delay[k];
          bool pdrop = (k > 0) && (par >= 0) && (touched[par] < 0);
          bool cong = (v < 0 || v >= g.num_nodes) ? true
                      : (occ[v] > g.capacity[v]);
          if (k > 0 && (pdrop || cong)) {
            touched[k] = -1;
            if (v >= 0 && v < g.num_nodes) atomicSub(&occ[v], 1);
            continue;
          }
          touched[k] = keep;
          t_node[keep] = v;
          t_parent[keep] = (k == 0) ? -1 : touched[par];
          t_sw[keep] = swk;
          t_delay[keep] = dl;
          ++keep;
        }
        trees.len[inet] = keep;
        sh.tree_len = keep;
        sh.fail = 0;
      }
      __syncthreads();
      tree_len = sh.tree_len;
      __syncthreads();
    }

    for (int32_t si = s0; si < s1; ++si) {
      SinkCtx S;
      S.sink_node = nets.sink_rr[si];
      S.sx = g.xlow[S.sink_node];
      S.sy = g.ylow[S.sink_node];
      S.crit = nets.crit[si];
      S.astar_fac = P.astar_fac;
      if (do_partial) {
        if (tid == 0) sh.qi = 0;
        __syncthreads();
        for (int k = tid; k < tree_len; k += CWG_THREADS)
          if (t_node[k] == S.sink_node) atomicOr(&sh.qi, 1);
        __syncthreads();
        const bool connected = sh.qi != 0;
        __syncthreads();
        if (connected) continue;   // kept subtree still reaches this sink
      }
      const float delta = P.delta_fac * (S.crit * P.seg_delay +
                                         (1.0f - S.crit) * P.seg_base);
      const float inv_delta = 1.0f / delta;

      if (tid == 0) {
        for (int b = 0; b < CAL_NB; ++b) sh.bcnt[b] = 0;
        sh.ov_cnt[0] = sh.ov_cnt[1] = 0;
        sh.ov_cur = 0;
        sh.best_sink_back = 0xffffffffu;
        sh.touched_cnt = 0;
        sh.base = 0;
        // calendar base: the source-side lower bound on tot
        sh.f0 = S.astar_fac * expected_cost(g, P, src, S);
        sh.bproc = 0;
      }
      __syncthreads();

      // push helper: bucket by (tot - f0) / delta relative to sh.base.
      // Entries >= CAL_NB-1 buckets ahead go to the overflow region.
      auto push_entry = [&](float tot, float back, int32_t node, int32_t prev) {
        int k = (int)((tot - sh.f0) * inv_delta);
        if (k < 0) k = 0;
        if (k >= CAL_NB - 1) {
          int oi = atomicAdd(&sh.ov_cnt[sh.ov_cur], 1);
          if (oi < ov_cap)
            ovf[sh.ov_cur][oi] = make_float4(tot, back,
                                             __int_as_float(node),
                                             __int_as_float(prev));
          atomicMin(&sh.ov_min, f32_bits(tot));
          return;
        }
        int bs = (sh.base + k) % CAL_NB;
        int bi = atomicAdd(&sh.bcnt[bs], 1);
        if (bi < bcap) {
          buckets[(int64_t)bs * bcap + bi] =
              make_float4(tot, back, __int_as_float(node),
                          __int_as_float(prev));
        } else {
          // bucket full: spill to overflow. Do NOT decrement the counter
          // (a concurrent fetch-add against a decremented counter can
          // hand out an already-used slot and silently overwrite an
          // entry); the scan clamps to bcap and the advance resets it.
          int oi = atomicAdd(&sh.ov_cnt[sh.ov_cur], 1);
          if (oi < ov_cap)
            ovf[sh.ov_cur][oi] = make_float4(tot, back,
                                             __int_as_float(node),
                                             __int_as_float(prev));
          atomicMin(&sh.ov_min, f32_bits(tot));
        }
      };

      if (tid == 0) sh.ov_min = 0xffffffffu;
      __syncthreads();

      // seeds
      for (int k = tid; k < tree_len; k += CWG_THREADS) {
        int32_t v = t_node[k];
        if (g.type[v] == 1) continue;
        if (!L.in_bb(g, v)) continue;
        float back = S.crit * t_delay[k];
        float tot = back + S.astar_fac * expected_cost(g, P, v, S);
        int64_t li = L(g, v);
        state[li] = pack_state(0.0f, v);
        int ti = atomicAdd(&sh.touched_cnt, 1);
        if (ti < t_cap) touched[ti] = (int32_t)li;
        push_entry(tot, back, v, v);
      }
      __syncthreads();

      int64_t scanned = 0;
      int rounds = 0;
      // ---- calendar main loop ----
      for (;;) {
        if (sh.fail) break;
        int cur = sh.base % CAL_NB;
        int n_in_bucket = min(sh.bcnt[cur], bcap);
        if (sh.bproc < n_in_bucket) {
          if (++rounds > P.max_rounds) {
            if (tid == 0) sh.fail = CFAIL_ROUNDS;
            __syncthreads();
            break;
          }
          int lo = sh.bproc, hi = n_in_bucket;
          __syncthreads();
          for (int i = lo + tid; i < hi; i += CWG_THREADS) {
            float4 e = buckets[(int64_t)cur * bcap + i];
            float back = e.y;
            int32_t v = __float_as_int(e.z);
            int32_t prev = __float_as_int(e.w);
            int64_t li = L(g, v);
            const uint64_t expect = (prev == v) ? pack_state(0.0f, v)
                                                : pack_state(back, prev);
            if (load_state(&state[li]) != expect) continue;
            if (v == S.sink_node) {
              atomicMin(&sh.best_sink_back, f32_bits(back));
              continue;
            }
            int32_t e0 = g.row_ptr[v], e1 = g.row_ptr[v + 1];
            for (int32_t ei = e0; ei < e1; ++ei) {
              int32_t w = g.edge_dst[ei];
              int8_t ty = g.type[w];
              if (ty == 1 && w != S.sink_node) continue;
              if (ty == 3 && (g.xlow[w] != S.sx || g.ylow[w] != S.sy))
                continue;
              if (!L.in_bb(g, w)) continue;
              int8_t sw = g.edge_sw[ei];
              float back_new = back + S.crit * hop_delay(g, sw, w) +
                               (1.0f - S.crit) * cong_cost(g, P, occ, acc, w);
              float tot_new = back_new +
                              S.astar_fac * expected_cost(g, P, w, S);
              int64_t lw = L(g, w);
              uint64_t pk = pack_state(back_new, v);
              uint64_t old = atomicMin((unsigned long long*)&state[lw],
                                       (unsigned long long)pk);
              if (pk < old) {
                if (old == CINF_STATE) {
                  int ti = atomicAdd(&sh.touched_cnt, 1);
                  if (ti < t_cap) touched[ti] = (int32_t)lw;
                }
                push_entry(tot_new, back_new, w, v);
                if (w == S.sink_node)
                  atomicMin(&sh.best_sink_back, f32_bits(back_new));
              }
            }
          }
          scanned += hi - lo;
          __syncthreads();
          if (tid == 0) {
            sh.bproc = hi;  // appended same-bucket entries get a next pass
            if (sh.touched_cnt > t_cap) sh.fail = CFAIL_TOUCHED;
          }
          __syncthreads();
          continue;
        }
        // current bucket exhausted: terminate / advance / redistribute.
        // Remaining CALENDAR entries all have tot >= f0 + delta (later
        // buckets) — but the OVERFLOW region also holds bucket-full
        // spills whose tot may be BELOW that (ov_min tracks them), so the
        // termination bound is min(f0 + delta, ov_min). Round-1 shipped
        // without the ov_min term and terminated past cheap spilled
        // entries under the pres_fac=0 tie explosion: LU32 A/B measured
        // +27% wirelength and a 20k-overused plateau. Strict mode must
        // still process ties at exactly the bound.
        float term_bound = sh.f0 + delta;
        if (sh.ov_cnt[sh.ov_cur] > 0)
          term_bound = fminf(term_bound, bits_f32(sh.ov_min));
        bool done = sh.best_sink_back != 0xffffffffu &&
                    (P.strict_term
                         ? bits_f32(sh.best_sink_back) < term_bound
                         : bits_f32(sh.best_sink_back) <= term_bound);
        if (done) break;
        // any entries left in the calendar?
        bool any = false;
        for (int b = 0; b < CAL_NB; ++b)
          if (sh.bcnt[b] > 0) { any = true; break; }
        if (any) {
          __syncthreads();
          if (tid == 0) {
            sh.bcnt[sh.base % CAL_NB] = 0;
            sh.base += 1;
            sh.f0 += delta;
            sh.bproc = 0;
          }
          __syncthreads();
          continue;
        }
        // calendar empty: redistribute overflow or finish
        if (sh.ov_cnt[sh.ov_cur] == 0) {
          if (tid == 0 && sh.best_sink_back == 0xffffffffu)
            sh.fail = CFAIL_NO_PATH;
          __syncthreads();
          break;
        }
        {
          int src_ov = sh.ov_cur;
          int n_ov = min(sh.ov_cnt[src_ov], ov_cap);
          __syncthreads();
          if (tid == 0) {
            if (sh.ov_cnt[src_ov] > ov_cap) sh.fail = CFAIL_FRONTIER;
            sh.ov_cur = src_ov ^ 1;
            sh.ov_cnt[src_ov ^ 1] = 0;
            // rebase the calendar at the overflow's minimum tot
            sh.f0 = bits_f32(sh.ov_min);
            sh.ov_min = 0xffffffffu;
            sh.bproc = 0;
          }
          __syncthreads();
          if (sh.fail) break;
          for (int i = tid; i < n_ov; i += CWG_THREADS) {
            float4 e = ovf[src_ov][i];
            float back = e.y;
            int32_t v = __float_as_int(e.z);
            int32_t prev = __float_as_int(e.w);
            int64_t li = L(g, v);
            const uint64_t expect = (prev == v) ? pack_state(0.0f, v)
                                                : pack_state(back, prev);
            if (load_state(&state[li]) != expect) continue;  // drop stale
            push_entry(e.x, back, v, prev);
          }
          scanned += n_ov;
          __syncthreads();
          if (tid == 0) sh.ov_cnt[src_ov] = 0;
          __syncthreads();
          continue;
        }
      }
      __syncthreads();
      if (tid == 0 && stats) {
        atomicAdd(&stats[0], (unsigned long long)rounds);
        atomicAdd(&stats[1], (unsigned long long)scanned);
        atomicAdd(&stats[2], 1ull);
        atomicAdd(&stats[3], (unsigned long long)sh.touched_cnt);
        if (net_scans) atomicAdd(&net_scans[inet], (unsigned long long)scanned);
      }

      // ---- backtrack + commit (identical to the production kernel) ----
      if (!sh.fail && tid == 0) {
        int n = 0;
        int32_t v = S.sink_node;
        for (;;) {
          uint64_t st = load_state(&state[(size_t)L(g, v)]);
          if (st == CINF_STATE) { sh.fail = CFAIL_NO_PATH; break; }
          int32_t prev = (int32_t)(st & 0xffffffffu);
          if (prev == v) break;
          if (n >= CPATH_CAP) { sh.fail = CFAIL_PATH_CAP; break; }
          sh.path[n++] = v;
          v = prev;
        }
        sh.path_len = n;
        sh.attach_idx = -1;
        sh.attach_node = (unsigned)v;
      }
      __syncthreads();
      if (sh.fail) break;
      {
        int32_t attach_node = (int32_t)sh.attach_node;
        for (int k = tid; k < tree_len; k += CWG_THREADS)
          if (t_node[k] == attach_node) sh.attach_idx = k;
      }
      __syncthreads();
      if (tid == 0) {
        int ai = sh.attach_idx;
        if (ai < 0) { sh.fail = CFAIL_NO_PATH; }
        else {
          float dacc = t_delay[ai];
          int parent = ai;
          int len = tree_len;
          for (int k = sh.path_len - 1; k >= 0; --k) {
            int32_t u = sh.path[k];
            int32_t pu = t_node[parent];
            int8_t sw = 0;
            for (int32_t ei = g.row_ptr[pu]; ei < g.row_ptr[pu + 1]; ++ei)
              if (g.edge_dst[ei] == u) { sw = g.edge_sw[ei]; break; }
            dacc += hop_delay(g, sw, u);
            if (len >= tcap) { sh.fail = CFAIL_TREE_CAP; break; }
            t_node[len] = u; t_parent[len] = parent; t_sw[len] = sw;
            t_delay[len] = dacc;
            parent = len;
            ++len;
            atomicAdd(&occ[u], 1);
          }
          tree_len = len;
          trees.len[inet] = len;
          if (!sh.fail) trees.sink_delay[nets.sink_orig[si]] = dacc;
          sh.tree_len = len;
        }
      }
      __syncthreads();
      if (sh.fail) break;
      tree_len = sh.tree_len;

      // sparse state reset
      int nt = min((int64_t)sh.touched_cnt, t_cap);
      for (int k = tid; k < nt; k += CWG_THREADS)
        state[(size_t)(uint32_t)touched[k]] = CINF_STATE;
      __syncthreads();
    }

    if (tid == 0 && sh.fail) fail_flags[inet] = sh.fail;
    __syncthreads();
    if (sh.fail) {
      int nt = min((int64_t)sh.touched_cnt, t_cap);
      for (int k = tid; k < nt; k += CWG_THREADS)
        state[(size_t)(uint32_t)touched[k]] = CINF_STATE;
      if (sh.touched_cnt > t_cap) {
        int64_t cap = is_small ? small_cap : large_cap;
        for (int64_t k = tid; k < cap; k += CWG_THREADS) state[k] = CINF_STATE;
      }
      __syncthreads();
    }
  }
}

}  // namespace pnrh
