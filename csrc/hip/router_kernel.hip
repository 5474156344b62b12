// CDNA4 wavefront router — the hot engine.
//
// Re-designs the reference's per-sink A*/delta-stepping sink router
// (vpr/SRC/parallel_route/dijkstra.h:16, delta_stepping.h:45,
//  SinkRouter partitioning_multi_sink_delta_stepping_route.cxx:360-780)
// for the MI355X execution model: one 256-thread workgroup (4 wave64) per
// net, delta-stepping bucket expansion with ping-pong frontier buffers in
// HBM, search state as a packed (back_cost, prev) u64 updated with a single
// 64-bit atomicMin (deterministic tie-break on prev id), congestion shared
// across concurrently-routed nets via device-scope atomics on occ
// (the reference's net-level-parallel family, locking_route.cxx semantics,
// with pres_cost computed on the fly from occ — no lock needed).
//
// Small nets use bb-dense local state (bb tiles x nodes-per-tile); large
// nets use whole-graph state in a few dedicated slots. Touched-list reset
// keeps per-sink state reset O(visited), like the reference's sparse reset
// (partitioning_multi_sink...cxx:769-779).
#include "pnr_hip.h"

namespace pnrh {
// EXPERIMENTAL calendar-frontier variant (csrc/hip/router_calendar.hip)
__global__ void route_nets_cal_kernel(
    RRDev g, NetsDev nets, TreesDev trees, RouteParams P,
    const int32_t*, int32_t, const int32_t*, int32_t, int32_t*,
    int32_t*, const float*, uint64_t*, int64_t, int64_t, int32_t,
    float4*, int64_t, int64_t, int32_t*, int64_t, int64_t,
    int32_t*, unsigned long long*, unsigned long long*);

#define WG_THREADS 256
#define PATH_CAP 4096      // max path nodes per sink backtrack (LDS)
#define FAIL_FRONTIER 1
#define FAIL_ROUNDS 2
#define FAIL_NO_PATH 3
#define FAIL_TREE_CAP 4
#define FAIL_PATH_CAP 5
#define FAIL_TOUCHED 6

#define INF_STATE 0xffffffffffffffffull

// Debug bounds checks must be CONTROL-FLOW NEUTRAL: they clamp the index
// and record a bit in sh.dbg (read only at net end). Writing sh.fail from
// arbitrary threads mid-round would diverge the round loop's exit and
// corrupt the workgroup at the next barrier.
#ifdef PNR_DEBUG_BOUNDS
#define DBG_LI(li, cap, code) \
  (((li) < 0 || (li) >= (cap)) ? (atomicOr(&sh.dbg, 1 << ((code) - 90)), (int64_t)0) : (li))
#define DBG_NODE(v, code) \
  (((v) < 0 || (v) >= g.num_nodes) ? (atomicOr(&sh.dbg, 1 << ((code) - 90)), 0) : (v))
#else
#define DBG_LI(li, cap, code) (li)
#define DBG_NODE(v, code) (v)
#endif

struct WgShared {
  int dbg;
  int fcnt[2];
  unsigned fmin_next;
  unsigned best_sink_back;
  int touched_cnt;
  int attach_idx;
  int fail;
  int n_cur;
  int path_len;
  int32_t path[PATH_CAP];
};

__launch_bounds__(WG_THREADS, 1)
__global__ void route_nets_kernel(
    RRDev g, NetsDev nets, TreesDev trees, RouteParams P,
    const int32_t* __restrict__ queue_small, int32_t n_queue_small,
    const int32_t* __restrict__ queue_large, int32_t n_queue_large,
    int32_t* q_cursors,                // [2]
    int32_t* occ, const float* __restrict__ acc,
    uint64_t* state_base, int64_t small_cap, int64_t large_cap,
    int32_t n_small_slots,
    float4* frontier_base, int64_t f_cap_small, int64_t f_cap_large,
    int32_t* touched_base, int64_t t_cap_small, int64_t t_cap_large,
    int32_t* fail_flags, unsigned long long* stats /*[8] or null*/,
    unsigned long long* net_scans /* per-net scan counts or null */,
    uint8_t* inq_base /* per-state-entry in-queue flag (frontier dedup) */) {
  const int tid = threadIdx.x;
  const bool is_small = (int)blockIdx.x < n_small_slots;
  const int slot = blockIdx.x;

  __shared__ WgShared sh;

  uint64_t* state;
  uint8_t* inq;
  float4* fr[2];
  int32_t* touched;
  int64_t f_cap, t_cap;
  if (is_small) {
    state = state_base + (int64_t)slot * small_cap;
    inq = inq_base + (int64_t)slot * small_cap;
    float4* fb = frontier_base + (int64_t)slot * 2 * f_cap_small;
    fr[0] = fb; fr[1] = fb + f_cap_small;
    touched = touched_base + (int64_t)slot * t_cap_small;
    f_cap = f_cap_small; t_cap = t_cap_small;
  } else {
    int ls = slot - n_small_slots;
    uint64_t* lbase = state_base + (int64_t)n_small_slots * small_cap;
    state = lbase + (int64_t)ls * large_cap;
    inq = inq_base + (int64_t)n_small_slots * small_cap
        + (int64_t)ls * large_cap;
    float4* fb = frontier_base + (int64_t)n_small_slots * 2 * f_cap_small
               + (int64_t)ls * 2 * f_cap_large;
    fr[0] = fb; fr[1] = fb + f_cap_large;
    touched = touched_base + (int64_t)n_small_slots * t_cap_small
            + (int64_t)ls * t_cap_large;
    f_cap = f_cap_large; t_cap = t_cap_large;
  }

  const int64_t s_cap = is_small ? small_cap : large_cap;
  (void)s_cap;
  const int32_t* queue = is_small ? queue_small : queue_large;
  const int32_t n_queue = is_small ? n_queue_small : n_queue_large;
  int32_t* cursor = q_cursors + (is_small ? 0 : 1);

  for (;;) {
    // pop a net
    if (tid == 0) sh.fcnt[0] = atomicAdd(cursor, 1);
    __syncthreads();
    int qi = sh.fcnt[0];
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
    // NOTE: the large class uses GLOBAL indexing (L.dense=false) but keeps
    // the net's own bb for pruning — searching the whole chip for a
    // bb-bounded net was the dominant cost of early profiles.

    // ---- rip-up previous tree (reference: route_tree rip-up, occ -1) ----
    int32_t old_len = trees.len[inet];
#ifdef PNR_DEBUG_BOUNDS
    if (tid == 0 && (old_len < 0 || old_len > tcap))
      printf("rip-up: net %d len %d cap %d\n", inet, old_len, tcap);
#endif
    // partial mode (P.partial, reference route_tree_mark_congested_...):
    // keep subtrees whose root-path avoids overused nodes; fall back to
    // a full rip when the tree is empty/oversized or the root itself is
    // congested. Decided by thread 0 (uniform branch for the workgroup).
    if (tid == 0) {
      int ok = 0;
      if (P.partial && old_len > 0 && old_len <= tcap &&
          old_len <= (int32_t)t_cap) {
        int32_t r = t_node[0];
        if (r >= 0 && r < g.num_nodes && occ[r] <= g.capacity[r]) ok = 1;
      }
      sh.fcnt[1] = ok;
    }
    __syncthreads();
    const bool do_partial = sh.fcnt[1] != 0;
    __syncthreads();
    int tree_len;
    if (!do_partial) {
      for (int k = tid; k < old_len && k < tcap; k += WG_THREADS) {
        int32_t rv = t_node[k];
#ifdef PNR_DEBUG_BOUNDS
        if (rv < 0 || rv >= g.num_nodes) {
          printf("rip-up: net %d k %d bad node %d\n", inet, k, rv);
          continue;
        }
#endif
        atomicSub(&occ[rv], 1);
      }
      __syncthreads();
      // ---- new tree root ----
      if (tid == 0) {
        t_node[0] = src; t_parent[0] = -1; t_sw[0] = -1; t_delay[0] = 0.0f;
        trees.len[inet] = 1;
        atomicAdd(&occ[src], 1);
        sh.fail = 0;
        sh.dbg = 0;
      }
      __syncthreads();
      tree_len = 1;
    } else {
      // serial drop + in-place compaction on thread 0 (endgame trees are
      // short; touched[] doubles as the old-index -> new-index remap)
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
          if ((k > 0 && (pdrop || cong)) ) {
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
        sh.fcnt[0] = keep;
        sh.fail = 0;
        sh.dbg = 0;
      }
      __syncthreads();
      tree_len = sh.fcnt[0];
      __syncthreads();
    }

    // ---- route each sink (pre-ordered by criticality on host) ----
    for (int32_t si = s0; si < s1; ++si) {
      SinkCtx S;
      S.sink_node = nets.sink_rr[si];
      S.sx = g.xlow[S.sink_node];
      S.sy = g.ylow[S.sink_node];
      S.crit = nets.crit[si];
      S.astar_fac = P.astar_fac;
      if (do_partial) {
        // skip sinks whose kept subtree still reaches them
        if (tid == 0) sh.fcnt[1] = 0;
        __syncthreads();
        for (int k = tid; k < tree_len; k += WG_THREADS)
          if (t_node[k] == S.sink_node) atomicOr(&sh.fcnt[1], 1);
        __syncthreads();
        const bool connected = sh.fcnt[1] != 0;
        __syncthreads();
        if (connected) continue;
      }
      // per-sink bucket width: delta_fac edge-steps in this sink's cost
      // units (wider buckets = fewer, fatter delta-stepping rounds;
      // PathFinder tolerates the relaxed expansion order)
      const float delta = P.delta_fac * (S.crit * P.seg_delay +
                                         (1.0f - S.crit) * P.seg_base);

      if (tid == 0) {
        sh.fcnt[0] = 0; sh.fcnt[1] = 0;
        sh.fmin_next = 0xffffffffu;
        sh.best_sink_back = 0xffffffffu;
        sh.touched_cnt = 0;
        sh.path_len = -1;
      }
      __syncthreads();

      // seed from the current route tree (reference: SinkRouter seeds
      // from tree nodes inside bb, partitioning_multi_sink...:707-745)
      for (int k = tid; k < tree_len; k += WG_THREADS) {
        int32_t v = t_node[k];
#ifdef PNR_DEBUG_BOUNDS
        if (v < 0 || v >= g.num_nodes) {
          atomicOr(&sh.dbg, 1 << 10);
          printf("SEEDJUNK net=%d k=%d v=0x%08x tree_len=%d si=%d\n",
                 inet, k, (unsigned)v, tree_len, (int)(si - s0));
          continue;
        }
#endif
        if (g.type[v] == 1 /*SINK*/) continue;
        if (!L.in_bb(g, v)) continue;
        float back = S.crit * t_delay[k];
        float tot = back + S.astar_fac * expected_cost(g, P, v, S);
        int64_t li = L(g, v);
        // prev==self marks a tree seed; stored back 0.0 makes the seed
        // UNBEATABLE by any ordinary entry (all edge costs > 0), so the
        // backtrack's prev==self stop condition is stable and a path can
        // never re-enter the tree (which would duplicate tree nodes and
        // leave phantom occupancy). The frontier entry carries the REAL
        // back cost for expansion.
        state[li] = pack_state(0.0f, v);
        inq[li] = 1;
        int ti = atomicAdd(&sh.touched_cnt, 1);
        if (ti < t_cap) touched[ti] = (int32_t)li;
        int fi = atomicAdd(&sh.fcnt[0], 1);
        if (fi < f_cap)
          fr[0][fi] = make_float4(tot, back, __int_as_float(v), __int_as_float(v));
        atomicMin(&sh.fmin_next, f32_bits(tot));
      }
      __syncthreads();
      if (sh.fcnt[0] > f_cap) { if (tid == 0) sh.fail = FAIL_FRONTIER; }
      __syncthreads();

      int cur = 0;
      int n_cur = sh.fcnt[0];
      unsigned fmin = sh.fmin_next;
      int rounds = 0;
      int64_t scanned = 0;
      (void)net_scans;

      while (!sh.fail) {
        if (n_cur == 0) { if (tid == 0 && sh.best_sink_back == 0xffffffffu) sh.fail = FAIL_NO_PATH; break; }
        // terminate when the sink's settled cost beats the min frontier f.
        // Deterministic mode uses STRICT <: the == bucket must be fully
        // processed so equal-cost tie-breaks reach their fixpoint (with an
        // admissible heuristic the result is then order-independent).
        // Normal mode uses <= — at pres_fac 0 the fabric is full of exact
        // cost ties and processing every tie plateau explodes the frontier.
        if (sh.best_sink_back != 0xffffffffu &&
            (P.strict_term ? (sh.best_sink_back < fmin)
                           : (sh.best_sink_back <= fmin))) break;
        if (++rounds > P.max_rounds) { if (tid == 0) sh.fail = FAIL_ROUNDS; break; }
        const float thr = bits_f32(fmin) + delta;
        const int nxt = cur ^ 1;
        if (tid == 0) { sh.fcnt[nxt] = 0; sh.fmin_next = 0xffffffffu; }
        __syncthreads();

        for (int i = tid; i < n_cur; i += WG_THREADS) {
          float4 e = fr[cur][i];
          float tot = e.x, back = e.y;
          int32_t v = DBG_NODE(__float_as_int(e.z), 95);
          int32_t prev = __float_as_int(e.w);
          int64_t li = DBG_LI(L(g, v), s_cap, 91);
          // stale check: seeds (prev==self) store back 0.0 in state while
          // their entry carries the real back cost
          const uint64_t expect = (prev == v) ? pack_state(0.0f, v)
                                              : pack_state(back, prev);
          const uint64_t st = load_state(&state[li]);
          if (st != expect) {
            // stale: the in-queue dedup keeps exactly ONE live entry per
            // node, so REPAIR it from the current state instead of
            // dropping it (a drop would lose the node's only entry)
            int32_t prev2 = (int32_t)(st & 0xffffffffu);
            if (st == INF_STATE || prev2 == v) continue;
            back = bits_f32((uint32_t)(st >> 32));
            prev = prev2;
            tot = back + S.astar_fac * expected_cost(g, P, v, S);
            e = make_float4(tot, back, __int_as_float(v),
                            __int_as_float(prev));
          }
          if (tot > thr) {
            // keep for a later bucket (stays in-queue)
            int fi = atomicAdd(&sh.fcnt[nxt], 1);
            if (fi < f_cap) fr[nxt][fi] = e;
            atomicMin(&sh.fmin_next, f32_bits(tot));
            continue;
          }
          // leaving the queue: clear the membership flag with an
          // L1-bypassing store so a racing relaxer's atomicExch sees it
          __hip_atomic_store(&inq[li], (uint8_t)0, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
          if (v == S.sink_node) continue;  // settled sink; no expansion
          // expand
          int32_t e0 = g.row_ptr[v], e1 = g.row_ptr[v + 1];
          for (int32_t ei = e0; ei < e1; ++ei) {
            int32_t w = g.edge_dst[ei];
#ifdef PNR_DEBUG_BOUNDS
            if (w < 0 || w >= g.num_nodes) {
              atomicOr(&sh.dbg, 1 << 11);
              printf("EDGEJUNK net=%d v=%d ei=%d e0=%d e1=%d w=0x%08x\n",
                     inet, v, ei, e0, e1, (unsigned)w);
              continue;
            }
#endif
            int8_t ty = g.type[w];
            if (ty == 1 && w != S.sink_node) continue;       // other SINK
            if (ty == 3 /*IPIN*/ && (g.xlow[w] != S.sx || g.ylow[w] != S.sy))
              continue;                                      // wrong-tile IPIN
            if (!L.in_bb(g, w)) continue;                    // bb prune
            int8_t sw = g.edge_sw[ei];
            float back_new = back + S.crit * hop_delay(g, sw, w) +
                             (1.0f - S.crit) * cong_cost(g, P, occ, acc, w);
            float tot_new = back_new + S.astar_fac * expected_cost(g, P, w, S);
            int64_t lw = DBG_LI(L(g, w), s_cap, 92);
            uint64_t pk = pack_state(back_new, v);
            uint64_t old = atomicMin((unsigned long long*)&state[lw],
                                     (unsigned long long)pk);
            if (pk < old) {
              if (old == INF_STATE) {
                int ti = atomicAdd(&sh.touched_cnt, 1);
                if (ti < t_cap) touched[ti] = (int32_t)lw;
              }
              // fmin/best always track the improvement; the entry is
              // pushed only if the node is not already queued (dedup —
              // the queued entry is repaired from state at pop)
              atomicMin(&sh.fmin_next, f32_bits(tot_new));
              if (w == S.sink_node) atomicMin(&sh.best_sink_back, f32_bits(back_new));
              if (__hip_atomic_exchange(&inq[lw], (uint8_t)1,
                                        __ATOMIC_RELAXED,
                                        __HIP_MEMORY_SCOPE_AGENT) == 0) {
                int fi = atomicAdd(&sh.fcnt[nxt], 1);
                if (fi < f_cap)
                  fr[nxt][fi] = make_float4(tot_new, back_new,
                                            __int_as_float(w), __int_as_float(v));
              }
            }
          }
        }
        __syncthreads();
        if (tid == 0) {
          if (sh.fcnt[nxt] > f_cap) sh.fail = FAIL_FRONTIER;
          if (sh.touched_cnt > t_cap) sh.fail = FAIL_TOUCHED;
        }
        scanned += n_cur;
        __syncthreads();
        n_cur = min((int64_t)sh.fcnt[nxt], f_cap);
        fmin = sh.fmin_next;
        cur = nxt;
      }
      __syncthreads();
      if (tid == 0 && stats) {
        atomicAdd(&stats[0], (unsigned long long)rounds);
        atomicAdd(&stats[1], (unsigned long long)scanned);
        atomicAdd(&stats[2], 1ull);  // sinks attempted
        atomicAdd(&stats[3], (unsigned long long)sh.touched_cnt);
        if (net_scans) atomicAdd(&net_scans[inet], (unsigned long long)scanned);
      }

      // ---- backtrack + commit (reference: backtrack
      //      partitioning_multi_sink...:613-680 + route_tree add) ----
      if (!sh.fail && tid == 0) {
        int n = 0;
        int32_t v = S.sink_node;
        for (;;) {
          v = DBG_NODE(v, 94);
          uint64_t st = load_state(&state[(size_t)DBG_LI(L(g, v), s_cap, 93)]);
          // belt-and-braces: a torn/stale INF would wild-walk; stop instead
          if (st == INF_STATE) { sh.fail = FAIL_NO_PATH; break; }
          int32_t prev = (int32_t)(st & 0xffffffffu);
#ifdef PNR_DEBUG_BOUNDS
          if (sh.dbg) { sh.fail = FAIL_NO_PATH; break; }  // tid0-only loop
#endif
          if (prev == v) break;  // reached a tree seed
          if (n >= PATH_CAP) { sh.fail = FAIL_PATH_CAP; break; }
          sh.path[n++] = v;
          v = prev;
        }
        sh.path_len = n;
        sh.attach_idx = -1;
        // stash attach node in fmin_next slot for the parallel scan
        sh.fmin_next = (unsigned)v;
      }
      __syncthreads();
      if (sh.fail) break;
      // parallel scan for the attach node's tree index
      {
        int32_t attach_node = (int32_t)sh.fmin_next;
        for (int k = tid; k < tree_len; k += WG_THREADS)
          if (t_node[k] == attach_node) sh.attach_idx = k;
      }
      __syncthreads();
      if (tid == 0) {
        int ai = sh.attach_idx;
        if (ai < 0) { sh.fail = FAIL_NO_PATH; }
        else {
          float dacc = t_delay[ai];
          int parent = ai;
          int len = tree_len;
          for (int k = sh.path_len - 1; k >= 0; --k) {
            int32_t u = sh.path[k];
            int32_t pu = t_node[parent];
#ifdef PNR_DEBUG_BOUNDS
            if (u < 0 || u >= g.num_nodes || pu < 0 || pu >= g.num_nodes) {
              sh.fail = FAIL_NO_PATH;  // tid0-only loop: safe to abort
              sh.dbg |= 1 << 9;
              printf("commit: net %d bad u=%d pu=%d\n", inet, u, pu);
              break;
            }
#endif
            // find the switch of edge pu->u
            int8_t sw = 0;
            for (int32_t ei = g.row_ptr[pu]; ei < g.row_ptr[pu + 1]; ++ei)
              if (g.edge_dst[ei] == u) { sw = g.edge_sw[ei]; break; }
            dacc += hop_delay(g, sw, u);
            if (len >= tcap) { sh.fail = FAIL_TREE_CAP; break; }
            t_node[len] = u; t_parent[len] = parent; t_sw[len] = sw;
            t_delay[len] = dacc;
            parent = len;
            ++len;
            atomicAdd(&occ[u], 1);
          }
          tree_len = len;
          trees.len[inet] = len;  // always accurate, even on FAIL_TREE_CAP,
                                  // so the retry's rip-up stays balanced
          if (!sh.fail) trees.sink_delay[nets.sink_orig[si]] = dacc;
        }
      }
      __syncthreads();
      if (sh.fail) break;
      // broadcast updated tree_len to all threads
      if (tid == 0) sh.fcnt[0] = tree_len;
      __syncthreads();
      tree_len = sh.fcnt[0];

      // ---- sparse state reset (touched list) ----
      int nt = min((int64_t)sh.touched_cnt, t_cap);
      for (int k = tid; k < nt; k += WG_THREADS) {
        size_t li = (size_t)(uint32_t)touched[k];
        state[li] = INF_STATE;
        inq[li] = 0;
      }
      __syncthreads();
    }

    if (tid == 0) {
      if (sh.dbg) fail_flags[inet] = 64 + sh.dbg;   // debug-detect bits
      else if (sh.fail) fail_flags[inet] = sh.fail;
    }
    __syncthreads();
    // if this net failed mid-sink, its touched entries were reset above only
    // on success; do a full reset of touched here for safety
    if (sh.fail) {
      int nt = min((int64_t)sh.touched_cnt, t_cap);
      for (int k = tid; k < nt; k += WG_THREADS) {
        size_t li = (size_t)(uint32_t)touched[k];
        state[li] = INF_STATE;
        inq[li] = 0;
      }
      // if the touched list overflowed, fall back to a full clear
      if (sh.touched_cnt > t_cap) {
        int64_t cap = is_small ? small_cap : large_cap;
        for (int64_t k = tid; k < cap; k += WG_THREADS) {
          state[k] = INF_STATE;
          inq[k] = 0;
        }
      }
      __syncthreads();
    }
  }
}

// ---------------- congestion sweeps ----------------
// reference: congestion.h:176-193 update_costs — acc_cost += overuse*acc_fac
__global__ void update_acc_kernel(const int32_t* __restrict__ occ,
                                  const int16_t* __restrict__ cap,
                                  float* acc, float acc_fac, int32_t n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += gridDim.x * blockDim.x) {
    int over = occ[i] - cap[i];
    if (over > 0) acc[i] += over * acc_fac;
  }
}

// overuse census (reference: feasible_routing router.h:103 + census)
__global__ void overuse_count_kernel(const int32_t* __restrict__ occ,
                                     const int16_t* __restrict__ cap,
                                     const int8_t* __restrict__ type,
                                     int32_t* out /*[8]*/, int32_t n) {
  __shared__ int32_t cnt[8];
  if (threadIdx.x < 8) cnt[threadIdx.x] = 0;
  __syncthreads();
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += gridDim.x * blockDim.x) {
    if (occ[i] > cap[i]) atomicAdd(&cnt[type[i]], 1);
  }
  __syncthreads();
  if (threadIdx.x < 8 && cnt[threadIdx.x])
    atomicAdd(&out[threadIdx.x], cnt[threadIdx.x]);
}

// flag nets whose current tree touches an overused node (selective
// reroute set; reference: phase-two congested-nets-only rebuild,
// partitioning_multi_sink...cxx:5530 build_phase_two)
__global__ void flag_congested_nets_kernel(TreesDev trees,
                                           const int32_t* __restrict__ occ,
                                           const int16_t* __restrict__ cap,
                                           int32_t num_nets,
                                           uint8_t* __restrict__ out) {
  int inet = blockIdx.x;
  if (inet >= num_nets) return;
  __shared__ int flag;
  if (threadIdx.x == 0) flag = 0;
  __syncthreads();
  int64_t off = trees.off[inet];
  int32_t len = trees.len[inet];
  for (int k = threadIdx.x; k < len && !flag; k += blockDim.x) {
    int32_t v = trees.node[off + k];
    if (occ[v] > cap[v]) flag = 1;
  }
  __syncthreads();
  if (threadIdx.x == 0) out[inet] = (uint8_t)flag;
}

// rip up listed nets (occ -1, len 0) without rerouting — ownership
// hand-off for load rebalancing (reference: move_route_tree analogue)
__global__ void rip_up_nets_kernel(TreesDev trees, const int32_t* __restrict__ ids,
                                   int32_t n, int32_t* occ) {
  int i = blockIdx.x;
  if (i >= n) return;
  int32_t inet = ids[i];
  int64_t off = trees.off[inet];
  int32_t len = trees.len[inet];
  for (int k = threadIdx.x; k < len; k += blockDim.x)
    atomicSub(&occ[trees.node[off + k]], 1);
  __syncthreads();
  if (threadIdx.x == 0) trees.len[inet] = 0;
}

// occupancy recount from route trees (debug cross-check; reference:
// recalculate_occ partitioning_multi_sink...:6194-6216)
__global__ void recount_occ_kernel(TreesDev trees, const int32_t* __restrict__ net_ids,
                                   int32_t n_nets, int32_t* recount) {
  int inet_i = blockIdx.x;
  if (inet_i >= n_nets) return;
  int32_t inet = net_ids[inet_i];
  int64_t off = trees.off[inet];
  int32_t len = trees.len[inet];
  for (int k = threadIdx.x; k < len; k += blockDim.x)
    atomicAdd(&recount[trees.node[off + k]], 1);
}

}  // namespace pnrh

// ---------------- C ABI launchers ----------------
using namespace pnrh;

extern "C" {

int pnr_route_nets(const RouteLaunchArgs* a, void* stream) {
  RRDev g{a->type, a->xlow, a->ylow, a->xhigh, a->yhigh, a->capacity,
          a->R, a->C, a->row_ptr, a->edge_dst, a->edge_sw,
          a->sw_R, a->sw_Tdel, a->base_cost, a->idx_in_tile,
          a->num_nodes, a->nx, a->ny, a->L, a->npt};
  NetsDev nets{a->net_src, a->sink_ptr, a->sink_rr, a->crit, a->sink_orig,
               a->bb, a->num_nets};
  TreesDev trees{a->tree_off, a->tree_node, a->tree_parent, a->tree_sw,
                 a->tree_delay, a->tree_len, a->sink_delay};
  RouteParams P{};
  P.astar_fac = a->astar_fac; P.pres_fac = a->pres_fac;
  P.seg_delay = a->seg_delay; P.ipin_delay = a->ipin_delay;
  P.seg_base = a->seg_base; P.ipin_base = a->ipin_base;
  P.delta_fac = a->delta_fac;
  P.cong_mult = a->cong_mult < 1.0f ? 1.0f : a->cong_mult;
  P.max_rounds = a->max_rounds;
  P.strict_term = a->strict_term;
  P.partial = a->partial;
  int grid = a->n_small_slots + a->n_large_slots;
  if (a->use_calendar) {
    hipLaunchKernelGGL(route_nets_cal_kernel, dim3(grid), dim3(WG_THREADS), 0,
                       (hipStream_t)stream,
                       g, nets, trees, P,
                       a->queue_small, a->n_queue_small,
                       a->queue_large, a->n_queue_large,
                       a->q_cursors, a->occ, a->acc,
                       a->state_base, a->small_cap, a->large_cap,
                       a->n_small_slots,
                       a->frontier_base, a->f_cap_small, a->f_cap_large,
                       a->touched_base, a->t_cap_small, a->t_cap_large,
                       a->fail_flags, a->stats, a->net_scans);
    return (int)hipGetLastError();
  }
  hipLaunchKernelGGL(route_nets_kernel, dim3(grid), dim3(WG_THREADS), 0,
                     (hipStream_t)stream,
                     g, nets, trees, P,
                     a->queue_small, a->n_queue_small,
                     a->queue_large, a->n_queue_large,
                     a->q_cursors, a->occ, a->acc,
                     a->state_base, a->small_cap, a->large_cap,
                     a->n_small_slots,
                     a->frontier_base, a->f_cap_small, a->f_cap_large,
                     a->touched_base, a->t_cap_small, a->t_cap_large,
                     a->fail_flags, a->stats, a->net_scans, a->inq_base);
  return (int)hipGetLastError();
}

int pnr_update_acc(const int32_t* occ, const int16_t* cap, float* acc,
                   float acc_fac, int32_t n, void* stream) {
  int grid = min((n + 255) / 256, 2048);
  hipLaunchKernelGGL(update_acc_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, occ, cap, acc, acc_fac, n);
  return (int)hipGetLastError();
}

int pnr_overuse_count(const int32_t* occ, const int16_t* cap, const int8_t* type,
                      int32_t* out, int32_t n, void* stream) {
  int grid = min((n + 255) / 256, 2048);
  hipLaunchKernelGGL(overuse_count_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, occ, cap, type, out, n);
  return (int)hipGetLastError();
}

int pnr_recount_occ(const int64_t* tree_off, int32_t* tree_node, int32_t* tree_len,
                    const int32_t* net_ids, int32_t n_nets, int32_t* recount,
                    void* stream) {
  TreesDev t{};
  t.off = tree_off; t.node = tree_node; t.len = tree_len;
  hipLaunchKernelGGL(recount_occ_kernel, dim3(n_nets), dim3(256), 0,
                     (hipStream_t)stream, t, net_ids, n_nets, recount);
  return (int)hipGetLastError();
}

int64_t pnr_route_args_sizeof() { return (int64_t)sizeof(RouteLaunchArgs); }

}  // extern "C"

namespace pnrh {
__global__ void fill_u64_kernel(uint64_t* p, uint64_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (int64_t)gridDim.x * blockDim.x) p[i] = v;
}
}

extern "C" int pnr_rip_up_nets(const int64_t* tree_off, int32_t* tree_node,
                               int32_t* tree_len, const int32_t* ids,
                               int32_t n, int32_t* occ, void* stream) {
  if (n <= 0) return 0;
  TreesDev t{};
  t.off = tree_off; t.node = tree_node; t.len = tree_len;
  hipLaunchKernelGGL(pnrh::rip_up_nets_kernel, dim3(n), dim3(256), 0,
                     (hipStream_t)stream, t, ids, n, occ);
  return (int)hipGetLastError();
}

extern "C" int pnr_flag_congested_nets(const int64_t* tree_off,
                                       int32_t* tree_node, int32_t* tree_len,
                                       const int32_t* occ, const int16_t* cap,
                                       int32_t num_nets, uint8_t* out,
                                       void* stream) {
  TreesDev t{};
  t.off = tree_off; t.node = tree_node; t.len = tree_len;
  hipLaunchKernelGGL(pnrh::flag_congested_nets_kernel, dim3(num_nets),
                     dim3(256), 0, (hipStream_t)stream, t, occ, cap,
                     num_nets, out);
  return (int)hipGetLastError();
}

extern "C" int pnr_fill_u64_launch(uint64_t* p, uint64_t v, int64_t n, void* stream) {
  int64_t g64 = (n + 255) / 256;
  int grid = (int)(g64 < 4096 ? g64 : 4096);
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(pnrh::fill_u64_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, p, v, n);
  return (int)hipGetLastError();
}
