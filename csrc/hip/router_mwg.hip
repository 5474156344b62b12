// Multi-workgroup cooperative single-net router (straggler engine).
//
// The per-net workgroup router (router_kernel.hip) is the right shape for
// thousands of concurrent small-bb nets, but the contested ENDGAME is a
// handful of nets whose bounding boxes have grown to a large fraction of
// the chip: one 256-thread workgroup grinding a 12M-node search space
// serializes the whole iteration (measured: 106 s for 123 nets at
// bitcoin_miner scale, 1/3 of total route time). This engine routes ONE
// net at a time with the WHOLE GPU: the delta-stepping bucket loop runs as
// a sequence of grid-wide round kernels over a global frontier, so a
// straggler's frontier (10^4-10^6 entries) spreads across all 256 CUs
// instead of 4 wave64s. Rounds are enqueued in stream-ordered batches
// (round + advance pairs); the host syncs once per batch to test the
// device-computed done flag — no cooperative-launch machinery, identical
// cost/termination semantics to the workgroup kernel's normal mode.
//
// Reference analogue: the reference shrinks the MPI communicator when the
// endgame no longer fills the machine (mpi_comm_shrink); on one MI355X the
// equivalent is inverting the parallelism axis from nets to nodes.
#include "pnr_hip.h"

namespace pnrh {
// defined in router_kernel.hip (same library)
__global__ void rip_up_nets_kernel(TreesDev trees, const int32_t* ids,
                                   int32_t n, int32_t* occ);
__global__ void fill_u64_kernel(uint64_t* p, uint64_t v, int64_t n);

#define MWG_THREADS 256
#define MWG_GRID 1024
#define MWG_PATH_CAP 8192
#define MWG_FAIL_FRONTIER 1
#define MWG_FAIL_ROUNDS 2
#define MWG_FAIL_NO_PATH 3
#define MWG_FAIL_TREE_CAP 4
#define MWG_FAIL_PATH_CAP 5

#define MWG_INF 0xffffffffu
#define MWG_INF_STATE 0xffffffffffffffffull

// ctrl block layout (uint32[16], device)
// 0 n_cur   1 n_next   2 fmin_cur(bits)  3 fmin_next(bits)
// 4 best_sink_back(bits)  5 fail  6 rounds  7 done  8 cur_buf
// 9 net_fail (sticky across sinks)

__device__ __forceinline__ uint32_t ld_ctrl(const uint32_t* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__global__ void mwg_set_root_kernel(TreesDev trees, NetsDev nets,
                                    int32_t inet, int32_t* occ,
                                    uint32_t* ctrl) {
  int32_t src = nets.src[inet];
  int64_t off = trees.off[inet];
  trees.node[off] = src; trees.parent[off] = -1;
  trees.sw[off] = -1; trees.delay[off] = 0.0f;
  trees.len[inet] = 1;
  atomicAdd(&occ[src], 1);
  ctrl[9] = 0;
}

__global__ void mwg_init_ctrl_kernel(uint32_t* ctrl) {
  ctrl[0] = 0; ctrl[1] = 0;
  ctrl[2] = MWG_INF; ctrl[3] = MWG_INF; ctrl[4] = MWG_INF;
  ctrl[5] = 0; ctrl[6] = 0; ctrl[7] = 0; ctrl[8] = 0;
}

// seed the frontier from the current route tree (same semantics as the
// workgroup kernel's per-sink seeding; state uses GLOBAL node indexing)
__global__ void mwg_seed_kernel(RRDev g, NetsDev nets, TreesDev trees,
                                RouteParams P, int32_t inet, int32_t si,
                                uint64_t* state, uint8_t* inq,
                                float4* frA, int64_t f_cap,
                                uint32_t* ctrl) {
  SinkCtx S;
  S.sink_node = nets.sink_rr[si];
  S.sx = g.xlow[S.sink_node];
  S.sy = g.ylow[S.sink_node];
  S.crit = nets.crit[si];
  S.astar_fac = P.astar_fac;
  LocalIdx L;
  L.dense = false;
  L.bx0 = nets.bb[4 * inet + 0];
  L.by0 = nets.bb[4 * inet + 1];
  L.bw = nets.bb[4 * inet + 2] - L.bx0 + 1;
  L.bh = nets.bb[4 * inet + 3] - L.by0 + 1;
  L.npt = g.npt;
  int64_t off = trees.off[inet];
  int32_t len = trees.len[inet];
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < len; i += gridDim.x * blockDim.x) {
    int32_t v = trees.node[off + i];
    if (g.type[v] == 1 /*SINK*/) continue;
    if (!L.in_bb(g, v)) continue;
    float back = S.crit * trees.delay[off + i];
    float tot = back + S.astar_fac * expected_cost(g, P, v, S);
    state[v] = pack_state(0.0f, v);   // seed: unbeatable, prev==self
    inq[v] = 1;
    uint32_t fi = atomicAdd(&ctrl[0], 1u);
    if (fi < (uint32_t)f_cap)
      frA[fi] = make_float4(tot, back, __int_as_float(v), __int_as_float(v));
    atomicMin(&ctrl[2], f32_bits(tot));
  }
}

__global__ __launch_bounds__(MWG_THREADS, 2)
void mwg_round_kernel(RRDev g, NetsDev nets, RouteParams P,
                      int32_t inet, int32_t si, uint64_t* state,
                      uint8_t* inq,
                      float4* frA, float4* frB, int64_t f_cap,
                      uint32_t* ctrl, const int32_t* __restrict__ occ,
                      const float* __restrict__ acc) {
  if (ld_ctrl(&ctrl[7])) return;   // done (batched enqueue over-runs)
  const uint32_t n_cur = min(ld_ctrl(&ctrl[0]), (uint32_t)f_cap);
  if (n_cur == 0) return;
  SinkCtx S;
  S.sink_node = nets.sink_rr[si];
  S.sx = g.xlow[S.sink_node];
  S.sy = g.ylow[S.sink_node];
  S.crit = nets.crit[si];
  S.astar_fac = P.astar_fac;
  LocalIdx L;
  L.dense = false;
  L.bx0 = nets.bb[4 * inet + 0];
  L.by0 = nets.bb[4 * inet + 1];
  L.bw = nets.bb[4 * inet + 2] - L.bx0 + 1;
  L.bh = nets.bb[4 * inet + 3] - L.by0 + 1;
  L.npt = g.npt;
  const float delta = P.delta_fac * (S.crit * P.seg_delay +
                                     (1.0f - S.crit) * P.seg_base);
  const float thr = bits_f32(ld_ctrl(&ctrl[2])) + delta;
  const int cur = (int)ld_ctrl(&ctrl[8]);
  const float4* __restrict__ fin = cur ? frB : frA;
  float4* __restrict__ fout = cur ? frA : frB;

  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n_cur; i += gridDim.x * blockDim.x) {
    float4 e = fin[i];
    float tot = e.x, back = e.y;
    int32_t v = __float_as_int(e.z);
    int32_t prev = __float_as_int(e.w);
    const uint64_t expect = (prev == v) ? pack_state(0.0f, v)
                                        : pack_state(back, prev);
    const uint64_t st = load_state(&state[v]);
    if (st != expect) {
      // stale: repair from current state (dedup keeps one entry/node)
      int32_t prev2 = (int32_t)(st & 0xffffffffu);
      if (st == MWG_INF_STATE || prev2 == v) continue;
      back = bits_f32((uint32_t)(st >> 32));
      prev = prev2;
      tot = back + S.astar_fac * expected_cost(g, P, v, S);
      e = make_float4(tot, back, __int_as_float(v), __int_as_float(prev));
    }
    if (tot > thr) {   // keep for a later bucket (stays in-queue)
      uint32_t fi = atomicAdd(&ctrl[1], 1u);
      if (fi < (uint32_t)f_cap) fout[fi] = e;
      atomicMin(&ctrl[3], f32_bits(tot));
      continue;
    }
    __hip_atomic_store(&inq[v], (uint8_t)0, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    if (v == S.sink_node) continue;   // settled sink; no expansion
    int32_t e0 = g.row_ptr[v], e1 = g.row_ptr[v + 1];
    for (int32_t ei = e0; ei < e1; ++ei) {
      int32_t w = g.edge_dst[ei];
      int8_t ty = g.type[w];
      if (ty == 1 && w != S.sink_node) continue;
      if (ty == 3 /*IPIN*/ && (g.xlow[w] != S.sx || g.ylow[w] != S.sy))
        continue;
      if (!L.in_bb(g, w)) continue;
      int8_t sw = g.edge_sw[ei];
      float back_new = back + S.crit * hop_delay(g, sw, w) +
                       (1.0f - S.crit) * cong_cost(g, P, occ, acc, w);
      float tot_new = back_new + S.astar_fac * expected_cost(g, P, w, S);
      uint64_t pk = pack_state(back_new, v);
      uint64_t old = atomicMin((unsigned long long*)&state[w],
                               (unsigned long long)pk);
      if (pk < old) {
        atomicMin(&ctrl[3], f32_bits(tot_new));
        if (w == S.sink_node) atomicMin(&ctrl[4], f32_bits(back_new));
        if (__hip_atomic_exchange(&inq[w], (uint8_t)1, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT) == 0) {
          uint32_t fi = atomicAdd(&ctrl[1], 1u);
          if (fi < (uint32_t)f_cap)
            fout[fi] = make_float4(tot_new, back_new,
                                   __int_as_float(w), __int_as_float(v));
        }
      }
    }
  }
}

__global__ void mwg_advance_kernel(uint32_t* ctrl, int64_t f_cap,
                                   int32_t max_rounds,
                                   unsigned long long* net_scans,
                                   int32_t inet) {
  if (ctrl[7]) return;
  if (net_scans) net_scans[inet] += ctrl[0];
  if (ctrl[5]) { ctrl[7] = 1; return; }
  uint32_t n_next = ctrl[1];
  if (n_next > (uint32_t)f_cap) {
    ctrl[5] = MWG_FAIL_FRONTIER; ctrl[7] = 1; return;
  }
  uint32_t best = ctrl[4], fminn = ctrl[3];
  if (n_next == 0) {
    if (best == MWG_INF) ctrl[5] = MWG_FAIL_NO_PATH;
    ctrl[7] = 1; return;
  }
  // normal-mode termination (<=); the deterministic path never uses MWG
  if (best != MWG_INF && best <= fminn) { ctrl[7] = 1; return; }
  if ((int32_t)(++ctrl[6]) > max_rounds) {
    ctrl[5] = MWG_FAIL_ROUNDS; ctrl[7] = 1; return;
  }
  ctrl[0] = n_next; ctrl[1] = 0;
  ctrl[2] = fminn; ctrl[3] = MWG_INF;
  ctrl[8] ^= 1;
}

// one workgroup: backtrack from the sink through state prevs, attach to
// the tree, occ++, record sink delay (same semantics as the workgroup
// kernel's commit, router_kernel.hip:394-468)
__global__ void mwg_backtrack_kernel(RRDev g, NetsDev nets, TreesDev trees,
                                     RouteParams P, int32_t inet, int32_t si,
                                     uint64_t* state, uint32_t* ctrl,
                                     int32_t* occ, int32_t* fail_flags) {
  __shared__ int32_t path[MWG_PATH_CAP];
  __shared__ int path_len;
  __shared__ int attach_idx;
  __shared__ int32_t attach_node;
  const int tid = threadIdx.x;
  if (ctrl[5]) {
    if (tid == 0) { fail_flags[inet] = ctrl[5]; ctrl[9] = ctrl[5]; }
    return;
  }
  int64_t toff = trees.off[inet];
  const int32_t tcap = (int32_t)(trees.off[inet + 1] - toff);
  int32_t* t_node = trees.node + toff;
  int32_t* t_parent = trees.parent + toff;
  int8_t* t_sw = trees.sw + toff;
  float* t_delay = trees.delay + toff;
  int32_t tree_len = trees.len[inet];

  if (tid == 0) {
    int n = 0;
    int32_t v = nets.sink_rr[si];
    int fail = 0;
    for (;;) {
      uint64_t st = load_state(&state[v]);
      if (st == MWG_INF_STATE) { fail = MWG_FAIL_NO_PATH; break; }
      int32_t prev = (int32_t)(st & 0xffffffffu);
      if (prev == v) break;   // tree seed
      if (n >= MWG_PATH_CAP) { fail = MWG_FAIL_PATH_CAP; break; }
      path[n++] = v;
      v = prev;
    }
    path_len = n;
    attach_idx = -1;
    attach_node = v;
    ctrl[5] = fail;
  }
  __syncthreads();
  if (ctrl[5]) {
    if (tid == 0) { fail_flags[inet] = ctrl[5]; ctrl[9] = ctrl[5]; }
    return;
  }
  for (int k = tid; k < tree_len; k += blockDim.x)
    if (t_node[k] == attach_node) attach_idx = k;
  __syncthreads();
  if (tid == 0) {
    int ai = attach_idx;
    int fail = 0;
    if (ai < 0) fail = MWG_FAIL_NO_PATH;
    else {
      float dacc = t_delay[ai];
      int parent = ai;
      int len = tree_len;
      for (int k = path_len - 1; k >= 0; --k) {
        int32_t u = path[k];
        int32_t pu = t_node[parent];
        int8_t sw = 0;
        for (int32_t ei = g.row_ptr[pu]; ei < g.row_ptr[pu + 1]; ++ei)
          if (g.edge_dst[ei] == u) { sw = g.edge_sw[ei]; break; }
        dacc += hop_delay(g, sw, u);
        if (len >= tcap) { fail = MWG_FAIL_TREE_CAP; break; }
        t_node[len] = u; t_parent[len] = parent; t_sw[len] = sw;
        t_delay[len] = dacc;
        parent = len;
        ++len;
        atomicAdd(&occ[u], 1);
      }
      trees.len[inet] = len;
      if (!fail) trees.sink_delay[nets.sink_orig[si]] = dacc;
    }
    if (fail) { fail_flags[inet] = fail; ctrl[9] = fail; }
  }
}

}  // namespace pnrh

using namespace pnrh;

extern "C" {

// Route one net with the whole device. Returns 0 on HIP success (routing
// failures land in fail_flags[inet], same contract as pnr_route_nets).
// state: num_nodes u64 (filled INF by caller once; this function restores
// INF after each sink with a full refill — O(num_nodes) at HBM bandwidth,
// microseconds, cheaper and simpler than a touched list at this scale).
// ctrl: 16 u32 device scratch. rounds_per_batch: rounds enqueued between
// host syncs (stream-ordered; over-enqueued rounds exit on the done flag).
int pnr_mwg_route_net(const RouteLaunchArgs* a, int32_t inet,
                      int32_t s_begin, int32_t s_end,
                      uint64_t* state, uint8_t* inq,
                      float4* frA, float4* frB,
                      int64_t f_cap, uint32_t* ctrl,
                      int32_t rounds_per_batch, void* stream_v) {
  hipStream_t stream = (hipStream_t)stream_v;
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

  // full rip-up of the previous tree, then root reset
  {
    // stage the net id through the ctrl block (reinterpreted) to avoid a
    // host alloc: rip_up wants a device pointer to the id list
    hipError_t rc = hipMemcpyAsync((void*)&ctrl[15], &inet, sizeof(int32_t),
                                   hipMemcpyHostToDevice, stream);
    if (rc != hipSuccess) return (int)rc;
    TreesDev t{};
    t.off = a->tree_off; t.node = a->tree_node; t.len = a->tree_len;
    hipLaunchKernelGGL(rip_up_nets_kernel, dim3(1), dim3(256), 0, stream,
                       t, (const int32_t*)&ctrl[15], 1, a->occ);
    hipLaunchKernelGGL(mwg_set_root_kernel, dim3(1), dim3(1), 0, stream,
                       trees, nets, inet, a->occ, ctrl);
  }

  uint32_t h_ctrl[10];
  for (int32_t si = s_begin; si < s_end; ++si) {
    hipLaunchKernelGGL(mwg_init_ctrl_kernel, dim3(1), dim3(1), 0, stream,
                       ctrl);
    hipLaunchKernelGGL(mwg_seed_kernel, dim3(64), dim3(MWG_THREADS), 0,
                       stream, g, nets, trees, P, inet, si,
                       state, inq, frA, f_cap, ctrl);
    // bucket rounds in batches; device computes done
    for (;;) {
      for (int k = 0; k < rounds_per_batch; ++k) {
        hipLaunchKernelGGL(mwg_round_kernel, dim3(MWG_GRID),
                           dim3(MWG_THREADS), 0, stream,
                           g, nets, P, inet, si, state, inq, frA, frB,
                           f_cap, ctrl, a->occ, a->acc);
        hipLaunchKernelGGL(mwg_advance_kernel, dim3(1), dim3(1), 0, stream,
                           ctrl, f_cap, P.max_rounds, a->net_scans, inet);
      }
      hipError_t rc = hipMemcpyAsync(h_ctrl, ctrl, sizeof(h_ctrl),
                                     hipMemcpyDeviceToHost, stream);
      if (rc != hipSuccess) return (int)rc;
      rc = hipStreamSynchronize(stream);
      if (rc != hipSuccess) return (int)rc;
      if (h_ctrl[7]) break;
    }
    hipLaunchKernelGGL(mwg_backtrack_kernel, dim3(1), dim3(MWG_THREADS), 0,
                       stream, g, nets, trees, P, inet, si, state, ctrl,
                       a->occ, a->fail_flags);
    // restore INF state + clear in-queue flags for the next sink (full
    // refill at HBM bandwidth)
    {
      int64_t n = a->num_nodes;
      int64_t g64 = (n + 255) / 256;
      int grid = (int)(g64 < 4096 ? g64 : 4096);
      hipLaunchKernelGGL(fill_u64_kernel, dim3(grid), dim3(256), 0, stream,
                         state, MWG_INF_STATE, n);
      hipError_t mrc = hipMemsetAsync(inq, 0, (size_t)n, stream);
      if (mrc != hipSuccess) return (int)mrc;
    }
    // sticky net failure? (backtrack can fail) — stop routing its sinks
    hipError_t rc = hipMemcpyAsync(h_ctrl, ctrl, sizeof(h_ctrl),
                                   hipMemcpyDeviceToHost, stream);
    if (rc != hipSuccess) return (int)rc;
    rc = hipStreamSynchronize(stream);
    if (rc != hipSuccess) return (int)rc;
    if (h_ctrl[9]) break;
  }
  return (int)hipGetLastError();
}

}  // extern "C"
