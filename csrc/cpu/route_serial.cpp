// Serial timing-driven PathFinder router — the CPU oracle.
//
// Re-implements the semantics of the reference's canonical router
// (vpr/SRC/route/route_timing.c:85 try_timing_driven_route,
//  :399 timing_driven_route_net; heap in route_common.c:65-81;
//  congestion update semantics of parallel_route/congestion.cxx:296):
// rip-up & reroute every net each iteration, per-sink A* over the rr graph,
// cost = crit*Tdel + (1-crit)*base*acc*pres + astar_fac*lookahead.
// All switches in our fabric are buffered, so per-hop Elmore
// (Tsw + Cnode*(Rsw + Rnode/2)) is the exact Elmore delay of the tree.
//
// The GPU wavefront router (csrc/hip/router_kernel.hip) must match this
// oracle's routing quality (iteration count, wirelength, critical path)
// on the test configs.
#include "pnr.h"
#include <queue>
#include <algorithm>
#include <cstring>
#include <unordered_set>

namespace pnr {

struct RouterOpts {
  float pres_fac_init = 0.5f;
  float pres_fac_mult = 1.3f;
  float acc_fac = 1.0f;
  float astar_fac = 1.2f;
  int max_iters = 50;
};

struct RouteTree {
  // parallel arrays; parent index -1 for root (SOURCE)
  std::vector<int32_t> nodes;
  std::vector<int32_t> parent;   // index into nodes
  std::vector<int8_t> sw;        // switch of edge parent->node
  std::vector<float> delay;      // source->node delay
  void clear() { nodes.clear(); parent.clear(); sw.clear(); delay.clear(); }
};

class SerialRouter {
 public:
  SerialRouter(const RRGraph* g, std::vector<int32_t> net_src,
               std::vector<int64_t> sink_ptr, std::vector<int32_t> sinks,
               RouterOpts opts)
      : g_(g), net_src_(std::move(net_src)), sink_ptr_(std::move(sink_ptr)),
        sinks_(std::move(sinks)), opts_(opts) {
    int n = g_->num_nodes;
    occ_.assign(n, 0);
    pres_.assign(n, 1.0f);
    acc_.assign(n, 1.0f);
    path_cost_.assign(n, 1e30f);
    back_cost_.assign(n, 1e30f);
    prev_node_.assign(n, -1);
    prev_sw_.assign(n, -1);
    R_up_.assign(n, 0.0f);
    tree_mark_.assign(n, -1);
    trees_.resize(net_src_.size());
    pres_fac_ = 0.0f;  // first iteration routes with pres_fac 0 (VPR style)
    // per-hop expected segment delay for the lookahead
    float Rw = 0, Cw = 0;
    for (int i = 0; i < n; ++i)
      if (g_->type[i] == CHANX) { Rw = g_->R[i]; Cw = g_->C[i]; break; }
    float Rsw = g_->sw_R[SW_SB], Tsw = g_->sw_Tdel[SW_SB];
    seg_delay_ = Tsw + Cw * (Rsw + 0.5f * Rw);
    ipin_delay_ = g_->sw_Tdel[SW_IPIN];
    seg_base_ = g_->base_cost[CHANX];
  }

  int num_nets() const { return (int)net_src_.size(); }
  int num_sinks_total() const { return (int)sinks_.size(); }

  // One rip-up-and-reroute pass over all nets. crit: per-sink criticality
  // aligned with sinks_. Returns number of overused rr nodes after the pass.
  int64_t route_iteration(const float* crit) {
    heap_pushes_ = heap_pops_ = 0;
    unrouted_sinks_ = 0;
    for (int inet = 0; inet < num_nets(); ++inet) route_net(inet, crit);
    return count_overused();
  }

  // sinks the last pass could not reach at all (disconnected fabric or
  // capacity-saturated SOURCE class) — must be surfaced, never silent
  int64_t unrouted_sinks() const { return unrouted_sinks_; }

  // Route only the given nets (distributed partitioning / selective reroute;
  // reference: mpi router routes partition_nets[rank], mpi_route...cxx:936).
  int64_t route_subset(const float* crit, const int32_t* ids, int64_t n) {
    for (int64_t i = 0; i < n; ++i) route_net(ids[i], crit);
    return count_overused();
  }

  // Partial rip-up + incremental reroute (reference:
  // route_tree_mark_congested_nodes_to_be_ripped family,
  // route_tree.h:157-165): drop every subtree whose root-path crosses an
  // overused node, keep the clean remainder as seeds, and re-route only
  // the sinks that lost their path. Full rip when the SOURCE is congested.
  // crit_rip_thr: additionally rip the private path of any sink whose
  // criticality meets the threshold, so near-critical connections are
  // re-optimized as slack tightens (reference: timing-driven incremental
  // rerouting of critical connections); > 1 disables.
  int64_t route_subset_incremental(const float* crit, const int32_t* ids,
                                   int64_t n, float crit_rip_thr = 2.0f) {
    unrouted_sinks_ = 0;
    for (int64_t i = 0; i < n; ++i)
      route_net_incremental(ids[i], crit, crit_rip_thr);
    return count_overused();
  }

  // Sink-parallel (virtual-net) routing oracle for the round-2 GPU
  // kernel (reference: MultiSinkParallelRouter :975-1064 routes >16-sink
  // nets' sinks independently from the source and merges the local
  // trees, merge :880). Each cluster routes BLIND — it sees other nets'
  // congestion but not its siblings' paths, exactly like concurrent
  // workgroups would — then the cluster trees merge with first-wins
  // parents and single-counted occupancy.
  // grp_ptr/grp: CSR of sink positions (0-based within the net).
  void route_net_sink_parallel(int inet, const float* crit_flat,
                               const int32_t* grp_ptr, const int32_t* grp,
                               int n_grp) {
    RouteTree& tree = trees_[inet];
    for (int32_t v : tree.nodes) update_one_cost(v, -1);
    tree.clear();
    int32_t src = net_src_[inet];
    int64_t s0 = sink_ptr_[inet];
    if ((int64_t)sink_delays_.size() < sink_ptr_[inet + 1])
      sink_delays_.resize(sinks_.size(), 0.f);
    // 1) each cluster routes from a private source-only tree; its own
    //    occupancy is withdrawn afterwards so siblings cannot see it
    std::vector<RouteTree> sub((size_t)n_grp);
    for (int gi = 0; gi < n_grp; ++gi) {
      RouteTree& t = sub[gi];
      t.nodes.push_back(src); t.parent.push_back(-1);
      t.sw.push_back(-1); t.delay.push_back(0.0f);
      update_one_cost(src, +1);
      for (int32_t k = grp_ptr[gi]; k < grp_ptr[gi + 1]; ++k) {
        int si = grp[k];
        int32_t sink = sinks_[s0 + si];
        float crit = crit_flat ? crit_flat[s0 + si] : 0.0f;
        sink_delays_[s0 + si] = route_one_sink(inet, sink, crit, t);
      }
      for (int32_t v : t.nodes) update_one_cost(v, -1);
    }
    // 2) merge with first-wins parents (tree order guarantees a node's
    //    parent is already merged), occupancy once per union node
    std::unordered_map<int32_t, int32_t> pos;
    tree.nodes.push_back(src); tree.parent.push_back(-1);
    tree.sw.push_back(-1); tree.delay.push_back(0.0f);
    update_one_cost(src, +1);
    pos.emplace(src, 0);
    for (int gi = 0; gi < n_grp; ++gi) {
      const RouteTree& t = sub[gi];
      for (size_t k = 1; k < t.nodes.size(); ++k) {
        int32_t n = t.nodes[k];
        if (pos.count(n)) continue;
        int32_t p = t.nodes[t.parent[k]];
        auto it = pos.find(p);
        // parent precedes child within the cluster tree, so it is merged
        int32_t pi = it->second;
        pos.emplace(n, (int32_t)tree.nodes.size());
        tree.nodes.push_back(n);
        tree.parent.push_back(pi);
        tree.sw.push_back(t.sw[k]);
        tree.delay.push_back(0.0f);  // recomputed below
        update_one_cost(n, +1);
      }
    }
    // 3) delays along the merged topology (a grafted subtree's upstream
    //    path may differ from its cluster-of-origin's)
    for (size_t k = 1; k < tree.nodes.size(); ++k) {
      int32_t n = tree.nodes[k];
      int8_t sw = tree.sw[k];
      float Tdel = g_->sw_Tdel[sw] + g_->C[n] * (g_->sw_R[sw] + 0.5f * g_->R[n]);
      tree.delay[k] = tree.delay[tree.parent[k]] + Tdel;
    }
    for (int64_t c = s0; c < sink_ptr_[inet + 1]; ++c) {
      auto it = pos.find(sinks_[c]);
      if (it != pos.end()) sink_delays_[c] = tree.delay[it->second];
    }
  }

  // Nets with at least one sink not present in their current tree
  // (never routed, or dropped by a partial rip) — must be in every
  // incremental reroute set or the flow could terminate with a missing
  // connection (check_routed would catch it, but late).
  std::vector<int32_t> incomplete_nets() const {
    std::vector<int32_t> out;
    std::unordered_set<int32_t> in_tree;
    for (int inet = 0; inet < num_nets(); ++inet) {
      in_tree.clear();
      in_tree.insert(trees_[inet].nodes.begin(), trees_[inet].nodes.end());
      for (int64_t k = sink_ptr_[inet]; k < sink_ptr_[inet + 1]; ++k)
        if (!in_tree.count(sinks_[k])) { out.push_back(inet); break; }
    }
    return out;
  }

  void route_net_incremental(int inet, const float* crit_flat,
                             float crit_rip_thr = 2.0f) {
    RouteTree& tree = trees_[inet];
    if (tree.nodes.empty()) { route_net(inet, crit_flat); return; }
    int nt = (int)tree.nodes.size();
    std::vector<char> drop(nt, 0);
    bool any = false;
    for (int k = 0; k < nt; ++k) {
      bool cong = occ_[tree.nodes[k]] > g_->capacity[tree.nodes[k]];
      bool par = tree.parent[k] >= 0 && drop[tree.parent[k]];
      drop[k] = cong || par;  // parents precede children in tree order
      any |= (bool)drop[k];
    }
    if (crit_flat && crit_rip_thr <= 1.0f) {
      // rip the PRIVATE path of each near-critical sink (up to the first
      // node still feeding a surviving branch) so it can re-route for
      // delay; shared trunk stays.
      std::vector<int16_t> surv(nt, 0);
      for (int k = 1; k < nt; ++k)
        if (!drop[k]) surv[tree.parent[k]]++;
      int64_t s0 = sink_ptr_[inet], s1 = sink_ptr_[inet + 1];
      for (int64_t c = s0; c < s1; ++c) {
        if (crit_flat[c] < crit_rip_thr) continue;
        int32_t snode = sinks_[c];
        for (int k = 0; k < nt; ++k) {
          if (tree.nodes[k] != snode || drop[k]) continue;
          int cur = k;
          while (cur > 0 && !drop[cur] && surv[cur] == 0) {
            drop[cur] = 1;
            any = true;
            int p = tree.parent[cur];
            if (p >= 0) surv[p]--;
            cur = p;
          }
          break;
        }
      }
    }
    if (any && drop[0]) { route_net(inet, crit_flat); return; }
    if (any) {
      RouteTree kept;
      std::vector<int32_t> remap(nt, -1);
      for (int k = 0; k < nt; ++k) {
        if (drop[k]) { update_one_cost(tree.nodes[k], -1); continue; }
        remap[k] = (int32_t)kept.nodes.size();
        kept.nodes.push_back(tree.nodes[k]);
        kept.parent.push_back(tree.parent[k] < 0 ? -1 : remap[tree.parent[k]]);
        kept.sw.push_back(tree.sw[k]);
        kept.delay.push_back(tree.delay[k]);
      }
      tree = std::move(kept);
    }
    // route sinks whose SINK node is no longer in the tree
    std::unordered_set<int32_t> in_tree(tree.nodes.begin(), tree.nodes.end());
    int64_t s0 = sink_ptr_[inet], s1 = sink_ptr_[inet + 1];
    int ns = (int)(s1 - s0);
    sink_order_.resize(ns);
    for (int i = 0; i < ns; ++i) sink_order_[i] = i;
    std::sort(sink_order_.begin(), sink_order_.end(), [&](int a, int b) {
      float ca = crit_flat ? crit_flat[s0 + a] : 0.f;
      float cb = crit_flat ? crit_flat[s0 + b] : 0.f;
      if (ca != cb) return ca > cb;
      return a < b;
    });
    if ((int64_t)sink_delays_.size() < s1) sink_delays_.resize(sinks_.size(), 0.f);
    for (int i = 0; i < ns; ++i) {
      int si = sink_order_[i];
      int32_t sink = sinks_[s0 + si];
      if (in_tree.count(sink)) continue;  // path intact; delay unchanged
      float crit = crit_flat ? crit_flat[s0 + si] : 0.0f;
      sink_delays_[s0 + si] = route_one_sink(inet, sink, crit, trees_[inet]);
      in_tree.insert(sink);
    }
  }

  // Rip up the given nets' trees (occ -1, trees cleared) without
  // rerouting — ownership hand-off during load rebalancing (the
  // reference migrates trees, mpi_route...cxx:172 move_route_tree; with
  // a replicated graph the new owner just reroutes from scratch).
  void rip_up_nets(const int32_t* ids, int64_t n) {
    for (int64_t i = 0; i < n; ++i) {
      RouteTree& t = trees_[ids[i]];
      for (int32_t v : t.nodes) update_one_cost(v, -1);
      t.clear();
    }
  }

  // Replace occupancy wholesale (after a distributed occ all-reduce) and
  // refresh pres_cost from it.
  void set_occ(const int32_t* occ) {
    for (int v = 0; v < g_->num_nodes; ++v) {
      occ_[v] = occ[v];
      int over = occ_[v] + 1 - g_->capacity[v];
      pres_[v] = (over > 0) ? 1.0f + over * pres_fac_ : 1.0f;
    }
  }

  // PathFinder cost-schedule update between iterations
  // (reference: congestion.h:176-193 update_costs).
  void update_costs(float pres_fac, float acc_fac) {
    pres_fac_ = pres_fac;
    int n = g_->num_nodes;
    for (int v = 0; v < n; ++v) {
      int over = occ_[v] - g_->capacity[v];
      if (over > 0) acc_[v] += over * acc_fac;
      pres_[v] = (over >= 0) ? 1.0f + (over + 1) * pres_fac_ : 1.0f;
    }
  }
  void set_pres_fac(float p) { pres_fac_ = p; }

  int64_t count_overused() const {
    int64_t c = 0;
    for (int v = 0; v < g_->num_nodes; ++v)
      if (occ_[v] > g_->capacity[v]) ++c;
    return c;
  }

  bool feasible() const { return count_overused() == 0; }

  // Nets whose current tree crosses an overused node (reference:
  // phase-two congested-net selection; GPU analogue:
  // pnr_flag_congested_nets in csrc/hip/router_kernel.hip). Only
  // meaningful for nets whose trees this router owns.
  std::vector<int32_t> congested_nets() const {
    std::vector<int32_t> out;
    for (int inet = 0; inet < num_nets(); ++inet) {
      for (int32_t v : trees_[inet].nodes)
        if (occ_[v] > g_->capacity[v]) { out.push_back(inet); break; }
    }
    return out;
  }

  // Elmore delay at each routed sink (aligned with sinks_).
  void sink_delays(float* out) const {
    for (size_t i = 0; i < sink_delays_.size(); ++i) out[i] = sink_delays_[i];
  }

  int64_t total_wirelength() const {
    int64_t wl = 0;
    for (auto& t : trees_)
      for (int32_t v : t.nodes)
        if (g_->type[v] == CHANX || g_->type[v] == CHANY)
          wl += g_->xhigh[v] - g_->xlow[v] + g_->yhigh[v] - g_->ylow[v] + 1;
    return wl;
  }

  // Validator (reference: route/check_route.c:27): every net's tree is
  // connected, rooted at its source, visits every sink; occ matches a
  // recount; returns true if valid.
  bool check_routed(std::string* err) const {
    std::vector<int32_t> recount(g_->num_nodes, 0);
    for (int inet = 0; inet < num_nets(); ++inet) {
      const RouteTree& t = trees_[inet];
      if (t.nodes.empty()) { *err = "net has empty tree"; return false; }
      if (t.nodes[0] != net_src_[inet]) { *err = "tree root != source"; return false; }
      std::vector<char> seen(t.nodes.size(), 0);
      for (size_t k = 0; k < t.nodes.size(); ++k) {
        if (t.parent[k] >= (int)k && k > 0) { *err = "parent after child"; return false; }
        recount[t.nodes[k]]++;
        // edge existence
        if (k > 0) {
          int32_t p = t.nodes[t.parent[k]], v = t.nodes[k];
          bool found = false;
          for (int64_t e = g_->row_ptr[p]; e < g_->row_ptr[p + 1]; ++e)
            if (g_->edge_dst[e] == v) { found = true; break; }
          if (!found) { *err = "tree edge not in rr graph"; return false; }
        }
      }
      (void)seen;
      // all sinks present
      for (int64_t s = sink_ptr_[inet]; s < sink_ptr_[inet + 1]; ++s) {
        int32_t snk = sinks_[s];
        bool found = false;
        for (int32_t v : t.nodes) if (v == snk) { found = true; break; }
        if (!found) { *err = "sink not reached"; return false; }
      }
    }
    for (int v = 0; v < g_->num_nodes; ++v)
      if (recount[v] != occ_[v]) { *err = "occ mismatch at node " + std::to_string(v); return false; }
    return true;
  }

  const std::vector<int32_t>& occ() const { return occ_; }
  const RouteTree& tree(int inet) const { return trees_[inet]; }
  int64_t heap_pushes() const { return heap_pushes_; }
  int64_t heap_pops() const { return heap_pops_; }

 private:
  struct HeapEnt {
    float total, back, Rup;
    int32_t node, prev, sw;
    bool operator<(const HeapEnt& o) const { return total > o.total; }  // min-heap
  };

  void update_one_cost(int32_t v, int delta) {
    // reference: congestion.cxx:296 update_one_cost_internal
    occ_[v] += delta;
    int over = occ_[v] + 1 - g_->capacity[v];
    pres_[v] = (over > 0) ? 1.0f + over * pres_fac_ : 1.0f;
  }

  float expected_cost(int32_t v, int32_t sink, float crit) const {
    // distance-based lookahead (reference: router.cxx:610
    // get_timing_driven_expected_cost / :445 get_expected_segs_to_target)
    int8_t ty = g_->type[v];
    if (ty == SINK) return 0.0f;
    int tx = g_->xlow[sink], ty2 = g_->ylow[sink];
    int dx = 0, dy = 0;
    if (g_->xlow[v] > tx) dx = g_->xlow[v] - tx;
    else if (g_->xhigh[v] < tx) dx = tx - g_->xhigh[v];
    if (g_->ylow[v] > ty2) dy = g_->ylow[v] - ty2;
    else if (g_->yhigh[v] < ty2) dy = ty2 - g_->yhigh[v];
    int dist = dx + dy;
    int nseg = (dist + g_->L - 1) / g_->L;
    float cong = nseg * seg_base_ + g_->base_cost[IPIN];
    float del = nseg * seg_delay_ + ipin_delay_;
    return crit * del + (1.0f - crit) * cong;
  }

  void route_net(int inet, const float* crit_flat) {
    RouteTree& tree = trees_[inet];
    // rip-up (reference rips up the whole net every iteration:
    // route_timing.c rip-up in timing_driven_route_net)
    for (int32_t v : tree.nodes) update_one_cost(v, -1);
    tree.clear();
    int32_t src = net_src_[inet];
    tree.nodes.push_back(src); tree.parent.push_back(-1);
    tree.sw.push_back(-1); tree.delay.push_back(0.0f);
    update_one_cost(src, +1);

    // sinks sorted by decreasing criticality
    int64_t s0 = sink_ptr_[inet], s1 = sink_ptr_[inet + 1];
    int ns = (int)(s1 - s0);
    sink_order_.resize(ns);
    for (int i = 0; i < ns; ++i) sink_order_[i] = i;
    std::sort(sink_order_.begin(), sink_order_.end(), [&](int a, int b) {
      float ca = crit_flat ? crit_flat[s0 + a] : 0.f;
      float cb = crit_flat ? crit_flat[s0 + b] : 0.f;
      if (ca != cb) return ca > cb;
      return a < b;
    });
    if ((int64_t)sink_delays_.size() < s1) sink_delays_.resize(sinks_.size(), 0.f);

    for (int i = 0; i < ns; ++i) {
      int si = sink_order_[i];
      int32_t sink = sinks_[s0 + si];
      float crit = crit_flat ? crit_flat[s0 + si] : 0.0f;
      float d = route_one_sink(inet, sink, crit, tree);
      sink_delays_[s0 + si] = d;
    }
  }

  // Returns delay at sink. Grows `tree` and occ along the new path.
  float route_one_sink(int inet, int32_t sink, float crit, RouteTree& tree) {
    (void)inet;
    std::priority_queue<HeapEnt> heap;
    touched_.clear();
    // seed with the existing route tree (congestion-free re-use;
    // reference: SinkRouter sources from tree nodes inside bb,
    // partitioning_multi_sink...cxx:707-745)
    for (size_t k = 0; k < tree.nodes.size(); ++k) {
      int32_t v = tree.nodes[k];
      if (g_->type[v] == SINK) continue;  // cannot expand from a sink
      float back = crit * tree.delay[k];
      float tot = back + opts_.astar_fac * expected_cost(v, sink, crit);
      // R_up at v in the tree: all switches buffered -> R resets per hop.
      float rup = g_->R[v];
      push(heap, v, tot, back, rup, -2 - (int32_t)k, -1);  // prev<-2 encodes tree index
    }
    int32_t found_prev = -1;
    float sink_back = 0.f;
    while (!heap.empty()) {
      HeapEnt e = heap.top(); heap.pop();
      ++heap_pops_;
      if (e.total > path_cost_[e.node]) continue;   // stale
      if (e.node == sink) { found_prev = e.prev; sink_back = e.back; (void)sink_back; break; }
      // expand
      for (int64_t ei = g_->row_ptr[e.node]; ei < g_->row_ptr[e.node + 1]; ++ei) {
        int32_t v = g_->edge_dst[ei];
        int8_t sw = g_->edge_sw[ei];
        if (g_->type[v] == SINK && v != sink) continue;  // don't route through other sinks
        if (g_->type[v] == IPIN) {
          // prune IPINs not at the sink tile (cheap bb pruning)
          if (g_->xlow[v] != g_->xlow[sink] || g_->ylow[v] != g_->ylow[sink])
            continue;
        }
        float Rsw = g_->sw_R[sw];
        float Tsw = g_->sw_Tdel[sw];
        float Rnode = g_->R[v], Cnode = g_->C[v];
        float Tdel = Tsw + Cnode * (Rsw + 0.5f * Rnode);
        float cong = g_->base_cost[g_->type[v]] * acc_[v] * pres_[v];
        float back = e.back + crit * Tdel + (1.0f - crit) * cong;
        float tot = back + opts_.astar_fac * expected_cost(v, sink, crit);
        if (tot < path_cost_[v]) {
          push(heap, v, tot, back, Rsw + Rnode, e.node, (int32_t)sw);
        }
      }
    }
    if (found_prev == -1) {
      // genuinely unreachable (should not happen on a validated fabric)
      ++unrouted_sinks_;
      reset_touched();
      return 0.0f;
    }
    // backtrack: collect path sink..tree-node (exclusive — a node whose own
    // prev is a seed marker IS the tree attach point and must not be
    // re-appended, else the tree duplicates it)
    path_buf_.clear();
    int32_t v = sink;
    int32_t tree_attach = -1;
    while (true) {
      int32_t pv = prev_node_[v];
      if (pv <= -2) { tree_attach = -2 - pv; break; }  // v is in the tree
      path_buf_.push_back(v);
      v = pv;
    }
    // walk from attach point forward, appending to tree
    int parent_idx = tree_attach;
    float delay_acc = tree.delay[tree_attach];
    for (int k = (int)path_buf_.size() - 1; k >= 0; --k) {
      int32_t n = path_buf_[k];
      int8_t sw = (int8_t)prev_sw_[n];
      float Tdel = g_->sw_Tdel[sw] + g_->C[n] * (g_->sw_R[sw] + 0.5f * g_->R[n]);
      delay_acc += Tdel;
      tree.nodes.push_back(n);
      tree.parent.push_back(parent_idx);
      tree.sw.push_back(sw);
      tree.delay.push_back(delay_acc);
      parent_idx = (int)tree.nodes.size() - 1;
      update_one_cost(n, +1);
    }
    float d = delay_acc;
    reset_touched();
    return d;
  }

  void push(std::priority_queue<HeapEnt>& heap, int32_t v, float tot,
            float back, float rup, int32_t prev, int32_t sw) {
    // a node seeded from the route tree is final for this sink's search:
    // re-entering the tree through real wires would duplicate tree nodes
    // on backtrack (and costs more congestion anyway)
    if (prev_node_[v] <= -2) return;
    if (path_cost_[v] >= 1e29f) touched_.push_back(v);
    if (tot < path_cost_[v]) {
      path_cost_[v] = tot;
      back_cost_[v] = back;
      prev_node_[v] = prev;
      prev_sw_[v] = sw;
      R_up_[v] = rup;
      heap.push({tot, back, rup, v, prev, sw});
      ++heap_pushes_;
    }
  }

  void reset_touched() {
    for (int32_t v : touched_) {
      path_cost_[v] = 1e30f; back_cost_[v] = 1e30f;
      prev_node_[v] = -1; prev_sw_[v] = -1;
    }
    touched_.clear();
  }

 public:
  std::shared_ptr<RRGraph> graph_holder_;  // lifetime pin for Python bindings
  const RRGraph* g_;
  std::vector<int32_t> net_src_;
  std::vector<int64_t> sink_ptr_;
  std::vector<int32_t> sinks_;
  RouterOpts opts_;
  std::vector<int32_t> occ_;
  std::vector<float> pres_, acc_;
  float pres_fac_;
  std::vector<float> sink_delays_;
  std::vector<RouteTree> trees_;
  int64_t heap_pushes_ = 0, heap_pops_ = 0;
  int64_t unrouted_sinks_ = 0;

 private:
  std::vector<float> path_cost_, back_cost_, R_up_;
  std::vector<int32_t> prev_node_, prev_sw_;
  std::vector<int32_t> tree_mark_;
  std::vector<int32_t> touched_;
  std::vector<int32_t> path_buf_;
  std::vector<int> sink_order_;
  float seg_delay_, ipin_delay_, seg_base_;
};

}  // namespace pnr
