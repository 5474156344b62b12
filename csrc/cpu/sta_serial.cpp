// Static timing analysis — levelized forward/backward sweeps.
//
// Re-implements the semantics of the reference STA
// (vpr/SRC/timing/path_delay.c:1994 do_timing_analysis_new, levelization
//  path_delay2.c:81): block-granularity timing graph, T_arr max-plus forward
//  sweep per level, T_req min-minus backward, per-connection slack and
//  criticality = 1 - slack/cpd; analyze_domains implements the
//  reference's multi-clock (src,sink)-domain-pair loop. Heterogeneous
//  blocks carry per-block combinational delays (blk_delay_).
//
// The level arrays built here are uploaded to HBM for the GPU STA kernels
// (csrc/hip/sta_kernel.hip), which run the same sweeps level-synchronously.
#include "pnr.h"
#include <algorithm>
#include <map>

namespace pnr {

class TimingGraph {
 public:
  // blk_delay: optional per-block combinational propagation delay
  // (heterogeneous blocks: CLB/RAM/DSP differ); empty => T_clb everywhere.
  TimingGraph(const Netlist* nl, float T_clb, float T_seq_out, float T_seq_in,
              std::vector<float> blk_delay = {})
      : nl_(nl), T_clb_(T_clb), T_seq_out_(T_seq_out), T_seq_in_(T_seq_in),
        blk_delay_(std::move(blk_delay)) {
    if (blk_delay_.empty()) blk_delay_.assign(nl_->num_blocks, T_clb_);
    if (blk_delay_.size() != (size_t)nl_->num_blocks)
      throw std::runtime_error("blk_delay size mismatch");
    levelize();
  }

  // conn_delay: per (net,sink) connection delay, aligned with nl_->net_sinks.
  // Outputs: slack (same alignment), crit = 1 - slack/cpd, returns cpd.
  float analyze(const float* conn_delay, float* slack, float* crit) {
    int nb = nl_->num_blocks;
    t_arr_.assign(nb, 0.0f);
    t_req_.assign(nb, 3.0e38f);
    // forward: blocks in topo order. T_arr[b] = output arrival time.
    for (int b : topo_) {
      bool seq = nl_->block_is_seq[b];
      if (seq) { t_arr_[b] = T_seq_out_; continue; }
      float a = in_arrival(b, conn_delay);
      t_arr_[b] = a + blk_delay_[b];
    }
    // cpd = max arrival at any sequential/output endpoint input
    float cpd = 0.0f;
    for (int b = 0; b < nb; ++b) {
      if (!nl_->block_is_seq[b]) continue;
      float a = in_arrival(b, conn_delay);
      if (a + T_seq_in_ > cpd) cpd = a + T_seq_in_;
    }
    if (cpd <= 0) cpd = 1e-12f;
    // backward: T_req at block OUTPUT. For seq endpoint b, required at its
    // input is cpd - T_seq_in.
    for (auto it = topo_.rbegin(); it != topo_.rend(); ++it) {
      int b = *it;
      float r = 3.0e38f;
      for (int64_t k = out_ptr_[b]; k < out_ptr_[b + 1]; ++k) {
        int64_t conn = out_conn_[k];
        int snk = nl_->net_sinks[conn];
        float req_in = nl_->block_is_seq[snk] ? (cpd - T_seq_in_)
                                              : (t_req_[snk] - blk_delay_[snk]);
        float rr = req_in - conn_delay[conn];
        if (rr < r) r = rr;
      }
      t_req_[b] = r;  // inf for blocks driving nothing
    }
    // per-connection slack + criticality
    int64_t nconn = (int64_t)nl_->net_sinks.size();
    for (int64_t c = 0; c < nconn; ++c) {
      int drv = conn_driver_[c];
      int snk = nl_->net_sinks[c];
      float req_in = nl_->block_is_seq[snk] ? (cpd - T_seq_in_)
                                            : (t_req_[snk] - blk_delay_[snk]);
      float s = req_in - (t_arr_[drv] + conn_delay[c]);
      slack[c] = s;
      float cr = 1.0f - s / cpd;
      crit[c] = cr < 0 ? 0.0f : (cr > 1 ? 1.0f : cr);
    }
    return cpd;
  }

  int num_levels() const { return num_levels_; }
  const std::vector<int32_t>& topo() const { return topo_; }
  const std::vector<int32_t>& level_of() const { return level_; }

  // Multi-clock analysis (reference: do_timing_analysis_new's per
  // (src_domain, sink_domain) pair loop, path_delay.c:1996-2085).
  // block_clock: domain id per block (-1 for combinational); periods[K].
  // Single-cycle setup constraint against the DESTINATION domain's period.
  // Outputs: per-conn worst slack and criticality (max over pairs, each
  // normalized by its pair's constraint). Returns the worst cpd/period
  // ratio times its period (the binding domain's achieved period).
  // pair_skip / pair_mult: optional KxK (src,sink)-domain-pair
  // constraints (reference read_sdc.c: set_false_path and
  // set_multicycle_path between clock domains). A skipped pair is not
  // analyzed; a multiplied pair's setup constraint is periods[cj]*mult
  // (the backward pass is re-run per distinct (cj, mult) so comb
  // fan-in cones see the adjusted requirement exactly).
  float analyze_domains(const float* conn_delay, const int32_t* block_clock,
                        const float* periods, int K,
                        float* slack, float* crit,
                        const uint8_t* pair_skip = nullptr,
                        const float* pair_mult = nullptr) {
    int nb = nl_->num_blocks;
    int64_t nconn = (int64_t)nl_->net_sinks.size();
    const float NEG = -3.0e38f, POS = 3.0e38f;
    std::vector<std::vector<float>> arr(K), req(K);
    // forward per source domain
    for (int ci = 0; ci < K; ++ci) {
      auto& a = arr[ci];
      a.assign(nb, NEG);
      for (int b : topo_) {
        if (nl_->block_is_seq[b]) {
          a[b] = (block_clock[b] == ci) ? T_seq_out_ : NEG;
          continue;
        }
        float m = NEG;
        for (int64_t k = in_ptr_[b]; k < in_ptr_[b + 1]; ++k) {
          int64_t c = in_conn_[k];
          float v = a[conn_driver_[c]];
          if (v > NEG) v += conn_delay[c];
          if (v > m) m = v;
        }
        a[b] = (m > NEG) ? m + blk_delay_[b] : NEG;
      }
    }
    // backward per (sink domain, constraint period)
    auto backward = [&](int cj, float period, std::vector<float>& r) {
      r.assign(nb, POS);
      float req_ep = period - T_seq_in_;
      for (auto it = topo_.rbegin(); it != topo_.rend(); ++it) {
        int b = *it;
        float m = POS;
        for (int64_t k = out_ptr_[b]; k < out_ptr_[b + 1]; ++k) {
          int64_t c = out_conn_[k];
          int snk = nl_->net_sinks[c];
          float ri;
          if (nl_->block_is_seq[snk])
            ri = (block_clock[snk] == cj) ? req_ep : POS;
          else
            ri = (r[snk] < POS) ? r[snk] - blk_delay_[snk] : POS;
          if (ri < POS) ri -= conn_delay[c];
          if (ri < m) m = ri;
        }
        r[b] = m;
      }
    };
    for (int cj = 0; cj < K; ++cj) backward(cj, periods[cj], req[cj]);
    // multicycle pairs need the backward pass at the adjusted period
    std::map<std::pair<int, float>, std::vector<float>> req_mult;
    if (pair_mult)
      for (int ci = 0; ci < K; ++ci)
        for (int cj = 0; cj < K; ++cj) {
          float mu = pair_mult[ci * K + cj];
          if (mu != 1.0f && !req_mult.count({cj, mu}))
            backward(cj, periods[cj] * mu, req_mult[{cj, mu}]);
        }
    for (int64_t c = 0; c < nconn; ++c) { slack[c] = POS; crit[c] = 0.0f; }
    float worst_ratio = 0.0f;  // achieved/required; >1 means violated
    float worst_period = periods[0];
    for (int ci = 0; ci < K; ++ci)
      for (int cj = 0; cj < K; ++cj) {
        if (pair_skip && pair_skip[ci * K + cj]) continue;  // false path
        float mu = pair_mult ? pair_mult[ci * K + cj] : 1.0f;
        float constraint = periods[cj] * mu;
        const std::vector<float>& R =
            (mu == 1.0f) ? req[cj] : req_mult[{cj, mu}];
        for (int64_t c = 0; c < nconn; ++c) {
          int drv = conn_driver_[c];
          int snk = nl_->net_sinks[c];
          if (arr[ci][drv] <= NEG) continue;
          float ri;
          if (nl_->block_is_seq[snk])
            ri = (block_clock[snk] == cj) ? constraint - T_seq_in_ : POS;
          else
            ri = (R[snk] < POS) ? R[snk] - blk_delay_[snk] : POS;
          if (ri >= POS) continue;
          float s = ri - (arr[ci][drv] + conn_delay[c]);
          if (s < slack[c]) slack[c] = s;
          float cr = 1.0f - s / constraint;
          if (cr > crit[c]) crit[c] = cr < 0 ? 0.0f : (cr > 1 ? 1.0f : cr);
          float achieved = constraint - s;
          if (achieved / constraint > worst_ratio) {
            worst_ratio = achieved / constraint;
            worst_period = achieved;
          }
        }
      }
    for (int64_t c = 0; c < nconn; ++c)
      if (slack[c] >= POS) slack[c] = 0.0f;
    return worst_period;
  }

  // GPU upload accessors: blocks sorted by level + CSRs
  void level_arrays(std::vector<int32_t>& blocks,
                    std::vector<int32_t>& start) const {
    int nb = nl_->num_blocks;
    start.assign(num_levels_ + 1, 0);
    for (int b = 0; b < nb; ++b) start[level_[b] + 1]++;
    for (int l = 0; l < num_levels_; ++l) start[l + 1] += start[l];
    blocks.assign(nb, -1);
    std::vector<int32_t> cur(start.begin(), start.end() - 1);
    for (int b = 0; b < nb; ++b) blocks[cur[level_[b]]++] = b;
  }
  const std::vector<int64_t>& in_ptr() const { return in_ptr_; }
  const std::vector<int64_t>& in_conn() const { return in_conn_; }
  const std::vector<int64_t>& out_ptr() const { return out_ptr_; }
  const std::vector<int64_t>& out_conn() const { return out_conn_; }
  const std::vector<int32_t>& conn_driver() const { return conn_driver_; }

 public:
  std::shared_ptr<Netlist> netlist_holder_;  // lifetime pin for Python bindings
  const Netlist* nl_;
  float T_clb_, T_seq_out_, T_seq_in_;
  std::vector<float> blk_delay_;

 private:
  std::vector<int32_t> topo_, level_;
  int num_levels_ = 0;
  std::vector<float> t_arr_, t_req_;
  // per-block incoming connections (conn index into net_sinks) and outgoing
  std::vector<int64_t> in_ptr_, out_ptr_;
  std::vector<int64_t> in_conn_, out_conn_;
  std::vector<int32_t> conn_driver_;

  float in_arrival(int b, const float* conn_delay) const {
    float a = 0.0f;
    for (int64_t k = in_ptr_[b]; k < in_ptr_[b + 1]; ++k) {
      int64_t c = in_conn_[k];
      float v = t_arr_[conn_driver_[c]] + conn_delay[c];
      if (v > a) a = v;
    }
    return a;
  }

  void levelize() {
    int nb = nl_->num_blocks, nn = nl_->num_nets;
    int64_t nconn = (int64_t)nl_->net_sinks.size();
    conn_driver_.assign(nconn, -1);
    std::vector<int64_t> in_cnt(nb, 0), out_cnt(nb, 0);
    for (int n = 0; n < nn; ++n) {
      int drv = nl_->net_driver[n];
      for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s) {
        conn_driver_[s] = drv;
        out_cnt[drv]++;
        in_cnt[nl_->net_sinks[s]]++;
      }
    }
    in_ptr_.assign(nb + 1, 0); out_ptr_.assign(nb + 1, 0);
    for (int b = 0; b < nb; ++b) {
      in_ptr_[b + 1] = in_ptr_[b] + in_cnt[b];
      out_ptr_[b + 1] = out_ptr_[b] + out_cnt[b];
    }
    in_conn_.assign(in_ptr_[nb], 0); out_conn_.assign(out_ptr_[nb], 0);
    std::vector<int64_t> ic(in_ptr_.begin(), in_ptr_.end() - 1);
    std::vector<int64_t> oc(out_ptr_.begin(), out_ptr_.end() - 1);
    for (int n = 0; n < nn; ++n) {
      for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s) {
        out_conn_[oc[nl_->net_driver[n]]++] = s;
        in_conn_[ic[nl_->net_sinks[s]]++] = s;
      }
    }
    // topo order over combinational edges (seq blocks are sources: their
    // arrival doesn't depend on inputs)
    std::vector<int32_t> pend(nb, 0);
    std::vector<int32_t> stack;
    level_.assign(nb, 0);
    for (int b = 0; b < nb; ++b) {
      if (nl_->block_is_seq[b]) { stack.push_back(b); continue; }
      int cnt = 0;
      for (int64_t k = in_ptr_[b]; k < in_ptr_[b + 1]; ++k) cnt++;
      pend[b] = cnt;
      if (cnt == 0) stack.push_back(b);
    }
    topo_.clear(); topo_.reserve(nb);
    size_t head = 0;
    std::vector<int32_t> q(std::move(stack));
    while (head < q.size()) {
      int b = q[head++];
      topo_.push_back(b);
      for (int64_t k = out_ptr_[b]; k < out_ptr_[b + 1]; ++k) {
        int snk = nl_->net_sinks[out_conn_[k]];
        if (nl_->block_is_seq[snk]) continue;  // edges into seq don't gate
        int lv = level_[b] + 1;
        if (lv > level_[snk]) level_[snk] = lv;
        if (--pend[snk] == 0) q.push_back(snk);
      }
    }
    if ((int)topo_.size() != nb)
      throw std::runtime_error("combinational cycle in netlist (" +
                               std::to_string(topo_.size()) + "/" +
                               std::to_string(nb) + " levelized)");
    num_levels_ = 0;
    for (int b = 0; b < nb; ++b) num_levels_ = std::max(num_levels_, level_[b] + 1);
  }
};

}  // namespace pnr
