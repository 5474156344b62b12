// Serial simulated-annealing placer — the CPU oracle.
//
// Re-implements the semantics of the reference placer
// (vpr/SRC/place/place.c:310 try_place; try_swap:1252; update_bb:2292;
//  get_net_cost:2204 = HPWL x crossing-count factor; adaptive schedule
//  update_t:983): range-limited swaps, incremental bounding-box cost with
//  boundary counts, Metropolis acceptance, success-rate-driven cooling.
// The outer temperature loop lives in Python (place/placer.py) so the GPU
// engine and this oracle share schedule + STA refresh logic; this class
// runs one temperature's worth of moves.
#include "pnr.h"
#include <random>
#include <algorithm>

namespace pnr {

// crossing-count factor q(fanout) — standard Cheng (1994) wirelength
// correction used by VPR (place.c:197 cross_count); piecewise-linear fit.
static inline float cross_count(int n) {
  static const float q3 = 1.0f, q50 = 2.79f;
  if (n <= 3) return q3;
  if (n >= 50) return q50 + 0.02616f * (n - 50);
  return q3 + (q50 - q3) * (n - 3) / 47.0f;
}

struct Bb { int16_t xmin, xmax, ymin, ymax; int16_t nxmin, nxmax, nymin, nymax; };

class SerialPlacer {
 public:
  // delay_mat: (nx+2)*(ny+2) delay lookup by |dx|,|dy| (flattened dy-major:
  // delay_mat[dx*(ny+2)+dy]); may be empty for bb-only placement.
  // tile_btype: (nx+2)*(ny+2) x-major block type per tile (-1 corner,
  // 0 IO, 1 CLB, 2 RAM, 3 DSP); empty => homogeneous (perimeter IO,
  // interior CLB), which reproduces the pre-heterogeneous behavior.
  SerialPlacer(const Netlist* nl, int nx, int ny, int io_cap,
               std::vector<float> delay_mat, uint64_t seed,
               std::vector<int8_t> tile_btype = {})
      : nl_(nl), nx_(nx), ny_(ny), io_cap_(io_cap),
        delay_mat_(std::move(delay_mat)), rng_(seed),
        tile_btype_(std::move(tile_btype)) {
    int nb = nl_->num_blocks;
    bx_.assign(nb, -1); by_.assign(nb, -1); bslot_.assign(nb, 0);
    // grid occupancy: per location, list of block ids (size cap)
    gx_ = nx + 2; gy_ = ny + 2;
    if (tile_btype_.empty()) {
      tile_btype_.assign((size_t)gx_ * gy_, -1);
      for (int x = 1; x <= nx_; ++x)
        for (int y = 1; y <= ny_; ++y) tile_btype_[(size_t)x * gy_ + y] = 1;
      for (int y = 1; y <= ny_; ++y) {
        tile_btype_[y] = 0; tile_btype_[(size_t)(gx_ - 1) * gy_ + y] = 0;
      }
      for (int x = 1; x <= nx_; ++x) {
        tile_btype_[(size_t)x * gy_] = 0;
        tile_btype_[(size_t)x * gy_ + gy_ - 1] = 0;
      }
    }
    if (tile_btype_.size() != (size_t)gx_ * gy_)
      throw std::runtime_error("tile_btype size mismatch");
    // per-type column lists for sparse-column (RAM/DSP) move proposals
    for (int x = 1; x <= nx_; ++x) {
      int8_t t = tile_btype_[(size_t)x * gy_ + 1];
      if (t >= 2 && t <= 3) type_cols_[t - 2].push_back(x);
    }
    grid_.assign((size_t)gx_ * gy_ * std::max(1, io_cap), -1);
    grid_cnt_.assign((size_t)gx_ * gy_, 0);
    build_net_arrays();
    initial_placement();
    recompute_bb_all();
    crit_.assign(nl_->net_sinks.size(), 0.0f);
    conn_delay_.assign(nl_->net_sinks.size(), 0.0f);
    recompute_td_all();
  }

  int8_t tile_type(int x, int y) const {
    return tile_btype_[(size_t)x * gy_ + y];
  }
  int cap_at(int x, int y) const {
    int8_t t = tile_type(x, y);
    if (t < 0) return 0;
    return t == 0 ? io_cap_ : 1;
  }
  bool is_io_loc(int x, int y) const {
    return (x == 0 || x == gx_ - 1 || y == 0 || y == gy_ - 1);
  }

  // ---- costs ----
  double bb_cost() const { return bb_cost_; }
  double td_cost() const { return td_cost_; }

  double recompute_bb_cost_from_scratch() {
    // drift check (reference: place.c recompute_bb_cost / check_place:2950)
    double c = 0;
    for (int n = 0; n < nl_->num_nets; ++n) c += net_cost_from_scratch(n);
    return c;
  }

  // One temperature step: nmoves moves at temperature T, range limit rlim.
  // timing_tradeoff: 0 => pure bb. crit must be set via set_crit.
  // Returns success rate.
  double run_moves(double T, double rlim, int64_t nmoves, double timing_tradeoff,
                   double bb_norm, double td_norm) {
    int64_t acc = 0;
    delta_sum_ = delta_sq_sum_ = 0; delta_n_ = 0;
    att_valid_ = acc_cnt_ = 0;
    for (int64_t m = 0; m < nmoves; ++m)
      acc += try_swap(T, rlim, timing_tradeoff, bb_norm, td_norm);
    acc_cnt_ = acc;
    return (double)acc / std::max<int64_t>(1, nmoves);
  }

  // std of move deltas over the last run_moves call — used for the
  // reference-style starting temperature (place.c:1045 starting_t).
  double last_delta_std() const {
    if (delta_n_ < 2) return 0.0;
    double mean = delta_sum_ / delta_n_;
    double var = delta_sq_sum_ / delta_n_ - mean * mean;
    return var > 0 ? std::sqrt(var) : 0.0;
  }

  void set_crit(const float* crit, int64_t n) {
    for (int64_t i = 0; i < n; ++i) crit_[i] = crit[i];
    recompute_td_all();
  }

  // connection delays by current placement (for STA)
  void get_conn_delays(float* out) const {
    for (size_t i = 0; i < conn_delay_.size(); ++i) out[i] = conn_delay_[i];
  }

  void get_placement(int32_t* x, int32_t* y, int32_t* slot) const {
    for (int b = 0; b < nl_->num_blocks; ++b) { x[b] = bx_[b]; y[b] = by_[b]; slot[b] = bslot_[b]; }
  }
  void set_placement(const int32_t* x, const int32_t* y, const int32_t* slot) {
    std::fill(grid_.begin(), grid_.end(), -1);
    std::fill(grid_cnt_.begin(), grid_cnt_.end(), 0);
    for (int b = 0; b < nl_->num_blocks; ++b) {
      bx_[b] = x[b]; by_[b] = y[b]; bslot_[b] = slot[b];
      grid_at(x[b], y[b], slot[b]) = b;
      grid_cnt_[(size_t)x[b] * gy_ + y[b]]++;
    }
    recompute_bb_all();
    recompute_td_all();
  }

  bool check_place(std::string* err) const {
    // every block in a legal location; grid consistent; cost drift small
    std::vector<int> cnt((size_t)gx_ * gy_, 0);
    for (int b = 0; b < nl_->num_blocks; ++b) {
      int x = bx_[b], y = by_[b];
      if (cap_at(x, y) <= bslot_[b]) { *err = "block in illegal slot"; return false; }
      if (tile_type(x, y) != nl_->block_type[b]) { *err = "type/location mismatch"; return false; }
      if (grid_at(x, y, bslot_[b]) != b) { *err = "grid inconsistent"; return false; }
      cnt[(size_t)x * gy_ + y]++;
    }
    if (!macro_ptr_.empty() && !macro_members_consistent(err)) return false;
    double fresh = const_cast<SerialPlacer*>(this)->net_cost_sum_check();
    if (std::abs(fresh - bb_cost_) > 1e-3 * std::max(1.0, fresh)) {
      *err = "bb cost drift: " + std::to_string(fresh) + " vs " + std::to_string(bb_cost_);
      return false;
    }
    return true;
  }

  uint64_t rand_u64() { return rng_(); }

  // Restrict moves to a column range [x0, x1] (inclusive), the grid-shard
  // domain decomposition of SURVEY section 7 step 6: a rank only moves
  // blocks currently inside its strip, and only within the strip, so
  // concurrent ranks touch disjoint grid cells and their placements
  // merge conflict-free. x0 = -1 disables.
  void set_move_region(int x0, int x1) { rx0_ = x0; rx1_ = x1; }
  // Per-rank stream divergence after a shared-seed initial placement.
  void reseed(uint64_t seed) { rng_.seed(seed); }
  int64_t last_valid_attempts() const { return att_valid_; }
  int64_t last_accepts() const { return acc_cnt_; }

  // Placement macros (reference: place/place_macro.c — carry chains):
  // groups of blocks at fixed relative offsets from a head, moved
  // atomically by try_swap. member_ptr/member_blk/dx/dy: CSR, first
  // member of each macro is the head at offset (0,0). Re-runs the
  // initial placement so macros start legal.
  void set_macros(const int64_t* member_ptr, const int32_t* member_blk,
                  const int32_t* mdx, const int32_t* mdy, int n_macros) {
    int nb = nl_->num_blocks;
    macro_ptr_.assign(member_ptr, member_ptr + n_macros + 1);
    int64_t total = member_ptr[n_macros];
    macro_blk_.assign(member_blk, member_blk + total);
    macro_dx_.assign(mdx, mdx + total);
    macro_dy_.assign(mdy, mdy + total);
    macro_of_.assign(nb, -1);
    for (int m = 0; m < n_macros; ++m) {
      if (macro_dx_[macro_ptr_[m]] != 0 || macro_dy_[macro_ptr_[m]] != 0)
        throw std::runtime_error("macro head must be at offset (0,0)");
      for (int64_t k = macro_ptr_[m]; k < macro_ptr_[m + 1]; ++k) {
        int b = macro_blk_[k];
        if (nl_->block_type[b] == 0)
          throw std::runtime_error("IO blocks cannot join macros");
        if (macro_of_[b] >= 0)
          throw std::runtime_error("block in two macros");
        macro_of_[b] = m;
      }
    }
    std::fill(grid_.begin(), grid_.end(), -1);
    std::fill(grid_cnt_.begin(), grid_cnt_.end(), 0);
    initial_placement_with_macros();
    recompute_bb_all();
    recompute_td_all();
  }

  bool macro_members_consistent(std::string* err) const {
    for (size_t m = 0; m + 1 < macro_ptr_.size(); ++m) {
      int head = macro_blk_[macro_ptr_[m]];
      for (int64_t k = macro_ptr_[m]; k < macro_ptr_[m + 1]; ++k) {
        int b = macro_blk_[k];
        if (bx_[b] != bx_[head] + macro_dx_[k] ||
            by_[b] != by_[head] + macro_dy_[k]) {
          *err = "macro " + std::to_string(m) + " member offset broken";
          return false;
        }
      }
    }
    return true;
  }

  // Pin blocks to fixed locations (reference: -pad_loc_file / fix_pins,
  // place.c initial_placement_location with pad constraints): teleport
  // each block to its location, then exclude it from all moves.
  void fix_blocks(const int32_t* ids, const int32_t* fx, const int32_t* fy,
                  const int32_t* fslot, int64_t n) {
    if (fixed_.empty()) fixed_.assign(nl_->num_blocks, 0);
    for (int64_t i = 0; i < n; ++i) {
      int b = ids[i];
      int x = fx[i], y = fy[i], sl = fslot[i];
      if (tile_type(x, y) != nl_->block_type[b])
        throw std::runtime_error("fix_blocks: type/location mismatch");
      if (sl < 0 || sl >= cap_at(x, y))
        throw std::runtime_error("fix_blocks: bad slot");
      int32_t occ = grid_at(x, y, sl);
      if (occ == b) { fixed_[b] = 1; continue; }
      // displaced occupant swaps into b's old slot
      grid_at(bx_[b], by_[b], bslot_[b]) = occ;
      if (occ >= 0) {
        if (fixed_[occ]) throw std::runtime_error("fix_blocks: collision");
        bx_[occ] = bx_[b]; by_[occ] = by_[b]; bslot_[occ] = bslot_[b];
      }
      bx_[b] = x; by_[b] = y; bslot_[b] = sl;
      grid_at(x, y, sl) = b;
      fixed_[b] = 1;
    }
    recompute_bb_all();
    recompute_td_all();
  }

 public:
  std::shared_ptr<Netlist> netlist_holder_;  // lifetime pin for Python bindings
  const Netlist* nl_;
  int nx_, ny_, io_cap_, gx_, gy_;
  std::vector<float> delay_mat_;
  std::mt19937_64 rng_;
  std::vector<int32_t> bx_, by_, bslot_;
  std::vector<int8_t> tile_btype_;       // (gx*gy) x-major; see ctor
  std::vector<int> type_cols_[2];        // columns of type RAM(0) / DSP(1)
  std::vector<uint8_t> fixed_;           // empty => nothing fixed
  int rx0_ = -1, rx1_ = -1;              // move region (column strip)
  int64_t att_valid_ = 0, acc_cnt_ = 0;
  std::vector<int64_t> macro_ptr_;       // empty => no macros
  std::vector<int32_t> macro_blk_, macro_dx_, macro_dy_;
  std::vector<int32_t> macro_of_;

 private:
  std::vector<int32_t> grid_;      // (x*gy+y)*cap + slot -> block
  std::vector<int32_t> grid_cnt_;
  std::vector<Bb> bbs_;
  std::vector<float> net_cost_;
  double bb_cost_ = 0, td_cost_ = 0;
  std::vector<float> crit_, conn_delay_;
  // per-block nets (CSR): nets touching each block
  std::vector<int64_t> blk_net_ptr_;
  std::vector<int32_t> blk_nets_;
  std::vector<int64_t> net_nblocks_;  // fanout+1 per net

  int32_t& grid_at(int x, int y, int slot) {
    return grid_[((size_t)x * gy_ + y) * std::max(1, io_cap_) + slot];
  }
  int32_t grid_at(int x, int y, int slot) const {
    return grid_[((size_t)x * gy_ + y) * std::max(1, io_cap_) + slot];
  }

  void build_net_arrays() {
    int nb = nl_->num_blocks, nn = nl_->num_nets;
    std::vector<int64_t> cnt(nb, 0);
    auto each_pin = [&](auto&& f) {
      for (int n = 0; n < nn; ++n) {
        f(n, nl_->net_driver[n]);
        for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s)
          f(n, nl_->net_sinks[s]);
      }
    };
    each_pin([&](int n, int b) { (void)n; cnt[b]++; });
    blk_net_ptr_.assign(nb + 1, 0);
    for (int b = 0; b < nb; ++b) blk_net_ptr_[b + 1] = blk_net_ptr_[b] + cnt[b];
    blk_nets_.assign(blk_net_ptr_[nb], -1);
    std::vector<int64_t> cur(blk_net_ptr_.begin(), blk_net_ptr_.end() - 1);
    each_pin([&](int n, int b) { blk_nets_[cur[b]++] = n; });
    net_nblocks_.assign(nn, 0);
    for (int n = 0; n < nn; ++n)
      net_nblocks_[n] = 1 + (nl_->net_sink_ptr[n + 1] - nl_->net_sink_ptr[n]);
    bbs_.resize(nn);
    net_cost_.assign(nn, 0.f);
  }

  void initial_placement() {
    // random legal placement: each block into a free tile of its type.
    // Location-list build + shuffle order (CLB, RAM, DSP, then IO) keeps
    // the rng stream identical to the pre-heterogeneous placer on
    // homogeneous fabrics (empty lists draw nothing from the rng).
    std::vector<std::pair<int, int>> locs[4];  // by block type
    for (int x = 1; x <= nx_; ++x)
      for (int y = 1; y <= ny_; ++y) {
        int8_t t = tile_type(x, y);
        if (t >= 1) locs[t].push_back({x, y});
      }
    for (int y = 1; y <= ny_; ++y) { locs[0].push_back({0, y}); locs[0].push_back({gx_ - 1, y}); }
    for (int x = 1; x <= nx_; ++x) { locs[0].push_back({x, 0}); locs[0].push_back({x, gy_ - 1}); }
    std::shuffle(locs[1].begin(), locs[1].end(), rng_);
    std::shuffle(locs[2].begin(), locs[2].end(), rng_);
    std::shuffle(locs[3].begin(), locs[3].end(), rng_);
    std::shuffle(locs[0].begin(), locs[0].end(), rng_);
    size_t cur[4] = {0, 0, 0, 0};
    int io_slot = 0;
    for (int b = 0; b < nl_->num_blocks; ++b) {
      int t = nl_->block_type[b];
      if (t < 0 || t > 3) throw std::runtime_error("bad block type");
      if (t != 0) {
        if (cur[t] >= locs[t].size())
          throw std::runtime_error("too many blocks of type " +
                                   std::to_string(t) + " for grid");
        auto [x, y] = locs[t][cur[t]++];
        bx_[b] = x; by_[b] = y; bslot_[b] = 0;
        grid_at(x, y, 0) = b; grid_cnt_[(size_t)x * gy_ + y]++;
      } else {
        if (cur[0] >= locs[0].size()) throw std::runtime_error("too many IOs for grid");
        auto [x, y] = locs[0][cur[0]];
        bx_[b] = x; by_[b] = y; bslot_[b] = io_slot;
        grid_at(x, y, io_slot) = b; grid_cnt_[(size_t)x * gy_ + y]++;
        if (++io_slot >= io_cap_) { io_slot = 0; ++cur[0]; }
      }
    }
  }

  void initial_placement_with_macros() {
    int nb = nl_->num_blocks;
    std::fill(bx_.begin(), bx_.end(), -1);
    std::fill(by_.begin(), by_.end(), -1);
    std::fill(bslot_.begin(), bslot_.end(), 0);
    // 1) macros: scan shuffled head locations where every member offset
    //    lands on a free tile of the member's type
    std::vector<std::pair<int, int>> cells;
    for (int x = 1; x <= nx_; ++x)
      for (int y = 1; y <= ny_; ++y) cells.push_back({x, y});
    std::shuffle(cells.begin(), cells.end(), rng_);
    int n_macros = (int)macro_ptr_.size() - 1;
    for (int m = 0; m < n_macros; ++m) {
      bool placed = false;
      for (auto [hx, hy] : cells) {
        bool ok = true;
        for (int64_t k = macro_ptr_[m]; k < macro_ptr_[m + 1] && ok; ++k) {
          int x = hx + macro_dx_[k], y = hy + macro_dy_[k];
          if (x < 1 || x > nx_ || y < 1 || y > ny_ ||
              tile_type(x, y) != nl_->block_type[macro_blk_[k]] ||
              grid_at(x, y, 0) >= 0)
            ok = false;
        }
        if (!ok) continue;
        for (int64_t k = macro_ptr_[m]; k < macro_ptr_[m + 1]; ++k) {
          int b = macro_blk_[k];
          bx_[b] = hx + macro_dx_[k]; by_[b] = hy + macro_dy_[k];
          bslot_[b] = 0;
          grid_at(bx_[b], by_[b], 0) = b;
          grid_cnt_[(size_t)bx_[b] * gy_ + by_[b]]++;
        }
        placed = true;
        break;
      }
      if (!placed)
        throw std::runtime_error("no legal location for macro " +
                                 std::to_string(m));
    }
    // 2) the rest: free tiles of the right type / IO slots
    std::vector<std::pair<int, int>> locs[4];
    for (int x = 1; x <= nx_; ++x)
      for (int y = 1; y <= ny_; ++y) {
        int8_t t = tile_type(x, y);
        if (t >= 1 && grid_at(x, y, 0) < 0) locs[t].push_back({x, y});
      }
    for (int y = 1; y <= ny_; ++y) { locs[0].push_back({0, y}); locs[0].push_back({gx_ - 1, y}); }
    for (int x = 1; x <= nx_; ++x) { locs[0].push_back({x, 0}); locs[0].push_back({x, gy_ - 1}); }
    for (int t = 0; t < 4; ++t) std::shuffle(locs[t].begin(), locs[t].end(), rng_);
    size_t cur[4] = {0, 0, 0, 0};
    int io_slot = 0;
    for (int b = 0; b < nb; ++b) {
      if (!macro_of_.empty() && macro_of_[b] >= 0) continue;
      int t = nl_->block_type[b];
      if (t != 0) {
        if (cur[t] >= locs[t].size())
          throw std::runtime_error("too many blocks of type " +
                                   std::to_string(t) + " for grid");
        auto [x, y] = locs[t][cur[t]++];
        bx_[b] = x; by_[b] = y; bslot_[b] = 0;
        grid_at(x, y, 0) = b; grid_cnt_[(size_t)x * gy_ + y]++;
      } else {
        if (cur[0] >= locs[0].size()) throw std::runtime_error("too many IOs for grid");
        auto [x, y] = locs[0][cur[0]];
        bx_[b] = x; by_[b] = y; bslot_[b] = io_slot;
        grid_at(x, y, io_slot) = b; grid_cnt_[(size_t)x * gy_ + y]++;
        if (++io_slot >= io_cap_) { io_slot = 0; ++cur[0]; }
      }
    }
  }

  // Atomic macro move (reference: find_affected_blocks place.c:1192):
  // shift every member by (dx,dy); member targets must be type-legal and
  // hold either nothing or single (non-macro) blocks, which swap into
  // the vacated member cells.
  int try_macro_move(int m, double T, double rlim, double timing_tradeoff,
                     double bb_norm, double td_norm) {
    int irlim = std::max(1, (int)rlim);
    int dx = (int)(rng_() % (2 * irlim + 1)) - irlim;
    int dy = (int)(rng_() % (2 * irlim + 1)) - irlim;
    if (dx == 0 && dy == 0) return 0;
    int64_t k0 = macro_ptr_[m], k1 = macro_ptr_[m + 1];
    // strip-sharded mode: the WHOLE macro must live inside this rank's
    // region — a boundary-straddling macro would mutate another rank's
    // cells and break the conflict-free fusion (parallel/dist_place.py)
    if (rx0_ >= 0)
      for (int64_t k = k0; k < k1; ++k) {
        int xb = bx_[macro_blk_[k]];
        if (xb < rx0_ || xb > rx1_) return 0;
      }
    // validate all targets
    mm_moves_.clear();
    for (int64_t k = k0; k < k1; ++k) {
      int b = macro_blk_[k];
      int tx = bx_[b] + dx, ty = by_[b] + dy;
      if (tx < 1 || tx > nx_ || ty < 1 || ty > ny_) return 0;
      if (rx0_ >= 0 && (tx < rx0_ || tx > rx1_)) return 0;
      if (tile_type(tx, ty) != nl_->block_type[b]) return 0;
      int occ = grid_at(tx, ty, 0);
      if (occ >= 0) {
        if (macro_of_[occ] >= 0) return 0;   // another macro's member
        if (!fixed_.empty() && fixed_[occ]) return 0;
        mm_moves_.push_back({occ, bx_[b], by_[b]});  // displaced -> vacated
      }
      mm_moves_.push_back({b, tx, ty});
    }
    ++att_valid_;
    // collect affected nets
    saved_bbs_.clear(); saved_costs_.clear(); saved_ids_.clear();
    for (auto& mv : mm_moves_) {
      int b = mv.b;
      for (int64_t kk = blk_net_ptr_[b]; kk < blk_net_ptr_[b + 1]; ++kk) {
        int n = blk_nets_[kk];
        if (!mark_net(n)) continue;
        saved_ids_.push_back(n);
        saved_bbs_.push_back(bbs_[n]);
        saved_costs_.push_back(net_cost_[n]);
      }
    }
    double before = 0, td_before = 0;
    for (int n : saved_ids_) before += net_cost_[n];
    if (timing_tradeoff > 0) td_before = td_of_move_blocks();
    // apply: clear all source cells, then write all targets
    mm_save_.clear();
    for (auto& mv : mm_moves_) mm_save_.push_back({mv.b, bx_[mv.b], by_[mv.b]});
    for (auto& mv : mm_moves_) grid_at(bx_[mv.b], by_[mv.b], 0) = -1;
    for (auto& mv : mm_moves_) {
      bx_[mv.b] = mv.x; by_[mv.b] = mv.y; bslot_[mv.b] = 0;
      grid_at(mv.x, mv.y, 0) = mv.b;
    }
    for (int n : saved_ids_) net_cost_from_scratch(n);
    double after = 0, td_after = 0;
    for (int n : saved_ids_) after += net_cost_[n];
    if (timing_tradeoff > 0) td_after = td_of_move_blocks();
    double d_bb = after - before, d_td = td_after - td_before;
    double delta = (1.0 - timing_tradeoff) * d_bb / bb_norm +
                   timing_tradeoff * d_td / td_norm;
    delta_sum_ += delta; delta_sq_sum_ += delta * delta; ++delta_n_;
    bool accept;
    if (delta <= 0) accept = true;
    else if (T <= 0) accept = false;
    else {
      double u = (double)(rng_() % (1ull << 53)) / (double)(1ull << 53);
      accept = u < std::exp(-delta / T);
    }
    if (accept) {
      bb_cost_ += d_bb; td_cost_ += d_td;
      for (int n : saved_ids_) unmark_net(n);
      update_conn_delays_of_move();
      return 1;
    }
    // revert
    for (auto& mv : mm_moves_) grid_at(bx_[mv.b], by_[mv.b], 0) = -1;
    for (auto& sv : mm_save_) {
      bx_[sv.b] = sv.x; by_[sv.b] = sv.y;
      grid_at(sv.x, sv.y, 0) = sv.b;
    }
    for (size_t i = 0; i < saved_ids_.size(); ++i) {
      bbs_[saved_ids_[i]] = saved_bbs_[i];
      net_cost_[saved_ids_[i]] = saved_costs_[i];
      unmark_net(saved_ids_[i]);
    }
    return 0;
  }

  double td_of_move_blocks() {
    double t = 0;
    for (auto& mv : mm_moves_) {
      for (int64_t k = blk_net_ptr_[mv.b]; k < blk_net_ptr_[mv.b + 1]; ++k) {
        int n = blk_nets_[k];
        if (!mark_net2(n)) continue;
        int drv = nl_->net_driver[n];
        for (int64_t c2 = nl_->net_sink_ptr[n]; c2 < nl_->net_sink_ptr[n + 1]; ++c2)
          t += (double)crit_[c2] * conn_delay(drv, nl_->net_sinks[c2]);
      }
    }
    for (int n : marked2_) net_mark2_[n] = 0;
    marked2_.clear();
    return t;
  }

  void update_conn_delays_of_move() {
    if (delay_mat_.empty()) return;
    for (auto& mv : mm_moves_) {
      for (int64_t k = blk_net_ptr_[mv.b]; k < blk_net_ptr_[mv.b + 1]; ++k) {
        int n = blk_nets_[k];
        if (!mark_net2(n)) continue;
        int drv = nl_->net_driver[n];
        for (int64_t c2 = nl_->net_sink_ptr[n]; c2 < nl_->net_sink_ptr[n + 1]; ++c2)
          conn_delay_[c2] = conn_delay(drv, nl_->net_sinks[c2]);
      }
    }
    for (int n : marked2_) net_mark2_[n] = 0;
    marked2_.clear();
  }

  struct MacroMove { int b, x, y; };
  std::vector<MacroMove> mm_moves_, mm_save_;

  float net_cost_from_scratch(int n) {
    Bb& b = bbs_[n];
    int16_t xmin = 32767, xmax = 0, ymin = 32767, ymax = 0;
    auto upd = [&](int blk) {
      int x = bx_[blk], y = by_[blk];
      if (x < xmin) xmin = x; if (x > xmax) xmax = x;
      if (y < ymin) ymin = y; if (y > ymax) ymax = y;
    };
    upd(nl_->net_driver[n]);
    for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s)
      upd(nl_->net_sinks[s]);
    b.xmin = xmin; b.xmax = xmax; b.ymin = ymin; b.ymax = ymax;
    // boundary counts
    b.nxmin = b.nxmax = b.nymin = b.nymax = 0;
    auto cntb = [&](int blk) {
      if (bx_[blk] == xmin) b.nxmin++;
      if (bx_[blk] == xmax) b.nxmax++;
      if (by_[blk] == ymin) b.nymin++;
      if (by_[blk] == ymax) b.nymax++;
    };
    cntb(nl_->net_driver[n]);
    for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s)
      cntb(nl_->net_sinks[s]);
    float c = cross_count((int)net_nblocks_[n]) *
              ((b.xmax - b.xmin + 1) + (b.ymax - b.ymin + 1));
    net_cost_[n] = c;
    return c;
  }

  void recompute_bb_all() {
    bb_cost_ = 0;
    for (int n = 0; n < nl_->num_nets; ++n) bb_cost_ += net_cost_from_scratch(n);
  }
  double net_cost_sum_check() {
    double c = 0;
    for (int n = 0; n < nl_->num_nets; ++n) {
      float saved = net_cost_[n];
      Bb sb = bbs_[n];
      c += net_cost_from_scratch(n);
      bbs_[n] = sb; net_cost_[n] = saved;
    }
    return c;
  }

  float conn_delay(int bsrc, int bsnk) const {
    if (delay_mat_.empty()) return 0.f;
    int dx = std::abs(bx_[bsrc] - bx_[bsnk]);
    int dy = std::abs(by_[bsrc] - by_[bsnk]);
    return delay_mat_[(size_t)dx * gy_ + dy];
  }

  void recompute_td_all() {
    td_cost_ = 0;
    for (int n = 0; n < nl_->num_nets; ++n) {
      int drv = nl_->net_driver[n];
      for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s) {
        float d = conn_delay(drv, nl_->net_sinks[s]);
        conn_delay_[s] = d;
        td_cost_ += (double)crit_[s] * d;
      }
    }
  }

  // incremental bb update for moving block blk from (x0,y0) to (x1,y1).
  // Mirrors reference update_bb (place.c:2292): O(1) via boundary counts,
  // full recompute only when a boundary with count 1 moves inward.
  void update_net_for_move(int n, int x0, int y0, int x1, int y1) {
    Bb& b = bbs_[n];
    bool recompute = false;
    // x dimension
    if (x1 < x0) {  // moving left
      if (x0 == b.xmax) { if (b.nxmax == 1) recompute = true; }
    } else if (x1 > x0) {
      if (x0 == b.xmin) { if (b.nxmin == 1) recompute = true; }
    }
    if (y1 < y0) {
      if (y0 == b.ymax) { if (b.nymax == 1) recompute = true; }
    } else if (y1 > y0) {
      if (y0 == b.ymin) { if (b.nymin == 1) recompute = true; }
    }
    if (recompute) { net_cost_from_scratch(n); return; }
    // grow / boundary-count updates
    auto leave_x = [&](int x) {
      if (x == b.xmin) b.nxmin--;
      if (x == b.xmax) b.nxmax--;
    };
    auto leave_y = [&](int y) {
      if (y == b.ymin) b.nymin--;
      if (y == b.ymax) b.nymax--;
    };
    leave_x(x0); leave_y(y0);
    if (x1 < b.xmin) { b.xmin = x1; b.nxmin = 1; }
    else if (x1 == b.xmin) b.nxmin++;
    if (x1 > b.xmax) { b.xmax = x1; b.nxmax = 1; }
    else if (x1 == b.xmax) b.nxmax++;
    if (y1 < b.ymin) { b.ymin = y1; b.nymin = 1; }
    else if (y1 == b.ymin) b.nymin++;
    if (y1 > b.ymax) { b.ymax = y1; b.nymax = 1; }
    else if (y1 == b.ymax) b.nymax++;
    net_cost_[n] = cross_count((int)net_nblocks_[n]) *
                   ((b.xmax - b.xmin + 1) + (b.ymax - b.ymin + 1));
  }

  int try_swap(double T, double rlim, double timing_tradeoff,
               double bb_norm, double td_norm) {
    int nb = nl_->num_blocks;
    int blk = (int)(rng_() % nb);
    if (!fixed_.empty() && fixed_[blk]) return 0;
    int x0 = bx_[blk], y0 = by_[blk];
    if (rx0_ >= 0 && (x0 < rx0_ || x0 > rx1_)) return 0;  // not my shard
    if (!macro_of_.empty() && macro_of_[blk] >= 0)
      return try_macro_move(macro_of_[blk], T, rlim, timing_tradeoff,
                            bb_norm, td_norm);
    int btype = nl_->block_type[blk];
    ++att_valid_;
    // find_to: range-limited destination of matching type (place.c:1520)
    int irlim = std::max(1, (int)rlim);
    int x1 = -1, y1 = -1, slot1 = 0;
    if (btype >= 2) {
      // sparse column types (RAM/DSP): draw the target column from this
      // type's column list clipped to the range window — rejection over
      // the square window would nearly always miss sparse columns.
      const std::vector<int>& cols = type_cols_[btype - 2];
      auto lo = std::lower_bound(cols.begin(), cols.end(), x0 - irlim);
      auto hi = std::upper_bound(cols.begin(), cols.end(), x0 + irlim);
      int ncol = (int)(hi - lo);
      if (ncol > 0) {
        for (int attempt = 0; attempt < 12; ++attempt) {
          int tx = *(lo + (int)(rng_() % ncol));
          if (rx0_ >= 0 && (tx < rx0_ || tx > rx1_)) continue;
          int ylo = std::max(1, y0 - irlim), yhi = std::min(ny_, y0 + irlim);
          int ty = ylo + (int)(rng_() % (yhi - ylo + 1));
          if (tx == x0 && ty == y0) continue;
          x1 = tx; y1 = ty; slot1 = 0;
          break;
        }
      }
    } else {
      bool io = btype == 0;
      for (int attempt = 0; attempt < 12; ++attempt) {
        int dx = (int)(rng_() % (2 * irlim + 1)) - irlim;
        int dy = (int)(rng_() % (2 * irlim + 1)) - irlim;
        int tx = x0 + dx, ty = y0 + dy;
        if (tx < 0 || tx >= gx_ || ty < 0 || ty >= gy_) continue;
        if (rx0_ >= 0 && (tx < rx0_ || tx > rx1_)) continue;
        if (is_io_loc(tx, ty) != io) continue;
        if (tile_type(tx, ty) != btype) continue;
        if (cap_at(tx, ty) <= 0) continue;
        if (tx == x0 && ty == y0) continue;
        x1 = tx; y1 = ty;
        slot1 = (int)(rng_() % cap_at(tx, ty));
        break;
      }
    }
    if (x1 < 0) return 0;
    int other = grid_at(x1, y1, slot1);
    if (other == blk) return 0;
    if (other >= 0 && !fixed_.empty() && fixed_[other]) return 0;
    if (other >= 0 && !macro_of_.empty() && macro_of_[other] >= 0)
      return 0;   // don't break a macro by swapping with its member

    // save + compute delta over affected nets
    saved_bbs_.clear(); saved_costs_.clear(); saved_ids_.clear();
    double d_bb = 0, d_td = 0;
    auto affect = [&](int b, int ox, int oy, int nx2, int ny2) {
      for (int64_t k = blk_net_ptr_[b]; k < blk_net_ptr_[b + 1]; ++k) {
        int n = blk_nets_[k];
        if (!mark_net(n)) continue;
        saved_ids_.push_back(n);
        saved_bbs_.push_back(bbs_[n]);
        saved_costs_.push_back(net_cost_[n]);
        (void)ox; (void)oy; (void)nx2; (void)ny2;
      }
    };
    affect(blk, x0, y0, x1, y1);
    if (other >= 0) affect(other, x1, y1, x0, y0);
    double before = 0;
    for (int n : saved_ids_) before += net_cost_[n];
    double td_before = 0, td_after = 0;
    if (timing_tradeoff > 0) td_before = td_of_blocks(blk, other);

    // tentatively move
    move_block(blk, x1, y1, slot1, other, x0, y0);
    for (size_t i = 0; i < saved_ids_.size(); ++i) {
      int n = saved_ids_[i];
      // use scratch recompute for any net touched by a swap of two blocks;
      // single-block moves use the incremental path.
      if (other >= 0) net_cost_from_scratch(n);
      else {
        // determine which endpoint moved
        update_net_for_move_of(n, blk, x0, y0, x1, y1);
      }
    }
    double after = 0;
    for (int n : saved_ids_) after += net_cost_[n];
    d_bb = after - before;
    if (timing_tradeoff > 0) { td_after = td_of_blocks(blk, other); d_td = td_after - td_before; }

    double delta = (1.0 - timing_tradeoff) * d_bb / bb_norm +
                   timing_tradeoff * d_td / td_norm;
    delta_sum_ += delta; delta_sq_sum_ += delta * delta; ++delta_n_;
    bool accept;
    if (delta <= 0) accept = true;
    else if (T <= 0) accept = false;
    else {
      double u = (double)(rng_() % (1ull << 53)) / (double)(1ull << 53);
      accept = u < std::exp(-delta / T);
    }
    if (accept) {
      bb_cost_ += d_bb; td_cost_ += d_td;
      for (int n : saved_ids_) unmark_net(n);
      update_conn_delays(blk, other);
      return 1;
    }
    // revert
    move_block(blk, x0, y0, bslot_saved_, other, x1, y1, /*revert_other_slot=*/slot1);
    for (size_t i = 0; i < saved_ids_.size(); ++i) {
      bbs_[saved_ids_[i]] = saved_bbs_[i];
      net_cost_[saved_ids_[i]] = saved_costs_[i];
      unmark_net(saved_ids_[i]);
    }
    return 0;
  }

  void update_net_for_move_of(int n, int blk, int x0, int y0, int x1, int y1) {
    (void)blk;
    update_net_for_move(n, x0, y0, x1, y1);
  }

  double td_of_blocks(int b1, int b2) {
    double t = 0;
    auto acc = [&](int b) {
      if (b < 0) return;
      for (int64_t k = blk_net_ptr_[b]; k < blk_net_ptr_[b + 1]; ++k) {
        int n = blk_nets_[k];
        if (!mark_net2(n)) continue;
        int drv = nl_->net_driver[n];
        for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s)
          t += (double)crit_[s] * conn_delay(drv, nl_->net_sinks[s]);
      }
    };
    acc(b1); acc(b2);
    for (int n : marked2_) net_mark2_[n] = 0;
    marked2_.clear();
    return t;
  }

  void update_conn_delays(int b1, int b2) {
    if (delay_mat_.empty()) return;
    auto acc = [&](int b) {
      if (b < 0) return;
      for (int64_t k = blk_net_ptr_[b]; k < blk_net_ptr_[b + 1]; ++k) {
        int n = blk_nets_[k];
        if (!mark_net2(n)) continue;
        int drv = nl_->net_driver[n];
        for (int64_t s = nl_->net_sink_ptr[n]; s < nl_->net_sink_ptr[n + 1]; ++s)
          conn_delay_[s] = conn_delay(drv, nl_->net_sinks[s]);
      }
    };
    acc(b1); acc(b2);
    for (int n : marked2_) net_mark2_[n] = 0;
    marked2_.clear();
  }

  int bslot_saved_ = 0;
  void move_block(int blk, int x1, int y1, int slot1, int other,
                  int x0, int y0, int revert_other_slot = -1) {
    bslot_saved_ = bslot_[blk];
    grid_at(bx_[blk], by_[blk], bslot_[blk]) = -1;
    if (other >= 0) grid_at(bx_[other], by_[other], bslot_[other]) = -1;
    int oslot = (revert_other_slot >= 0) ? revert_other_slot : bslot_saved_;
    bx_[blk] = x1; by_[blk] = y1; bslot_[blk] = slot1;
    grid_at(x1, y1, slot1) = blk;
    if (other >= 0) {
      bx_[other] = x0; by_[other] = y0; bslot_[other] = oslot;
      grid_at(x0, y0, oslot) = other;
    }
  }

  // net marking for dedup within a move
  std::vector<uint8_t> net_mark_, net_mark2_;
  std::vector<int> marked2_;
  bool mark_net(int n) {
    if (net_mark_.empty()) net_mark_.assign(nl_->num_nets, 0);
    if (net_mark_[n]) return false;
    net_mark_[n] = 1;
    return true;
  }
  void unmark_net(int n) { net_mark_[n] = 0; }
  bool mark_net2(int n) {
    if (net_mark2_.empty()) net_mark2_.assign(nl_->num_nets, 0);
    if (net_mark2_[n]) return false;
    net_mark2_[n] = 1; marked2_.push_back(n);
    return true;
  }

  std::vector<Bb> saved_bbs_;
  std::vector<float> saved_costs_;
  std::vector<int> saved_ids_;
  double delta_sum_ = 0, delta_sq_sum_ = 0;
  int64_t delta_n_ = 0;
};

}  // namespace pnr
