// Routing-resource graph builder for segmented unidirectional island FPGAs.
//
// Re-implements the semantics of the reference's build_rr_graph
// (vpr/SRC/route/rr_graph.c:385, rr_graph2.c) for the unidir segmented case,
// designed MI355X-first: flat SoA arrays + CSR edges built in two passes
// (count, fill) so the result uploads to HBM without any transformation.
//
// Fabric model (see parallel_eda_amd/arch/archdef.py for the grid convention):
//  * CHANX channel y in 0..ny spans x in 1..nx; CHANY channel x in 0..nx
//    spans y in 1..ny.  W tracks per channel; track t direction is INC for
//    even t, DEC for odd t; wire length L with stagger offset (t/2) % L.
//  * A wire is driven (mux) at its start SB and can be left at its end SB
//    (switch-block edges, Fs=3: one outgoing wire on each other side,
//    deterministic rotation) or at any tile along its span (IPIN edges).
//  * OPINs drive wires that START at their tile; IPINs tap any wire
//    covering their tile.  fc_in/fc_out are absolute track counts.
#include <numeric>

#include "pnr.h"

namespace pnr {

namespace {

struct ChanWires {
  // node id of the wire in (track t, pos p), p in 1..N; -1 if absent
  // flat index: t*N + (p-1)
  std::vector<int32_t> node;
  int N = 0;
  int32_t at(int t, int p) const { return node[(size_t)t * N + (p - 1)]; }
  int32_t& at(int t, int p) { return node[(size_t)t * N + (p - 1)]; }
};

// Span arithmetic for track t (stagger s0 = (t/2)%L) in a channel 1..N:
// span boundaries at a==1 or (a-1-s0) % L == 0.
inline bool is_span_start(int pos, int t, int L) {
  int s0 = (t / 2) % L;
  if (pos == 1) return true;
  int m = (pos - 1 - s0) % L;
  return m == 0;
}
inline int span_low(int pos, int t, int L) {
  // largest start <= pos
  for (int a = pos; a >= 1; --a)
    if (is_span_start(a, t, L)) return a;
  return 1;
}
inline int span_high(int pos, int t, int L, int N) {
  for (int b = pos + 1; b <= N; ++b)
    if (is_span_start(b, t, L)) return b - 1;
  return N;
}

}  // namespace

RRGraph build_rr_graph(const ArchParams& ap) {
  RRGraph g;
  g.nx = ap.nx; g.ny = ap.ny; g.W = ap.W; g.L = ap.L;
  const int nx = ap.nx, ny = ap.ny, W = ap.W, L = ap.L;
  if (W % 2) throw std::runtime_error("W must be even");
  // Per-track wire length: the first w_l1 tracks are LENGTH-1 segments,
  // the rest length L. See ArchParams::w_l1 for why a single-length
  // fabric cannot route locally.
  int w_l1 = ap.w_l1;
  if (w_l1 < 0) {
    // auto: W/8 length-1 tracks, but only where phase confinement can
    // occur at all — on grids within ~2L of the edge every wire is
    // clipped and already phase-mixes, so tiny fabrics keep the exact
    // single-length channel the arch asked for
    bool interior = L > 1 && nx > 4 * L && ny > 4 * L;
    w_l1 = interior ? (((W / 8) & ~1) < 2 ? 2 : ((W / 8) & ~1)) : 0;
  }
  if (w_l1 > W) w_l1 = W;
  auto track_L = [w_l1, L](int t) { return t < w_l1 ? 1 : L; };

  for (int i = 0; i < 6; ++i) g.base_cost[i] = ap.base_cost[i];
  // Switch table
  g.sw_R[SW_ZERO] = 0;        g.sw_Cin[SW_ZERO] = 0;          g.sw_Tdel[SW_ZERO] = 0;        g.sw_buffered[SW_ZERO] = 1;
  g.sw_R[SW_SB] = ap.R_sw;    g.sw_Cin[SW_SB] = ap.C_sw_in;   g.sw_Tdel[SW_SB] = ap.T_sw;    g.sw_buffered[SW_SB] = 1;
  g.sw_R[SW_OPIN] = ap.R_sw;  g.sw_Cin[SW_OPIN] = ap.C_sw_in; g.sw_Tdel[SW_OPIN] = ap.T_opin; g.sw_buffered[SW_OPIN] = 1;
  g.sw_R[SW_IPIN] = 0;        g.sw_Cin[SW_IPIN] = ap.C_sw_in; g.sw_Tdel[SW_IPIN] = ap.T_ipin; g.sw_buffered[SW_IPIN] = 1;

  const int gx = nx + 2, gy = ny + 2;
  g.tile_source.assign((size_t)gx * gy, -1);
  g.tile_sink.assign((size_t)gx * gy, -1);

  // ---------------- pass 0: enumerate nodes ----------------
  auto add_node = [&](int8_t ty, int xl, int yl, int xh, int yh, int ptc,
                      int cap, float R, float C) -> int32_t {
    g.type.push_back(ty);
    g.xlow.push_back((int16_t)xl); g.ylow.push_back((int16_t)yl);
    g.xhigh.push_back((int16_t)xh); g.yhigh.push_back((int16_t)yh);
    g.ptc.push_back((int16_t)ptc);
    g.capacity.push_back((int16_t)cap);
    g.R.push_back(R); g.C.push_back(C);
    return (int32_t)g.type.size() - 1;
  };

  // Per-tile pin nodes. opin_nodes/ipin_nodes: flat per tile, variable count.
  // We keep small per-tile vectors during build only.
  struct TilePins { int32_t src = -1, snk = -1; std::vector<int32_t> opins, ipins;
                    std::vector<int8_t> opin_side, ipin_side; };
  std::vector<TilePins> tiles((size_t)gx * gy);

  // Which tiles exist and their pin counts/sides.
  // side: 0=BOTTOM(ChanX y-1) 1=RIGHT(ChanY x) 2=TOP(ChanX y) 3=LEFT(ChanY x-1)
  auto make_tile = [&](int x, int y, int n_in, int n_out, int cap,
                       std::vector<int8_t> sides) {
    TilePins& tp = tiles[g.tile_id(x, y)];
    tp.src = add_node(SOURCE, x, y, x, y, 0, n_out > cap ? n_out : cap, 0, 0);
    tp.snk = add_node(SINK, x, y, x, y, 0, n_in > cap ? n_in : cap, 0, 0);
    g.tile_source[g.tile_id(x, y)] = tp.src;
    g.tile_sink[g.tile_id(x, y)] = tp.snk;
    for (int p = 0; p < n_out; ++p) {
      int8_t s = sides[p % sides.size()];
      tp.opins.push_back(add_node(OPIN, x, y, x, y, p, 1, 0, 0));
      tp.opin_side.push_back(s);
    }
    for (int p = 0; p < n_in; ++p) {
      int8_t s = sides[p % sides.size()];
      tp.ipins.push_back(add_node(IPIN, x, y, x, y, p, 1, 0, 0));
      tp.ipin_side.push_back(s);
    }
  };

  // Logic tiles: per-column block type (CLB / RAM column / DSP column;
  // homogeneous when ram_col_every == dsp_col_every == 0)
  for (int x = 1; x <= nx; ++x) {
    int bt = col_btype(ap, x);
    int t_in = ap.clb_in, t_out = ap.clb_out;
    if (bt == 2) { t_in = ap.ram_in; t_out = ap.ram_out; }
    else if (bt == 3) { t_in = ap.dsp_in; t_out = ap.dsp_out; }
    for (int y = 1; y <= ny; ++y)
      make_tile(x, y, t_in, t_out, 1, {0, 1, 2, 3});
  }
  // IO tiles: io_cap slots, each with 1 OPIN + 1 IPIN, one facing side.
  for (int y = 1; y <= ny; ++y) {
    make_tile(0, y, ap.io_cap, ap.io_cap, ap.io_cap, {1});       // left edge faces RIGHT
    make_tile(nx + 1, y, ap.io_cap, ap.io_cap, ap.io_cap, {3});  // right edge faces LEFT
  }
  for (int x = 1; x <= nx; ++x) {
    make_tile(x, 0, ap.io_cap, ap.io_cap, ap.io_cap, {2});       // bottom edge faces TOP
    make_tile(x, ny + 1, ap.io_cap, ap.io_cap, ap.io_cap, {0});  // top edge faces BOTTOM
  }

  // Channel wires. chanx[y], chany[x].
  std::vector<ChanWires> chanx(ny + 1), chany(nx + 1);
  for (int y = 0; y <= ny; ++y) {
    chanx[y].N = nx; chanx[y].node.assign((size_t)W * nx, -1);
    for (int t = 0; t < W; ++t) {
      int p = 1;
      while (p <= nx) {
        int b = span_high(p, t, track_L(t), nx);
        float len = (float)(b - p + 1);
        int32_t id = add_node(CHANX, p, y, b, y, t, 1,
                              ap.R_wire * len, ap.C_wire * len);
        for (int q = p; q <= b; ++q) chanx[y].at(t, q) = id;
        p = b + 1;
      }
    }
  }
  for (int x = 0; x <= nx; ++x) {
    chany[x].N = ny; chany[x].node.assign((size_t)W * ny, -1);
    for (int t = 0; t < W; ++t) {
      int p = 1;
      while (p <= ny) {
        int b = span_high(p, t, track_L(t), ny);
        float len = (float)(b - p + 1);
        int32_t id = add_node(CHANY, x, p, x, b, t, 1,
                              ap.R_wire * len, ap.C_wire * len);
        for (int q = p; q <= b; ++q) chany[x].at(t, q) = id;
        p = b + 1;
      }
    }
  }

  g.num_nodes = (int)g.type.size();

  // ---------------- edge generation (two passes) ----------------
  // Precomputed per-position track lists (span arithmetic is O(L) per
  // query; at Titan scale the naive form dominated the build at ~18 Gops).
  // starts_at[N][pos]: tracks whose wire is DRIVEN at tile pos
  //   (INC with span_low==pos, DEC with span_high==pos)
  // inc_end[pos]: INC tracks whose wire ENDS (xhigh) at pos
  // dec_lowstart[pos]: DEC tracks whose span_low == pos
  struct TrackLists {
    std::vector<std::vector<int16_t>> starts_at, inc_end, dec_lowstart;
    void build(int N, int W, int w_l1, int L) {
      auto track_L = [w_l1, L](int t) { return t < w_l1 ? 1 : L; };
      starts_at.assign(N + 1, {});
      inc_end.assign(N + 1, {});
      dec_lowstart.assign(N + 1, {});
      for (int pos = 1; pos <= N; ++pos)
        for (int t = 0; t < W; ++t) {
          int Lt = track_L(t);
          if ((t & 1) == 0) {
            if (span_low(pos, t, Lt) == pos) starts_at[pos].push_back(t);
            if (span_high(pos, t, Lt, N) == pos) inc_end[pos].push_back(t);
          } else {
            if (span_high(pos, t, Lt, N) == pos) starts_at[pos].push_back(t);
            if (span_low(pos, t, Lt) == pos) dec_lowstart[pos].push_back(t);
          }
        }
    }
  };
  TrackLists tlx, tly;
  tlx.build(nx, W, w_l1, L);
  tly.build(ny, W, w_l1, L);
  // channel access for a tile side: returns (is_x, chan_index, pos)
  struct SideRef { bool is_x; int chan; int pos; bool valid; };
  auto side_ref = [&](int x, int y, int8_t side) -> SideRef {
    switch (side) {
      case 0: return {true, y - 1, x, y - 1 >= 0};   // BOTTOM -> CHANX y-1
      case 2: return {true, y, x, y <= ny};          // TOP -> CHANX y
      case 3: return {false, x - 1, y, x - 1 >= 0};  // LEFT -> CHANY x-1
      default: return {false, x, y, x <= nx};        // RIGHT -> CHANY x
    }
  };
  auto chan_node = [&](bool is_x, int chan, int t, int pos) -> int32_t {
    return is_x ? chanx[chan].at(t, pos) : chany[chan].at(t, pos);
  };

  // Incoming-wire candidates per SB(i,j) by arrival side.
  // side: 0=W (INC CHANX ending at SB, travelled E), 1=E (DEC CHANX ending),
  //       2=S (INC CHANY ending, travelled N), 3=N (DEC CHANY ending)
  auto sb_in_list = [&](int i, int j, int side, std::vector<int32_t>& in) {
    in.clear();
    if (side == 0) { if (i < 1) return;
      for (int16_t t : tlx.inc_end[i]) in.push_back(chanx[j].at(t, i));
    } else if (side == 1) { if (i + 1 > nx) return;
      for (int16_t t : tlx.dec_lowstart[i + 1]) in.push_back(chanx[j].at(t, i + 1));
    } else if (side == 2) { if (j < 1) return;
      for (int16_t t : tly.inc_end[j]) in.push_back(chany[i].at(t, j));
    } else { if (j + 1 > ny) return;
      for (int16_t t : tly.dec_lowstart[j + 1]) in.push_back(chany[i].at(t, j + 1));
    }
  };

  // Generation driver: calls emit(src, dst, sw) for every edge.
  auto generate = [&](auto&& emit) {
    std::vector<int32_t> cand;
    // SOURCE -> OPIN, IPIN -> SINK
    for (int x = 0; x < gx; ++x) for (int y = 0; y < gy; ++y) {
      TilePins& tp = tiles[g.tile_id(x, y)];
      if (tp.src < 0) continue;
      for (int32_t o : tp.opins) emit(tp.src, o, SW_ZERO);
      for (int32_t i : tp.ipins) emit(i, tp.snk, SW_ZERO);
      // OPIN -> wire starts
      for (size_t pi = 0; pi < tp.opins.size(); ++pi) {
        SideRef sr = side_ref(x, y, tp.opin_side[pi]);
        if (!sr.valid) continue;
        cand.clear();
        const auto& starts = (sr.is_x ? tlx : tly).starts_at[sr.pos];
        for (int16_t t : starts)
          cand.push_back(chan_node(sr.is_x, sr.chan, t, sr.pos));
        if (cand.empty()) continue;
        int n = (int)cand.size();
        int fc = ap.fc_out < n ? ap.fc_out : n;
        int off = (x * 131 + y * 31 + (int)pi * 7) % n;
        for (int k = 0; k < fc; ++k)
          emit(tp.opins[pi], cand[(off + k * n / fc) % n], SW_OPIN);
      }
      // wire -> IPIN (from the IPIN's perspective)
      for (size_t pi = 0; pi < tp.ipins.size(); ++pi) {
        SideRef sr = side_ref(x, y, tp.ipin_side[pi]);
        if (!sr.valid) continue;
        int fc = ap.fc_in < W ? ap.fc_in : W;
        int off = (x * 71 + y * 37 + (int)pi * 13) % W;
        for (int k = 0; k < fc; ++k) {
          int t = (off + k * W / fc) % W;
          emit(chan_node(sr.is_x, sr.chan, t, sr.pos), tp.ipins[pi], SW_IPIN);
        }
      }
    }
    // Switch blocks, constructed per OUTGOING wire: each wire's driver mux
    // at its start SB takes (a) the SAME-TRACK straight continuation when
    // it exists (the standard unidir "straight through" — guarantees long
    // straight chains to the chip edge), and (b) one incoming wire from
    // each other non-U-turn side, selected BIJECTIVELY by the out-wire's
    // position in its side's start list (a pure modular-track rotation can
    // systematically orphan the fan-out of whole track classes: a fuzzed
    // 3x7/W28/L3 fabric had sources reaching <10% of sinks).
    // out_dir: 0=E 1=W 2=N 3=S.
    auto connect_out_wire = [&](int32_t wnode, int sb_i, int sb_j, int out_dir,
                                int out_pos) {
      static const int ins[4][3] = {
          {0, 2, 3},   // out E: in W(straight), S, N
          {1, 2, 3},   // out W: in E(straight), S, N
          {2, 0, 1},   // out N: in S(straight), W, E
          {3, 0, 1}};  // out S: in N(straight), W, E
      int t_out = g.ptc[wnode];
      for (int k = 0; k < 3; ++k) {
        int s = ins[out_dir][k];
        sb_in_list(sb_i, sb_j, s, cand);
        if (cand.empty()) continue;
        int n = (int)cand.size();
        if (k == 0) {
          // straight side: the same track's incoming wire (standard
          // unidir straight-through; guarantees long straight chains)
          int idx = -1;
          for (int c = 0; c < n; ++c)
            if (g.ptc[cand[c]] == t_out) { idx = c; break; }
          if (idx < 0) idx = (out_pos + sb_i + 2 * sb_j) % n;
          emit(cand[idx], wnode, SW_SB);
          continue;
        }
        // TURN sides take sb_turn_fanin in-wires each (default 1: the
        // bijective rotation). NOTE the permutation structure here is NOT
        // what limits local routability — the wire-length mix is (see
        // w_l1 above): a single-length fabric moves in strides of exactly
        // L, so forward-reachable and can-reach-sink wire sets inside a
        // bb live on disjoint (mod L, mod L) SB sublattices (measured at
        // bitcoin scale: zero intersection for 11% of nets, which then
        // needed bb margin ~124 — the chip edge — to route). Extra picks
        // (fan-in > 1) spread by an affine map with a step coprime to the
        // list size.
        int picks = ap.sb_turn_fanin < n ? ap.sb_turn_fanin : n;
        int base = (out_pos + k + sb_i + 2 * sb_j) % n;
        if (picks == 1) {
          emit(cand[base], wnode, SW_SB);
        } else {
          int b = 1 + ((sb_i + sb_j * 11 + k * 3) % 8);
          while (std::gcd(b, n) != 1) ++b;
          for (int r = 0; r < picks; ++r)
            emit(cand[(base + r * b) % n], wnode, SW_SB);
        }
      }
    };
    for (int y = 0; y <= ny; ++y) {
      std::vector<int> pos_inc(nx + 2, 0), pos_dec(nx + 2, 0);
      for (int t = 0; t < W; ++t) {
        int p = 1;
        while (p <= nx) {
          int b = span_high(p, t, track_L(t), nx);
          int32_t id = chanx[y].at(t, p);
          if ((t & 1) == 0) connect_out_wire(id, p - 1, y, 0, pos_inc[p]++);
          else connect_out_wire(id, b, y, 1, pos_dec[b]++);
          p = b + 1;
        }
      }
    }
    for (int x = 0; x <= nx; ++x) {
      std::vector<int> pos_inc(ny + 2, 0), pos_dec(ny + 2, 0);
      for (int t = 0; t < W; ++t) {
        int p = 1;
        while (p <= ny) {
          int b = span_high(p, t, track_L(t), ny);
          int32_t id = chany[x].at(t, p);
          if ((t & 1) == 0) connect_out_wire(id, x, p - 1, 2, pos_inc[p]++);
          else connect_out_wire(id, x, b, 3, pos_dec[b]++);
          p = b + 1;
        }
      }
    }
  };

  // pass 1: count
  std::vector<int32_t> deg(g.num_nodes, 0);
  int64_t total = 0;
  generate([&](int32_t s, int32_t d, int8_t sw) { (void)d; (void)sw; deg[s]++; total++; });
  g.row_ptr.assign(g.num_nodes + 1, 0);
  for (int i = 0; i < g.num_nodes; ++i) g.row_ptr[i + 1] = g.row_ptr[i] + deg[i];
  g.num_edges = total;
  g.edge_dst.assign(total, -1);
  g.edge_sw.assign(total, 0);
  // pass 2: fill
  std::vector<int64_t> cur(g.row_ptr.begin(), g.row_ptr.end() - 1);
  generate([&](int32_t s, int32_t d, int8_t sw) {
    int64_t at = cur[s]++;
    g.edge_dst[at] = d; g.edge_sw[at] = sw;
  });
  int dm = 0;
  for (int i = 0; i < g.num_nodes; ++i) if (deg[i] > dm) dm = deg[i];
  g.degree_max = dm;
  return g;
}

}  // namespace pnr
