// CDNA4 static timing analysis — level-synchronous sweeps.
//
// Re-implements the reference STA (vpr/SRC/timing/path_delay.c:1994
// do_timing_analysis_new; levels path_delay2.c:81) as GPU kernels over the
// block-granularity timing graph: one kernel launch per level for the
// forward T_arr max-plus sweep and the backward T_req min-minus sweep,
// then a per-connection slack/criticality kernel. Levels and CSRs are
// built once on the host (csrc/cpu/sta_serial.cpp TimingGraph) and stay
// resident in HBM.
#include "pnr_hip.h"

namespace pnrh {

struct StaDev {
  const int32_t* level_blocks;   // blocks sorted by level
  const int32_t* level_start;    // [num_levels+1]
  const int64_t* in_ptr;         // per block incoming conn CSR
  const int64_t* in_conn;
  const int64_t* out_ptr;        // per block outgoing conn CSR
  const int64_t* out_conn;
  const int32_t* conn_driver;    // per conn
  const int32_t* conn_sink;      // per conn
  const uint8_t* is_seq;
  const float* blk_delay;        // per-block comb delay; nullptr => T_clb
  float T_clb, T_seq_out, T_seq_in;
  int32_t num_blocks;
  int64_t num_conns;
  float* t_arr;                  // per block (output arrival)
  float* t_req;                  // per block (output required)
  float* cpd_out;                // [1]
};

__global__ void sta_forward_level(StaDev s, const float* __restrict__ delay,
                                  int32_t lv0, int32_t lv1) {
  int i = lv0 + blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= lv1) return;
  int32_t b = s.level_blocks[i];
  if (s.is_seq[b]) { s.t_arr[b] = s.T_seq_out; return; }
  float a = 0.0f;
  for (int64_t k = s.in_ptr[b]; k < s.in_ptr[b + 1]; ++k) {
    int64_t c = s.in_conn[k];
    float v = s.t_arr[s.conn_driver[c]] + delay[c];
    a = fmaxf(a, v);
  }
  s.t_arr[b] = a + (s.blk_delay ? s.blk_delay[b] : s.T_clb);
}

// cpd = max over seq endpoints of (input arrival + T_seq_in)
__global__ void sta_cpd_kernel(StaDev s, const float* __restrict__ delay) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  __shared__ unsigned smax;
  if (threadIdx.x == 0) smax = 0;
  __syncthreads();
  for (; b < s.num_blocks; b += gridDim.x * blockDim.x) {
    if (!s.is_seq[b]) continue;
    float a = 0.0f;
    for (int64_t k = s.in_ptr[b]; k < s.in_ptr[b + 1]; ++k) {
      int64_t c = s.in_conn[k];
      a = fmaxf(a, s.t_arr[s.conn_driver[c]] + delay[c]);
    }
    float v = a + s.T_seq_in;
    atomicMax(&smax, __float_as_uint(v));  // v >= 0: bits are ordered
  }
  __syncthreads();
  if (threadIdx.x == 0 && smax)
    atomicMax((unsigned*)s.cpd_out, smax);
}

__global__ void sta_backward_level(StaDev s, const float* __restrict__ delay,
                                   int32_t lv0, int32_t lv1) {
  int i = lv0 + blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= lv1) return;
  int32_t b = s.level_blocks[i];
  float cpd = s.cpd_out[0];
  float r = 3.0e38f;
  for (int64_t k = s.out_ptr[b]; k < s.out_ptr[b + 1]; ++k) {
    int64_t c = s.out_conn[k];
    int32_t snk = s.conn_sink[c];
    float req_in = s.is_seq[snk]
        ? (cpd - s.T_seq_in)
        : (s.t_req[snk] - (s.blk_delay ? s.blk_delay[snk] : s.T_clb));
    r = fminf(r, req_in - delay[c]);
  }
  s.t_req[b] = r;
}

__global__ void sta_slack_kernel(StaDev s, const float* __restrict__ delay,
                                 float* slack, float* crit, float max_crit) {
  int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  float cpd = s.cpd_out[0];
  float inv_cpd = 1.0f / fmaxf(cpd, 1e-12f);
  for (; c < s.num_conns; c += (int64_t)gridDim.x * blockDim.x) {
    int32_t drv = s.conn_driver[c];
    int32_t snk = s.conn_sink[c];
    float req_in = s.is_seq[snk]
        ? (cpd - s.T_seq_in)
        : (s.t_req[snk] - (s.blk_delay ? s.blk_delay[snk] : s.T_clb));
    float sl = req_in - (s.t_arr[drv] + delay[c]);
    slack[c] = sl;
    float cr = 1.0f - sl * inv_cpd;
    crit[c] = fminf(fmaxf(cr, 0.0f), max_crit);
  }
}

// ---- multi-clock analysis (reference: do_timing_analysis_new's
// (src,sink)-domain-pair loop, path_delay.c:1996-2085) ----
// Per-domain arrival/required arrays [K][num_blocks]; forward/backward
// level sweeps per domain, then one conn x pair slack/criticality pass.

__global__ void sta_fwd_domain_level(StaDev s, const float* __restrict__ delay,
                                     const int32_t* __restrict__ block_clock,
                                     float* __restrict__ arr, int ci,
                                     int32_t lv0, int32_t lv1) {
  const float NEG = -3.0e38f;
  int i = lv0 + blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= lv1) return;
  int32_t b = s.level_blocks[i];
  if (s.is_seq[b]) {
    arr[b] = (block_clock[b] == ci) ? s.T_seq_out : NEG;
    return;
  }
  float m = NEG;
  for (int64_t k = s.in_ptr[b]; k < s.in_ptr[b + 1]; ++k) {
    int64_t c = s.in_conn[k];
    float v = arr[s.conn_driver[c]];
    if (v > NEG) v += delay[c];
    m = fmaxf(m, v);
  }
  float td = s.blk_delay ? s.blk_delay[b] : s.T_clb;
  arr[b] = (m > NEG) ? m + td : NEG;
}

__global__ void sta_bwd_domain_level(StaDev s, const float* __restrict__ delay,
                                     const int32_t* __restrict__ block_clock,
                                     float* __restrict__ req, int cj,
                                     float period,
                                     int32_t lv0, int32_t lv1) {
  const float POS = 3.0e38f;
  int i = lv0 + blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= lv1) return;
  int32_t b = s.level_blocks[i];
  float req_ep = period - s.T_seq_in;
  float m = POS;
  for (int64_t k = s.out_ptr[b]; k < s.out_ptr[b + 1]; ++k) {
    int64_t c = s.out_conn[k];
    int32_t snk = s.conn_sink[c];
    float ri;
    if (s.is_seq[snk]) {
      ri = (block_clock[snk] == cj) ? req_ep : POS;
    } else {
      float td = s.blk_delay ? s.blk_delay[snk] : s.T_clb;
      ri = (req[snk] < POS) ? req[snk] - td : POS;
    }
    if (ri < POS) ri -= delay[c];
    m = fminf(m, ri);
  }
  req[b] = m;
}

// one thread per (conn, ci, cj): worst slack / max criticality per conn
__global__ void sta_slack_domains(StaDev s, const float* __restrict__ delay,
                                  const int32_t* __restrict__ block_clock,
                                  const float* __restrict__ arr_all,
                                  const float* __restrict__ req_all,
                                  const float* __restrict__ periods, int K,
                                  float* slack, float* crit, float max_crit,
                                  unsigned* worst_bits /*achieved period*/) {
  const float NEG = -3.0e38f, POS = 3.0e38f;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = s.num_conns * (int64_t)K * K;
  for (; idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t c = idx / (K * K);
    int pair = (int)(idx % (K * K));
    int ci = pair / K, cj = pair % K;
    const float* arr = arr_all + (int64_t)ci * s.num_blocks;
    const float* req = req_all + (int64_t)cj * s.num_blocks;
    int32_t drv = s.conn_driver[c];
    int32_t snk = s.conn_sink[c];
    if (arr[drv] <= NEG) continue;
    float constraint = periods[cj];
    float ri;
    if (s.is_seq[snk]) {
      ri = (block_clock[snk] == cj) ? constraint - s.T_seq_in : POS;
    } else {
      float td = s.blk_delay ? s.blk_delay[snk] : s.T_clb;
      ri = (req[snk] < POS) ? req[snk] - td : POS;
    }
    if (ri >= POS) continue;
    float sl = ri - (arr[drv] + delay[c]);
    // min-slack per conn via atomicMin on ordered float bits
    // (slack can be negative: flip bits accordingly)
    unsigned sb = __float_as_uint(sl);
    sb = (sb & 0x80000000u) ? ~sb : (sb | 0x80000000u);
    atomicMin((unsigned*)&slack[c], sb);
    float cr = 1.0f - sl / constraint;
    cr = fminf(fmaxf(cr, 0.0f), max_crit);
    atomicMax((unsigned*)&crit[c], __float_as_uint(cr));  // cr >= 0
    float achieved = constraint - sl;
    // track the worst achieved/constraint ratio's achieved period
    unsigned ab = __float_as_uint(achieved / constraint);
    unsigned prev = atomicMax(&worst_bits[0], ab);
    if (ab > prev) worst_bits[1] = __float_as_uint(achieved);
  }
}

__global__ void sta_slack_domains_finish(StaDev s, float* slack) {
  int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; c < s.num_conns; c += (int64_t)gridDim.x * blockDim.x) {
    unsigned sb = __float_as_uint(slack[c]);
    // undo the ordered-bits encoding; untouched sentinel (all ones
    // from the init fill) -> slack 0
    if (sb == 0xFFFFFFFFu) { slack[c] = 0.0f; continue; }
    sb = (sb & 0x80000000u) ? (sb & 0x7FFFFFFFu) : ~sb;
    slack[c] = __uint_as_float(sb);
  }
}

}  // namespace pnrh

using namespace pnrh;

extern "C" {

struct StaLaunchArgs {
  const int32_t* level_blocks; const int32_t* level_start;
  const int64_t* in_ptr; const int64_t* in_conn;
  const int64_t* out_ptr; const int64_t* out_conn;
  const int32_t* conn_driver; const int32_t* conn_sink;
  const uint8_t* is_seq;
  const float* blk_delay;  // nullptr => scalar T_clb
  float T_clb, T_seq_out, T_seq_in, max_crit;
  int32_t num_blocks, num_levels;
  int64_t num_conns;
  float* t_arr; float* t_req; float* cpd_out;
  const float* delay; float* slack; float* crit;
  const int32_t* level_start_host;  // host copy for launch bounds
};

int pnr_sta_analyze(const StaLaunchArgs* a, void* stream) {
  StaDev s{a->level_blocks, a->level_start, a->in_ptr, a->in_conn,
           a->out_ptr, a->out_conn, a->conn_driver, a->conn_sink, a->is_seq,
           a->blk_delay,
           a->T_clb, a->T_seq_out, a->T_seq_in, a->num_blocks, a->num_conns,
           a->t_arr, a->t_req, a->cpd_out};
  hipStream_t st = (hipStream_t)stream;
  const int32_t* ls = a->level_start_host;
  hipError_t me = hipMemsetAsync(a->cpd_out, 0, sizeof(float), st);
  if (me != hipSuccess) return (int)me;
  for (int lv = 0; lv < a->num_levels; ++lv) {
    int n = ls[lv + 1] - ls[lv];
    if (n <= 0) continue;
    hipLaunchKernelGGL(sta_forward_level, dim3((n + 255) / 256), dim3(256), 0,
                       st, s, a->delay, ls[lv], ls[lv + 1]);
  }
  int cg = (a->num_blocks + 255) / 256;
  cg = cg < 2048 ? cg : 2048;
  hipLaunchKernelGGL(sta_cpd_kernel, dim3(cg), dim3(256), 0, st, s, a->delay);
  for (int lv = a->num_levels - 1; lv >= 0; --lv) {
    int n = ls[lv + 1] - ls[lv];
    if (n <= 0) continue;
    hipLaunchKernelGGL(sta_backward_level, dim3((n + 255) / 256), dim3(256), 0,
                       st, s, a->delay, ls[lv], ls[lv + 1]);
  }
  int64_t sg64 = (a->num_conns + 255) / 256;
  int sg = (int)(sg64 < 2048 ? sg64 : 2048);
  hipLaunchKernelGGL(sta_slack_kernel, dim3(sg), dim3(256), 0, st,
                     s, a->delay, a->slack, a->crit, a->max_crit);
  return (int)hipGetLastError();
}

// multi-clock entry (separate args; the single-domain path is untouched)
struct StaDomainsArgs {
  StaLaunchArgs base;           // graph + buffers (t_arr/t_req unused)
  const int32_t* block_clock;   // per block, -1 comb
  const float* periods;         // [K] device
  int32_t K;
  float* arr_all;               // [K * num_blocks]
  float* req_all;               // [K * num_blocks]
  unsigned* worst_bits;         // [2] device scratch (ratio bits, achieved)
};

int pnr_sta_analyze_domains(const StaDomainsArgs* d, void* stream) {
  const StaLaunchArgs* a = &d->base;
  StaDev s{a->level_blocks, a->level_start, a->in_ptr, a->in_conn,
           a->out_ptr, a->out_conn, a->conn_driver, a->conn_sink, a->is_seq,
           a->blk_delay,
           a->T_clb, a->T_seq_out, a->T_seq_in, a->num_blocks, a->num_conns,
           a->t_arr, a->t_req, a->cpd_out};
  hipStream_t st = (hipStream_t)stream;
  const int32_t* ls = a->level_start_host;
  hipError_t me;
  me = hipMemsetAsync(d->worst_bits, 0, 2 * sizeof(unsigned), st);
  if (me != hipSuccess) return (int)me;
  // slack sentinel: all-ones encoded bits (= +inf in the ordering);
  // crit zero
  me = hipMemsetAsync(a->slack, 0xFF, a->num_conns * sizeof(float), st);
  if (me != hipSuccess) return (int)me;
  me = hipMemsetAsync(a->crit, 0, a->num_conns * sizeof(float), st);
  if (me != hipSuccess) return (int)me;
  for (int ci = 0; ci < d->K; ++ci) {
    float* arr = d->arr_all + (int64_t)ci * a->num_blocks;
    for (int lv = 0; lv < a->num_levels; ++lv) {
      int n = ls[lv + 1] - ls[lv];
      if (n <= 0) continue;
      hipLaunchKernelGGL(sta_fwd_domain_level, dim3((n + 255) / 256),
                         dim3(256), 0, st, s, a->delay, d->block_clock,
                         arr, ci, ls[lv], ls[lv + 1]);
    }
  }
  // per-domain constraints live on the host too? periods are device;
  // backward needs the scalar period per cj: read from a host mirror is
  // not available here, so the caller passes periods ALSO via
  // base.max_crit-sized side channel — instead we launch with a device
  // gather: copy periods to host once per call.
  float periods_h[64];
  if (d->K > 64) return (int)hipErrorInvalidValue;
  me = hipMemcpyAsync(periods_h, d->periods, d->K * sizeof(float),
                      hipMemcpyDeviceToHost, st);
  if (me != hipSuccess) return (int)me;
  me = hipStreamSynchronize(st);
  if (me != hipSuccess) return (int)me;
  for (int cj = 0; cj < d->K; ++cj) {
    float* req = d->req_all + (int64_t)cj * a->num_blocks;
    for (int lv = a->num_levels - 1; lv >= 0; --lv) {
      int n = ls[lv + 1] - ls[lv];
      if (n <= 0) continue;
      hipLaunchKernelGGL(sta_bwd_domain_level, dim3((n + 255) / 256),
                         dim3(256), 0, st, s, a->delay, d->block_clock,
                         req, cj, periods_h[cj], ls[lv], ls[lv + 1]);
    }
  }
  int64_t total = a->num_conns * (int64_t)d->K * d->K;
  int64_t g64 = (total + 255) / 256;
  int grid = (int)(g64 < 2048 ? g64 : 2048);
  hipLaunchKernelGGL(sta_slack_domains, dim3(grid), dim3(256), 0, st,
                     s, a->delay, d->block_clock, d->arr_all, d->req_all,
                     d->periods, d->K, a->slack, a->crit, a->max_crit,
                     d->worst_bits);
  int64_t c64 = (a->num_conns + 255) / 256;
  int cgrid = (int)(c64 < 2048 ? c64 : 2048);
  hipLaunchKernelGGL(sta_slack_domains_finish, dim3(cgrid), dim3(256), 0, st,
                     s, a->slack);
  return (int)hipGetLastError();
}

int64_t pnr_sta_domains_args_sizeof() { return (int64_t)sizeof(StaDomainsArgs); }

int64_t pnr_sta_args_sizeof() { return (int64_t)sizeof(StaLaunchArgs); }

}  // extern "C"
