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

int64_t pnr_sta_args_sizeof() { return (int64_t)sizeof(StaLaunchArgs); }

}  // extern "C"
