// MFMA (matrix-core) kernels — gfx950 f32 16x16x4 tiles.
//
// Used for the routed placement delay-matrix post-pass (reference:
// place/timing_place_lookup.c:981 compute_delay_lookup_tables — the
// (dx,dy)->delay table the SA placer's timing cost reads): the raw
// router-measured matrix is smoothed with separable band-averaging
// expressed as two dense products R = S_r * D * S_c^T, which puts the
// work on the MFMA pipes (v_mfma_f32_16x16x4_f32; one wave64 computes a
// 16x16 C tile, K stepped by 4).
//
// Lane maps (cdna_hip_programming.md / cdna4_isa.md section 10):
//   A[l&15][l>>4]   B[l>>4][l&15]   C/D: col = l&15, row = (l>>4)*4 + i
#include <hip/hip_runtime.h>
#include <cstdint>

namespace pnrh {

using f32x4 = __attribute__((ext_vector_type(4))) float;

__launch_bounds__(64, 4)
__global__ void mfma_gemm_f32_kernel(const float* __restrict__ A,
                                     const float* __restrict__ B,
                                     float* __restrict__ C,
                                     int M, int N, int K) {
  const int tm = blockIdx.x * 16;
  const int tn = blockIdx.y * 16;
  const int lane = threadIdx.x;   // wave64
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int ar = tm + (lane & 15);
  const int bc = tn + (lane & 15);
  const int ks = lane >> 4;       // 0..3
  for (int k0 = 0; k0 < K; k0 += 4) {
    const int ak = k0 + ks;
    float a = (ar < M && ak < K) ? A[(int64_t)ar * K + ak] : 0.f;
    float b = (ak < K && bc < N) ? B[(int64_t)ak * N + bc] : 0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  const int col = tn + (lane & 15);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int row = tm + (lane >> 4) * 4 + i;
    if (row < M && col < N) C[(int64_t)row * N + col] = acc[i];
  }
}

}  // namespace pnrh

extern "C" {

// C = A(MxK) * B(KxN), f32 row-major, any sizes (tiles masked).
int pnr_mfma_gemm_f32(const float* A, const float* B, float* C,
                      int M, int N, int K, void* stream) {
  dim3 grid((M + 15) / 16, (N + 15) / 16);
  hipLaunchKernelGGL(pnrh::mfma_gemm_f32_kernel, grid, dim3(64), 0,
                     (hipStream_t)stream, A, B, C, M, N, K);
  return (int)hipGetLastError();
}

}  // extern "C"
