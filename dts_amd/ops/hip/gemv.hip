// Skinny-batch (M <= 8) bf16 GEMV/GEMM for decode projections on MI355X.
//
// The decode step's linears are out[M, N] = x[M, K] @ W[N, K]^T with
// M = decode batch width (1-8 on the search workload) — pure
// weight-streaming: W's N*K*2 bytes dominate. hipBLASLt at these M runs
// ~50% of the HBM roofline on gfx950 (measured, profiles/); this kernel
// follows the guide's GEMV row ("operand streamed once per block, not
// shared across waves: load straight to VGPRs, deep unroll, late vmcnt").
//
// Structure: each WAVE owns RPW=4 consecutive output rows. Per k-chunk
// (8 bf16 per lane, 16 B coalesced):
//   - load the 4 W-row fragments (4 x 16 B per lane),
//   - load the M x-row fragments ONCE (amortized over the 4 W rows —
//     the single-row variant re-read x per row, which made it ALU/L2
//     bound beyond M≈2),
//   - 4*M*8 FMAs into fp32 accumulators.
// End: 64-lane shuffle reduction per (row, m). fp32 accumulation,
// bf16 I/O — same numeric class as the hipBLASLt path it replaces.

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

using bf16x8 = __attribute__((ext_vector_type(8))) short;

#define RPW 4  // rows per wave

template <int M>
__global__ void __launch_bounds__(256)
gemv_bf16_kernel(short* __restrict__ out,      // [M, N]
                 const short* __restrict__ x,  // [M, K] (row stride x_ts)
                 const short* __restrict__ w,  // [N, K]
                 int N, int K, long x_tstride, long out_tstride) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int waves_per_wg = blockDim.x / WAVE;
  const int groups_per_grid = gridDim.x * waves_per_wg;

  for (int n0 = (blockIdx.x * waves_per_wg + wave) * RPW; n0 < N;
       n0 += groups_per_grid * RPW) {
    float acc[RPW][M];
#pragma unroll
    for (int r = 0; r < RPW; ++r)
#pragma unroll
      for (int m = 0; m < M; ++m) acc[r][m] = 0.f;

    for (int k0 = lane * 8; k0 < K; k0 += WAVE * 8) {
      float xf[M][8];
#pragma unroll
      for (int m = 0; m < M; ++m) {
        bf16x8 xv = *(const bf16x8*)(x + (long)m * x_tstride + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j) xf[m][j] = bf2f(xv[j]);
      }
#pragma unroll
      for (int r = 0; r < RPW; ++r) {
        const int n = n0 + r;
        if (n >= N) break;
        bf16x8 wv = *(const bf16x8*)(w + (long)n * K + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float wf = bf2f(wv[j]);
#pragma unroll
          for (int m = 0; m < M; ++m) acc[r][m] += xf[m][j] * wf;
        }
      }
    }
#pragma unroll
    for (int r = 0; r < RPW; ++r) {
      const int n = n0 + r;
      if (n >= N) break;
#pragma unroll
      for (int m = 0; m < M; ++m) {
        float v = wave_sum(acc[r][m]);
        if (lane == 0) out[(long)m * out_tstride + n] = f2bf(v);
      }
    }
  }
}

void gemv_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.is_contiguous(), "weight must be contiguous [N, K]");
  TORCH_CHECK(x.stride(1) == 1, "x rows must be dense");
  TORCH_CHECK(out.stride(1) == 1);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(K % 512 == 0, "K must be a multiple of 512");
  TORCH_CHECK(M >= 1 && M <= 8, "gemv path is for M<=8");
  const int row_groups = (N + RPW - 1) / RPW;
  int wgs = std::min((row_groups + 3) / 4, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
#define GEMV_CASE(m)                                                         \
  hipLaunchKernelGGL((gemv_bf16_kernel<m>), dim3(wgs), dim3(256), 0, stream, \
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),     \
                     (const short*)w.data_ptr(), N, K, x.stride(0),          \
                     out.stride(0))
  switch (M) {
    case 1: GEMV_CASE(1); break;
    case 2: GEMV_CASE(2); break;
    case 3: GEMV_CASE(3); break;
    case 4: GEMV_CASE(4); break;
    case 5: GEMV_CASE(5); break;
    case 6: GEMV_CASE(6); break;
    case 7: GEMV_CASE(7); break;
    case 8: GEMV_CASE(8); break;
  }
#undef GEMV_CASE
  HIP_CHECK_LAST();
}
