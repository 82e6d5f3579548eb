// Skinny-batch (M <= 8) bf16 GEMV/GEMM for decode projections on MI355X.
//
// The decode step's linears are out[M, N] = x[M, K] @ W[N, K]^T with
// M = decode batch width (1-8 on the search workload) — pure
// weight-streaming: W's N*K*2 bytes dominate. hipBLASLt at these M runs
// ~50% of the HBM roofline on gfx950 (measured, profiles/); this kernel
// follows the guide's GEMV row ("operand streamed once per block, not
// shared across waves: load straight to VGPRs, deep unroll, late vmcnt"):
//
//   one WAVE per output row n: lane l streams W[n, 8l :: 512] as bf16x8
//   (16 B/lane, 1 KiB per wave-instruction, fully coalesced), multiplies
//   into M fp32 accumulators against x[m] fragments re-read from L2
//   (x is tiny and every wave reads the same lines), then one 64-lane
//   shuffle reduction per m. 4 waves per WG, grid-strided over N rows;
//   XCD-aware row swizzle keeps neighbor rows' x reads in one L2.
//
// fp32 accumulation, bf16 I/O — numerically the same class as the
// hipBLASLt path it replaces.

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

using bf16x8 = __attribute__((ext_vector_type(8))) short;

template <int M>
__global__ void __launch_bounds__(256)
gemv_bf16_kernel(short* __restrict__ out,      // [M, N]
                 const short* __restrict__ x,  // [M, K] (row stride x_ts)
                 const short* __restrict__ w,  // [N, K]
                 int N, int K, long x_tstride, long out_tstride) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int waves_per_wg = blockDim.x / WAVE;
  const int rows_per_grid = gridDim.x * waves_per_wg;

  for (int n = blockIdx.x * waves_per_wg + wave; n < N; n += rows_per_grid) {
    const short* wrow = w + (long)n * K;
    float acc[M];
#pragma unroll
    for (int m = 0; m < M; ++m) acc[m] = 0.f;

    // lane l covers k = 8l, 8l+512, ... ; 16 B per load, coalesced
    for (int k0 = lane * 8; k0 < K; k0 += WAVE * 8) {
      bf16x8 wv = *(const bf16x8*)(wrow + k0);
      float wf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) wf[j] = bf2f(wv[j]);
#pragma unroll
      for (int m = 0; m < M; ++m) {
        bf16x8 xv = *(const bf16x8*)(x + (long)m * x_tstride + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[m] += bf2f(xv[j]) * wf[j];
      }
    }
#pragma unroll
    for (int m = 0; m < M; ++m) {
      float v = wave_sum(acc[m]);
      if (lane == 0) out[(long)m * out_tstride + n] = f2bf(v);
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
  // enough waves to cover the chip several times over; grid-stride the rest
  int wgs = std::min((N + 3) / 4, 2048);
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
