// Skinny-M MFMA GEMM for decode projections on MI355X (gfx950).
//
// out[M, N] = x[M, K] @ W[N, K]^T with M = decode batch width (3..16).
// The round-1 VALU GEMV (gemv.hip) hits the HBM roofline at M<=2 but
// turns ALU-bound past M≈8 (RPW*M*8 scalar FMAs per 64 B of W), and
// hipBLASLt runs ~50% of the roofline at these M — so the bench's B≈6
// decode steps were paying ~2x on every weight read. GEMM-shaped work
// belongs on the matrix cores: this kernel streams W once through
// v_mfma_f32_16x16x32_bf16 tiles, which makes the arithmetic free and
// leaves pure weight streaming.
//
// Structure: one 16-row N-tile per 4-wave workgroup. Each wave owns a
// quarter of K (split-K inside the workgroup, no global atomics); per
// k-chunk of 32:
//   A-frag: lane (row = lane&15 -> x row m, k = k0 + (lane>>4)*8 + i)
//           — 16 B contiguous per lane; rows m >= M are zero.
//   B-frag: lane (col = lane&15 -> W row n0+(lane&15), same k split)
//           — 16 B contiguous per lane; 4 lanes stride-16 cover a full
//           64 B cacheline of each W row.
//   acc = mfma(a, b, acc)  (C/D: row = (lane>>4)*4 + i, col = lane&15)
// Epilogue: the 4 waves' C tiles reduce through LDS; wave 0 writes bf16.
// fp32 accumulation, bf16 I/O — same numeric class as hipBLASLt.

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define NWAVES 4  // waves per workgroup = K split factor
#define KSTEP 32  // one mfma k-chunk
#define KUNROLL 4  // k-chunks in flight per iteration

template <int KU>
__global__ void __launch_bounds__(NWAVES* WAVE)
gemm_skinny_kernel(short* __restrict__ out,      // [M, N] (row stride out_ts)
                   const short* __restrict__ x,  // [M, K] (row stride x_ts)
                   const short* __restrict__ w,  // [N, K]
                   int M, int N, int K, long x_ts, long out_ts) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n0 = blockIdx.x * 16;

  // this wave's K quarter (K % (NWAVES*KSTEP*KUNROLL) checked host-side)
  const int kq = K / NWAVES;
  const int kbeg = wave * kq;
  const int kend = kbeg + kq;

  const int a_row = lane & 15;          // x row (m)
  const int k_off = (lane >> 4) * 8;    // this lane's k sub-offset
  const int b_row = n0 + (lane & 15);   // W row (n)
  const bool a_live = a_row < M;
  const bool b_live = b_row < N;
  const short* xrow = x + (long)a_row * x_ts;
  const short* wrow = w + (long)b_row * K;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const bf16x8 zero8 = {0, 0, 0, 0, 0, 0, 0, 0};

  for (int k0 = kbeg; k0 < kend; k0 += KSTEP * KU) {
    bf16x8 a[KU], b[KU];
#pragma unroll
    for (int u = 0; u < KU; ++u) {
      const int kk = k0 + u * KSTEP + k_off;
      a[u] = a_live ? *(const bf16x8*)(xrow + kk) : zero8;
      b[u] = b_live ? *(const bf16x8*)(wrow + kk) : zero8;
    }
#pragma unroll
    for (int u = 0; u < KU; ++u)
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u], acc, 0, 0, 0);
  }

  // cross-wave reduction: C tiles are [16 m x 16 n] fp32, 1 KB per wave
  __shared__ float red[NWAVES][16][16];
  const int c_col = lane & 15;
#pragma unroll
  for (int i = 0; i < 4; ++i) red[wave][(lane >> 4) * 4 + i][c_col] = acc[i];
  __syncthreads();
  if (wave == 0) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int m = (lane >> 4) * 4 + i;
      const int n = n0 + c_col;
      if (m < M && n < N) {
        float v = red[0][m][c_col] + red[1][m][c_col] + red[2][m][c_col] +
                  red[3][m][c_col];
        out[(long)m * out_ts + n] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Grouped skinny GEMM for MoE decode (capture-safe expert dispatch).
//
// out[p, :] = x[p, :] @ W[e, :, :]^T for rows p in expert e's contiguous
// segment [offsets[e], offsets[e]+counts[e]). Counts/offsets are DEVICE
// tensors — no host-side shapes depend on routing, so a Mixtral decode
// step is hipGraph-capturable (the round-1 masked-gather path called
// nonzero() per expert per layer: a stream sync that aborted capture).
// Each (n-tile, expert) block loops the segment in 16-row chunks; decode
// segments are <= tokens*top_k (<= 32), so expert weights stream at most
// twice.
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(NWAVES* WAVE)
moe_grouped_kernel(short* __restrict__ out,      // [P, N]
                   const short* __restrict__ x,  // [P, K] dense
                   const short* __restrict__ w,  // [E, N, K]
                   const int* __restrict__ counts,   // [E]
                   const int* __restrict__ offsets,  // [E]
                   int N, int K) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n0 = blockIdx.x * 16;
  const int e = blockIdx.y;
  const int cnt = counts[e];
  if (cnt == 0) return;
  const int off = offsets[e];
  const short* we = w + (long)e * N * K;

  const int kq = K / NWAVES;
  const int kbeg = wave * kq;
  const int kend = kbeg + kq;
  const int a_row = lane & 15;
  const int k_off = (lane >> 4) * 8;
  const int b_row = n0 + (lane & 15);
  const bool b_live = b_row < N;
  const short* wrow = we + (long)b_row * K;
  const bf16x8 zero8 = {0, 0, 0, 0, 0, 0, 0, 0};

  __shared__ float red[NWAVES][16][16];
  const int c_col = lane & 15;

  for (int mb = 0; mb < cnt; mb += 16) {
    const bool a_live = mb + a_row < cnt;
    const short* xrow = x + (long)(off + mb + a_row) * K;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = kbeg; k0 < kend; k0 += KSTEP * KUNROLL) {
      bf16x8 a[KUNROLL], b[KUNROLL];
#pragma unroll
      for (int u = 0; u < KUNROLL; ++u) {
        const int kk = k0 + u * KSTEP + k_off;
        a[u] = a_live ? *(const bf16x8*)(xrow + kk) : zero8;
        b[u] = b_live ? *(const bf16x8*)(wrow + kk) : zero8;
      }
#pragma unroll
      for (int u = 0; u < KUNROLL; ++u)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u], acc, 0, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) red[wave][(lane >> 4) * 4 + i][c_col] = acc[i];
    __syncthreads();
    if (wave == 0) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = (lane >> 4) * 4 + i;
        const int n = n0 + c_col;
        if (mb + m < cnt && n < N) {
          float v = red[0][m][c_col] + red[1][m][c_col] + red[2][m][c_col] +
                    red[3][m][c_col];
          out[(long)(off + mb + m) * N + n] = f2bf(v);
        }
      }
    }
    __syncthreads();
  }
}

void moe_grouped_linear(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                        torch::Tensor counts, torch::Tensor offsets) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(w.is_contiguous(), "expert weights must be contiguous [E,N,K]");
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(counts.scalar_type() == torch::kInt32 &&
              offsets.scalar_type() == torch::kInt32);
  const int K = x.size(1), E = w.size(0), N = w.size(1);
  TORCH_CHECK(K % (NWAVES * KSTEP * KUNROLL) == 0, "K must be /512");
  const int tiles = (N + 15) / 16;
  hipLaunchKernelGGL(moe_grouped_kernel, dim3(tiles, E), dim3(NWAVES * WAVE),
                     0, c10::hip::getCurrentHIPStream(),
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(),
                     (const int*)counts.data_ptr(),
                     (const int*)offsets.data_ptr(), N, K);
  HIP_CHECK_LAST();
}

void gemm_skinny_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.is_contiguous(), "weight must be contiguous [N, K]");
  TORCH_CHECK(x.stride(1) == 1, "x rows must be dense");
  TORCH_CHECK(out.stride(1) == 1);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M >= 1 && M <= 16, "skinny path is for M<=16");
  TORCH_CHECK(K % (NWAVES * KSTEP * KUNROLL) == 0, "K must be /512");
  const int tiles = (N + 15) / 16;
  // deeper unroll keeps 128 B/operand/lane in flight (A/B via env)
  static const int ku = [] {
    const char* e = getenv("DTS_SKINNY_UNROLL");
    return (e && e[0] == '8') ? 8 : 4;
  }();
  auto stream = c10::hip::getCurrentHIPStream();
  if (ku == 8 && K % (NWAVES * KSTEP * 8) == 0)
    hipLaunchKernelGGL((gemm_skinny_kernel<8>), dim3(tiles),
                       dim3(NWAVES * WAVE), 0, stream, (short*)out.data_ptr(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       M, N, K, x.stride(0), out.stride(0));
  else
    hipLaunchKernelGGL((gemm_skinny_kernel<4>), dim3(tiles),
                       dim3(NWAVES * WAVE), 0, stream, (short*)out.data_ptr(),
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       M, N, K, x.stride(0), out.stride(0));
  HIP_CHECK_LAST();
}
