// Fused temperature-softmax + nucleus (top-p) sampling for MI355X.
//
// Sort-free design: sorting a 128k-entry vocab per row is the classic
// CUDA approach; on MI355X we instead bisect a logit threshold t such
// that the probability mass of {i : logit_i >= t} reaches top_p (~24
// fixed iterations), then draw from the kept set by a two-level
// prefix-sum walk. Every pass streams the row out of L2 (a 128k-fp32 row
// is 512 KB, far under the 4 MiB per-XCD L2), so the whole sampler is a
// few L2-bandwidth passes — no global sort, no scratch allocation.
//
// Each row is one 256-thread workgroup; thread t owns the contiguous
// range [t*V/256, (t+1)*V/256) in every pass. RNG: per-row xorshift from
// a seed (deterministic given seed).
//
// Semantics match dts_amd/ops/torch_ref.py top_p_sample (temperature<=0
// => greedy; top-p mass cut with at-least-one-token guarantee).

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

DEV unsigned long long xorshift64(unsigned long long x) {
  x ^= x << 13;
  x ^= x >> 7;
  x ^= x << 17;
  return x;
}

template <int BLOCK>
__global__ void __launch_bounds__(BLOCK)
top_p_sample_kernel(long* __restrict__ out, const float* __restrict__ logits,
                    const float* __restrict__ temps,
                    const float* __restrict__ top_ps,
                    const long* __restrict__ seeds, int V) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const float* x = logits + (long)row * V;
  const float T = temps[row];
  const float p_target = top_ps[row];

  __shared__ float red[BLOCK / WAVE];
  __shared__ float s_bcast[2];
  __shared__ int s_argmax;

  const int per = (V + BLOCK - 1) / BLOCK;
  const int lo_i = tid * per;
  const int hi_i = min(V, lo_i + per);

  // ---- pass 1: max (and argmax for greedy)
  float mymax = -1e30f;
  int myarg = 0;
  for (int i = lo_i; i < hi_i; ++i) {
    float v = x[i];
    if (v > mymax) {
      mymax = v;
      myarg = i;
    }
  }
  // block reduce max
  {
    float wm = mymax;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float o = __shfl_xor(wm, off, 64);
      wm = fmaxf(wm, o);
    }
    if ((tid & 63) == 0) red[tid / 64] = wm;
    __syncthreads();
    float bm = -1e30f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) bm = fmaxf(bm, red[i]);
    // greedy argmax via ballot on equality (first match wins)
    if (T <= 0.f) {
      if (tid == 0) s_argmax = -1;
      __syncthreads();
      if (mymax == bm) atomicCAS(&s_argmax, -1, myarg);
      __syncthreads();
      if (tid == 0) out[row] = s_argmax;
      return;
    }
    if (tid == 0) s_bcast[0] = bm;
    __syncthreads();
  }
  const float m = s_bcast[0];
  const float invT = 1.f / T;

  // ---- pass 2: Z = sum exp((x-m)/T)
  float zpart = 0.f;
  for (int i = lo_i; i < hi_i; ++i) zpart = zpart + __expf((x[i] - m) * invT);
  {
    float z = wave_sum(zpart);
    if ((tid & 63) == 0) red[tid / 64] = z;
    __syncthreads();
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) tot += red[i];
    if (tid == 0) s_bcast[1] = tot;
    __syncthreads();
  }
  const float Z = s_bcast[1];
  const float target_mass = fminf(p_target, 1.0f) * Z;

  // ---- bisect threshold tau on y = (x-m)/T in [-30, 0]
  float tau = -30.f;
  if (p_target < 1.0f) {
    float lo = -30.f, hi = 0.f;
    for (int it = 0; it < 24; ++it) {
      float mid = 0.5f * (lo + hi);
      float kept = 0.f;
      for (int i = lo_i; i < hi_i; ++i) {
        float y = (x[i] - m) * invT;
        if (y >= mid) kept += __expf(y);
      }
      kept = wave_sum(kept);
      if ((tid & 63) == 0) red[tid / 64] = kept;
      __syncthreads();
      float tot = 0.f;
#pragma unroll
      for (int i = 0; i < BLOCK / WAVE; ++i) tot += red[i];
      __syncthreads();
      if (tot >= target_mass) lo = mid;  // keep raising the floor
      else hi = mid;
    }
    tau = lo;  // mass(tau=lo) >= target (contains at least the max, y=0)
  }

  // ---- kept mass per thread + prefix over threads
  __shared__ float s_pref[BLOCK + 1];
  float mymass = 0.f;
  for (int i = lo_i; i < hi_i; ++i) {
    float y = (x[i] - m) * invT;
    if (y >= tau) mymass += __expf(y);
  }
  s_pref[tid + 1] = mymass;
  __syncthreads();
  if (tid == 0) {
    s_pref[0] = 0.f;
    for (int i = 1; i <= BLOCK; ++i) s_pref[i] += s_pref[i - 1];
  }
  __syncthreads();
  const float total_kept = s_pref[BLOCK];

  // ---- draw u in [0, total_kept) and locate the owning thread/token
  unsigned long long rng = xorshift64((unsigned long long)seeds[row] * 2685821657736338717ULL + 1);
  rng = xorshift64(rng);
  const float u = (float)((rng >> 11) * (1.0 / 9007199254740992.0)) * total_kept;

  __shared__ long s_result;
  if (tid == 0) s_result = -1;
  __syncthreads();
  if (u >= s_pref[tid] && u < s_pref[tid + 1]) {
    float acc = s_pref[tid];
    long pick = -1;
    for (int i = lo_i; i < hi_i; ++i) {
      float y = (x[i] - m) * invT;
      if (y >= tau) {
        acc += __expf(y);
        if (acc > u) {
          pick = i;
          break;
        }
      }
    }
    if (pick < 0) {  // numeric edge: last kept in range
      for (int i = hi_i - 1; i >= lo_i; --i) {
        float y = (x[i] - m) * invT;
        if (y >= tau) {
          pick = i;
          break;
        }
      }
    }
    s_result = pick;
  }
  __syncthreads();
  if (tid == 0) {
    long r = s_result;
    if (r < 0) {
      // degenerate: nothing kept (shouldn't happen — tau <= 0 keeps max);
      // fall back to greedy via a serial scan
      float bm = -1e30f;
      for (int i = 0; i < V; ++i)
        if (x[i] > bm) {
          bm = x[i];
          r = i;
        }
    }
    out[row] = r;
  }
}

void top_p_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temps,
                  torch::Tensor top_ps, torch::Tensor seeds) {
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32 && logits.is_contiguous());
  const int S = logits.size(0), V = logits.size(1);
  hipLaunchKernelGGL((top_p_sample_kernel<256>), dim3(S), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(), (long*)out.data_ptr(),
                     (const float*)logits.data_ptr(),
                     (const float*)temps.data_ptr(),
                     (const float*)top_ps.data_ptr(),
                     (const long*)seeds.data_ptr(), V);
  HIP_CHECK_LAST();
}
