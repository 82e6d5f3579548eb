// Fused temperature-softmax + nucleus (top-p) sampling for MI355X.
//
// Sort-free, histogram-thresholded design: instead of sorting 128k logits
// per row (the classic CUDA approach) or bisecting a threshold with ~24
// full passes, one pass builds a 1024-bin LDS histogram of the softmax
// masses over y = (logit - max)/T ∈ [-30, 0]; a suffix scan of the bins
// picks the threshold bin whose cumulative mass reaches top_p, and one
// final pass draws the token from the kept set via a two-level prefix
// walk. 3 full passes total (max; histogram+Z; select) — each streams the
// row out of L2 (a 128k-fp32 row is 512 KB ≪ the 4 MiB per-XCD L2).
//
// The kept-mass overshoot from taking a whole boundary bin is ≤ the bin
// mass (bin width 0.03 in y), i.e. the cut differs from an exact top-p by
// at most a handful of borderline tokens — the same class of tie-handling
// slack that sorting implementations have at equal probabilities.
//
// Each row is one 256-thread workgroup; thread t owns the STRIDED element
// set {t, t+256, ...} in every pass — contiguous per-thread ranges made
// every lane touch a different cache line per 4-byte read (measured 16x
// bandwidth waste, ~436 us/row). The final draw walks the owner thread's
// strided set in stride-order: a fixed permutation of the kept set, which
// leaves each token's selection probability unchanged. RNG: per-row
// xorshift from a seed (deterministic given seed).
//
// Semantics match dts_amd/ops/torch_ref.py top_p_sample (temperature<=0
// => greedy; at-least-one-token guarantee).

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define NBINS 1024
#define YMIN (-30.0f)

DEV unsigned long long xorshift64(unsigned long long x) {
  x ^= x << 13;
  x ^= x >> 7;
  x ^= x << 17;
  return x;
}

// splitmix64-style stateless seed mix over (base, position): the SAME
// formula runs on the host (dts_amd/ops/torch_ref.py mix_seed) so the
// chained-decode path (seeds derived on device from static positions)
// draws the exact token the per-step host path would.
DEV unsigned long long mix_seed64(unsigned long long base,
                                  unsigned long long pos) {
  unsigned long long x = base ^ (pos * 0x9E3779B97F4A7C15ull);
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  x ^= x >> 31;
  return x & 0x7FFFFFFFull;
}

// out[i] = mix(base[i], positions[i] + 1): positions are QUERY positions
// of the decode rows; the drawn token's index is one past the query.
__global__ void derive_seeds_kernel(long* __restrict__ out,
                                    const long* __restrict__ bases,
                                    const long* __restrict__ positions,
                                    int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n)
    out[i] = (long)mix_seed64((unsigned long long)bases[i],
                              (unsigned long long)(positions[i] + 1));
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
  __shared__ float s_bins[NBINS];

  // ---- pass 1: max (and argmax for greedy)
  float mymax = -1e30f;
  int myarg = 0;
  for (int i = tid; i < V; i += BLOCK) {
    float v = x[i];
    if (v > mymax) {
      mymax = v;
      myarg = i;
    }
  }
  {
    float wm = mymax;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      wm = fmaxf(wm, __shfl_xor(wm, off, 64));
    if ((tid & 63) == 0) red[tid / 64] = wm;
    __syncthreads();
    float bm = -1e30f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) bm = fmaxf(bm, red[i]);
    if (T <= 0.f) {
      // deterministic tie-break: LOWEST index among equal maxima (the
      // torch reference argmax does the same; an atomicCAS race here
      // made chained-vs-stepped greedy streams diverge on real ties)
      if (tid == 0) s_argmax = 0x7FFFFFFF;
      __syncthreads();
      if (mymax == bm) atomicMin(&s_argmax, myarg);
      __syncthreads();
      if (tid == 0) out[row] = s_argmax;
      return;
    }
    if (tid == 0) s_bcast[0] = bm;
    __syncthreads();
  }
  const float m = s_bcast[0];
  const float invT = 1.f / T;

  // ---- pass 2: mass histogram over y + total Z
  for (int b = tid; b < NBINS; b += BLOCK) s_bins[b] = 0.f;
  __syncthreads();
  float zpart = 0.f;
  const float bin_scale = NBINS / (-YMIN);  // bins per unit y
  for (int i = tid; i < V; i += BLOCK) {
    float y = (x[i] - m) * invT;
    float e = __expf(y);
    zpart += e;
    if (y > YMIN) {
      int b = (int)fminf((y - YMIN) * bin_scale, (float)(NBINS - 1));
      atomicAdd(&s_bins[b], e);
    }
  }
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

  // ---- threshold: suffix-scan the bins from the top until >= p*Z
  // (serial over 1024 bins on thread 0 — trivial vs a vocab pass)
  if (tid == 0) {
    const float target = fminf(p_target, 1.0f) * Z;
    float acc = 0.f;
    int b = NBINS - 1;
    for (; b >= 0; --b) {
      acc += s_bins[b];
      if (acc >= target) break;
    }
    if (b < 0) b = 0;
    s_bcast[0] = YMIN + b / bin_scale;  // tau = lower edge of boundary bin
  }
  __syncthreads();
  const float tau = s_bcast[0];

  // ---- kept mass per thread + prefix over threads
  __shared__ float s_pref[BLOCK + 1];
  float mymass = 0.f;
  for (int i = tid; i < V; i += BLOCK) {
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

  // ---- draw u and locate the owning thread/token
  unsigned long long rng =
      xorshift64((unsigned long long)seeds[row] * 2685821657736338717ULL + 1);
  rng = xorshift64(rng);
  const float u =
      (float)((rng >> 11) * (1.0 / 9007199254740992.0)) * total_kept;

  __shared__ long s_result;
  if (tid == 0) s_result = -1;
  __syncthreads();
  if (total_kept > 0.f && u >= s_pref[tid] && u < s_pref[tid + 1]) {
    float acc = s_pref[tid];
    long pick = -1, last_kept = -1;
    for (int i = tid; i < V; i += BLOCK) {
      float y = (x[i] - m) * invT;
      if (y >= tau) {
        last_kept = i;
        acc += __expf(y);
        if (acc > u) {
          pick = i;
          break;
        }
      }
    }
    s_result = (pick >= 0) ? pick : last_kept;
  }
  __syncthreads();
  if (tid == 0) {
    long r = s_result;
    if (r < 0) {
      // degenerate: nothing kept — greedy fallback via serial scan
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

// ---------------------------------------------------------------------------
// Gridded sampler (v2): the one-workgroup-per-row design above leaves
// 250 of 256 CUs idle at decode batch widths (B=1..8) — measured 364 us
// avg on the r2 bench (4.9% of GPU busy). v2 splits each row's vocab
// across SLICES workgroups and runs five tiny gridded kernels:
//   s2_max   : per-slice max -> packed 64-bit atomicMax (value bits
//              high, ~index low => deterministic LOWEST-index tie-break)
//   s2_hist  : per-slice LDS histogram of exp masses over
//              y=(x-max)/T in [-30,0] -> atomicAdd into global bins
//   s2_tau   : per-row 1024-bin suffix scan -> threshold tau
//   s2_kept  : per-slice EXACT kept mass (y >= tau)
//   s2_draw  : per-row: prefix over slice masses, draw u, walk the one
//              owning slice with a two-level prefix to pick the token
// Same bins / y-space / boundary-bin slack as v1; the draw walks in
// slice-index order (a different fixed permutation of the kept set than
// v1's strided order — same distribution, different exact picks for a
// given seed). Greedy (T<=0) short-circuits after s2_max.
// ---------------------------------------------------------------------------

#define S2_SLICES 16

DEV unsigned long long pack_max(float v, int idx) {
  union { float f; unsigned int i; } c;
  c.f = v;
  // order-preserving float->uint (sign flip trick)
  unsigned int bits = (c.i & 0x80000000u) ? ~c.i : (c.i | 0x80000000u);
  return ((unsigned long long)bits << 32) | (unsigned int)(0x7FFFFFFF - idx);
}

__global__ void __launch_bounds__(256)
s2_max_kernel(unsigned long long* __restrict__ gmax,
              const float* __restrict__ logits, int V) {
  const int row = blockIdx.x, slice = blockIdx.y, tid = threadIdx.x;
  const float* x = logits + (long)row * V;
  const int lo = slice * V / S2_SLICES, hi = (slice + 1) * V / S2_SLICES;
  float m = -1e30f;
  int arg = lo;
  for (int i = lo + tid; i < hi; i += 256) {
    if (x[i] > m) { m = x[i]; arg = i; }
  }
  // wave+wg reduce via LDS on packed values (max keeps lowest index)
  __shared__ unsigned long long red[4];
  unsigned long long p = pack_max(m, arg);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    unsigned long long o = __shfl_xor((long long)p, off, 64);
    if (o > p) p = o;
  }
  if ((tid & 63) == 0) red[tid / 64] = p;
  __syncthreads();
  if (tid == 0) {
    for (int i = 1; i < 4; ++i) if (red[i] > p) p = red[i];
    atomicMax(gmax + row, p);
  }
}

__global__ void __launch_bounds__(256)
s2_hist_kernel(float* __restrict__ gbins,  // [S, NBINS]
               const unsigned long long* __restrict__ gmax,
               const float* __restrict__ logits,
               const float* __restrict__ temps, int V) {
  const int row = blockIdx.x, slice = blockIdx.y, tid = threadIdx.x;
  const float T = temps[row];
  if (T <= 0.f) return;  // greedy: resolved from gmax alone
  union { unsigned int i; float f; } c;
  unsigned int bits = (unsigned int)(gmax[row] >> 32);
  c.i = (bits & 0x80000000u) ? (bits & 0x7FFFFFFFu) : ~bits;
  const float m = c.f;
  const float invT = 1.f / T;
  const float* x = logits + (long)row * V;
  const int lo = slice * V / S2_SLICES, hi = (slice + 1) * V / S2_SLICES;
  __shared__ float s_bins[NBINS];
  for (int b = tid; b < NBINS; b += 256) s_bins[b] = 0.f;
  __syncthreads();
  const float bin_scale = NBINS / (-YMIN);
  for (int i = lo + tid; i < hi; i += 256) {
    float y = (x[i] - m) * invT;
    if (y > YMIN) {
      int b = (int)fminf((y - YMIN) * bin_scale, (float)(NBINS - 1));
      atomicAdd(&s_bins[b], __expf(y));
    }
  }
  __syncthreads();
  float* dst = gbins + (long)row * NBINS;
  for (int b = tid; b < NBINS; b += 256)
    if (s_bins[b] != 0.f) atomicAdd(dst + b, s_bins[b]);
}

__global__ void __launch_bounds__(256)
s2_tau_kernel(float* __restrict__ gtau, long* __restrict__ out,
              const float* __restrict__ gbins,
              const unsigned long long* __restrict__ gmax,
              const float* __restrict__ temps,
              const float* __restrict__ top_ps) {
  const int row = blockIdx.x;
  if (temps[row] <= 0.f) {
    if (threadIdx.x == 0)
      out[row] = 0x7FFFFFFF - (int)(gmax[row] & 0xFFFFFFFFull);
    return;
  }
  if (threadIdx.x != 0) return;
  const float* bins = gbins + (long)row * NBINS;
  float Z = 0.f;
  for (int b = 0; b < NBINS; ++b) Z += bins[b];
  const float target = fminf(top_ps[row], 1.0f) * Z;
  float acc = 0.f;
  int b = NBINS - 1;
  for (; b >= 0; --b) {
    acc += bins[b];
    if (acc >= target) break;
  }
  if (b < 0) b = 0;
  gtau[row] = YMIN + b * (-YMIN) / NBINS;
}

__global__ void __launch_bounds__(256)
s2_kept_kernel(float* __restrict__ gslice,  // [S, S2_SLICES]
               const float* __restrict__ gtau,
               const unsigned long long* __restrict__ gmax,
               const float* __restrict__ logits,
               const float* __restrict__ temps, int V) {
  const int row = blockIdx.x, slice = blockIdx.y, tid = threadIdx.x;
  const float T = temps[row];
  if (T <= 0.f) return;
  union { unsigned int i; float f; } c;
  unsigned int bits = (unsigned int)(gmax[row] >> 32);
  c.i = (bits & 0x80000000u) ? (bits & 0x7FFFFFFFu) : ~bits;
  const float m = c.f;
  const float invT = 1.f / T, tau = gtau[row];
  const float* x = logits + (long)row * V;
  const int lo = slice * V / S2_SLICES, hi = (slice + 1) * V / S2_SLICES;
  float mass = 0.f;
  for (int i = lo + tid; i < hi; i += 256) {
    float y = (x[i] - m) * invT;
    if (y >= tau) mass += __expf(y);
  }
  mass = wave_sum(mass);
  __shared__ float red[4];
  if ((tid & 63) == 0) red[tid / 64] = mass;
  __syncthreads();
  if (tid == 0)
    gslice[(long)row * S2_SLICES + slice] = red[0] + red[1] + red[2] + red[3];
}

__global__ void __launch_bounds__(256)
s2_draw_kernel(long* __restrict__ out, const float* __restrict__ gslice,
               const float* __restrict__ gtau,
               const unsigned long long* __restrict__ gmax,
               const float* __restrict__ logits,
               const float* __restrict__ temps,
               const long* __restrict__ seeds, int V) {
  const int row = blockIdx.x, tid = threadIdx.x;
  const float T = temps[row];
  if (T <= 0.f) return;  // already written by s2_tau
  union { unsigned int i; float f; } c;
  unsigned int bits = (unsigned int)(gmax[row] >> 32);
  c.i = (bits & 0x80000000u) ? (bits & 0x7FFFFFFFu) : ~bits;
  const float m = c.f;
  const float invT = 1.f / T, tau = gtau[row];
  const float* sl = gslice + (long)row * S2_SLICES;
  __shared__ float s_pref_sl[S2_SLICES + 1];
  __shared__ int s_slice;
  __shared__ float s_u;
  if (tid == 0) {
    s_pref_sl[0] = 0.f;
    for (int i = 0; i < S2_SLICES; ++i) s_pref_sl[i + 1] = s_pref_sl[i] + sl[i];
    const float total = s_pref_sl[S2_SLICES];
    unsigned long long rng =
        xorshift64((unsigned long long)seeds[row] * 2685821657736338717ULL + 1);
    rng = xorshift64(rng);
    float u = (float)((rng >> 11) * (1.0 / 9007199254740992.0)) * total;
    int s = 0;
    while (s < S2_SLICES - 1 && u >= s_pref_sl[s + 1]) ++s;
    s_slice = s;
    s_u = u - s_pref_sl[s];
  }
  __syncthreads();
  const int slice = s_slice;
  const float u = s_u;
  const float* x = logits + (long)row * V;
  const int lo = slice * V / S2_SLICES, hi = (slice + 1) * V / S2_SLICES;
  // two-level prefix inside the slice: thread t owns [lo + t*chunk ...)
  const int n = hi - lo;
  const int chunk = (n + 255) / 256;
  const int my_lo = lo + tid * chunk;
  const int my_hi = min(hi, my_lo + chunk);
  float mymass = 0.f;
  for (int i = my_lo; i < my_hi; ++i) {
    float y = (x[i] - m) * invT;
    if (y >= tau) mymass += __expf(y);
  }
  __shared__ float s_pref[257];
  s_pref[tid + 1] = mymass;
  __syncthreads();
  if (tid == 0) {
    s_pref[0] = 0.f;
    for (int i = 1; i <= 256; ++i) s_pref[i] += s_pref[i - 1];
  }
  __syncthreads();
  __shared__ long s_result;
  if (tid == 0) s_result = -1;
  __syncthreads();
  if (u >= s_pref[tid] && u < s_pref[tid + 1]) {
    float acc = s_pref[tid];
    long pick = -1, last_kept = -1;
    for (int i = my_lo; i < my_hi; ++i) {
      float y = (x[i] - m) * invT;
      if (y >= tau) {
        last_kept = i;
        acc += __expf(y);
        if (acc > u) { pick = i; break; }
      }
    }
    s_result = (pick >= 0) ? pick : last_kept;
  }
  __syncthreads();
  if (tid == 0) {
    long r = s_result;
    if (r < 0) {
      // degenerate (fp slack at the edge): greedy fallback from gmax
      r = 0x7FFFFFFF - (int)(gmax[row] & 0xFFFFFFFFull);
    }
    out[row] = r;
  }
}

void top_p_sample_v2(torch::Tensor out, torch::Tensor logits,
                     torch::Tensor temps, torch::Tensor top_ps,
                     torch::Tensor seeds, torch::Tensor ws_max,
                     torch::Tensor ws_bins, torch::Tensor ws_slice,
                     torch::Tensor ws_tau) {
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32 && logits.is_contiguous());
  const int S = logits.size(0), V = logits.size(1);
  auto stream = c10::hip::getCurrentHIPStream();
  hipMemsetAsync(ws_max.data_ptr(), 0, S * 8, stream);
  hipMemsetAsync(ws_bins.data_ptr(), 0, (long)S * NBINS * 4, stream);
  dim3 gs(S, S2_SLICES);
  hipLaunchKernelGGL(s2_max_kernel, gs, dim3(256), 0, stream,
                     (unsigned long long*)ws_max.data_ptr(),
                     (const float*)logits.data_ptr(), V);
  hipLaunchKernelGGL(s2_hist_kernel, gs, dim3(256), 0, stream,
                     (float*)ws_bins.data_ptr(),
                     (const unsigned long long*)ws_max.data_ptr(),
                     (const float*)logits.data_ptr(),
                     (const float*)temps.data_ptr(), V);
  hipLaunchKernelGGL(s2_tau_kernel, dim3(S), dim3(256), 0, stream,
                     (float*)ws_tau.data_ptr(), (long*)out.data_ptr(),
                     (const float*)ws_bins.data_ptr(),
                     (const unsigned long long*)ws_max.data_ptr(),
                     (const float*)temps.data_ptr(),
                     (const float*)top_ps.data_ptr());
  hipLaunchKernelGGL(s2_kept_kernel, gs, dim3(256), 0, stream,
                     (float*)ws_slice.data_ptr(),
                     (const float*)ws_tau.data_ptr(),
                     (const unsigned long long*)ws_max.data_ptr(),
                     (const float*)logits.data_ptr(),
                     (const float*)temps.data_ptr(), V);
  hipLaunchKernelGGL(s2_draw_kernel, dim3(S), dim3(256), 0, stream,
                     (long*)out.data_ptr(), (const float*)ws_slice.data_ptr(),
                     (const float*)ws_tau.data_ptr(),
                     (const unsigned long long*)ws_max.data_ptr(),
                     (const float*)logits.data_ptr(),
                     (const float*)temps.data_ptr(),
                     (const long*)seeds.data_ptr(), V);
  HIP_CHECK_LAST();
}

void derive_seeds(torch::Tensor out, torch::Tensor bases,
                  torch::Tensor positions) {
  const int n = out.size(0);
  hipLaunchKernelGGL(derive_seeds_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(), (long*)out.data_ptr(),
                     (const long*)bases.data_ptr(),
                     (const long*)positions.data_ptr(), n);
  HIP_CHECK_LAST();
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
