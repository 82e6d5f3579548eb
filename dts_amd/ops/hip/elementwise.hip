// Elementwise / normalization kernels for MI355X (gfx950).
//
// All memory-bound: the design targets the HBM roofline via 16-B-per-lane
// vectorized bf16 access (8x bf16 per load — guide G13: scalar bf16 loads
// cost ~2-2.5x) and fp32 accumulation. One 256-thread workgroup per token
// row; grid-stride over rows.
//
// Semantics match dts_amd/ops/torch_ref.py (rmsnorm, fused_add_rmsnorm,
// silu_mul) — numerics tests compare against that fp32 reference.

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

// ---------------------------------------------------------------------------
// RMSNorm: y = x * rsqrt(mean(x^2) + eps) * w       x: [T, H] bf16
// ---------------------------------------------------------------------------

template <int BLOCK>
__global__ void rmsnorm_kernel(short* __restrict__ out,
                               const short* __restrict__ x,
                               const short* __restrict__ w, float eps, int T,
                               int H) {
  __shared__ float red[BLOCK / WAVE];
  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    short* yr = out + (long)row * H;
    float ss = 0.f;
    // 8 bf16 per thread per iter
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v v = *(const short8v*)(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v[j]);
        ss += f * f;
      }
    }
    ss = wave_sum(ss);
    if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = ss;
    __syncthreads();
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) tot += red[i];
    float inv = rsqrtf(tot / H + eps);
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v v = *(const short8v*)(xr + i);
      short8v wv = *(const short8v*)(w + i);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(v[j]) * inv * bf2f(wv[j]));
      *(short8v*)(yr + i) = o;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Fused residual-add + RMSNorm (in place):
//   residual += x;  x = rmsnorm(residual) * w
// Saves one full HBM round trip of the residual stream per layer.
// ---------------------------------------------------------------------------

template <int BLOCK>
__global__ void fused_add_rmsnorm_kernel(short* __restrict__ x,
                                         short* __restrict__ residual,
                                         const short* __restrict__ w,
                                         float eps, int T, int H) {
  __shared__ float red[BLOCK / WAVE];
  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    short* xr = x + (long)row * H;
    short* rr = residual + (long)row * H;
    float ss = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v xv = *(const short8v*)(xr + i);
      short8v rv = *(const short8v*)(rr + i);
      short8v nv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float s = bf2f(xv[j]) + bf2f(rv[j]);
        nv[j] = f2bf(s);
        float f = bf2f(nv[j]);  // norm of the bf16-rounded sum (matches ref)
        ss += f * f;
      }
      *(short8v*)(rr + i) = nv;
    }
    ss = wave_sum(ss);
    if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = ss;
    __syncthreads();
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) tot += red[i];
    float inv = rsqrtf(tot / H + eps);
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v rv = *(const short8v*)(rr + i);
      short8v wv = *(const short8v*)(w + i);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(rv[j]) * inv * bf2f(wv[j]));
      *(short8v*)(xr + i) = o;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// LayerNorm (GPT-2 path): y = (x - mu) / sqrt(var + eps) * w + b
// ---------------------------------------------------------------------------

template <int BLOCK>
__global__ void layernorm_kernel(short* __restrict__ out,
                                 const short* __restrict__ x,
                                 const short* __restrict__ w,
                                 const short* __restrict__ b, float eps,
                                 int T, int H) {
  __shared__ float red[BLOCK / WAVE];
  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    short* yr = out + (long)row * H;
    float sum = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v v = *(const short8v*)(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) sum += bf2f(v[j]);
    }
    sum = wave_sum(sum);
    if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = sum;
    __syncthreads();
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) tot += red[i];
    const float mu = tot / H;
    __syncthreads();
    float ss = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v v = *(const short8v*)(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float d = bf2f(v[j]) - mu;
        ss += d * d;
      }
    }
    ss = wave_sum(ss);
    if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = ss;
    __syncthreads();
    float vtot = 0.f;
#pragma unroll
    for (int i = 0; i < BLOCK / WAVE; ++i) vtot += red[i];
    const float inv = rsqrtf(vtot / H + eps);
    for (int i = threadIdx.x * 8; i < H; i += BLOCK * 8) {
      short8v v = *(const short8v*)(xr + i);
      short8v wv = *(const short8v*)(w + i);
      short8v bv = *(const short8v*)(b + i);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf((bf2f(v[j]) - mu) * inv * bf2f(wv[j]) + bf2f(bv[j]));
      *(short8v*)(yr + i) = o;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: out[t, i] = silu(in[t, i]) * in[t, I + i]     in: [T, 2I]
// ---------------------------------------------------------------------------

__global__ void silu_mul_kernel(short* __restrict__ out,
                                const short* __restrict__ in, long T, long I) {
  long idx = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long total = T * I;
  for (; idx < total; idx += (long)gridDim.x * blockDim.x * 8) {
    long t = idx / I, i = idx % I;
    short8v g = *(const short8v*)(in + t * 2 * I + i);
    short8v u = *(const short8v*)(in + t * 2 * I + I + i);
    short8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = f2bf(s * bf2f(u[j]));
    }
    *(short8v*)(out + t * I + i) = o;
  }
}

// ---------------------------------------------------------------------------
// Fused RoPE (rotate-half) + paged-KV append.
//   q: [T, Hq, D] bf16 (rotated in place)
//   k: [T, Hk, D] bf16 (rotated in place, then written to k_cache)
//   v: [T, Hk, D] bf16 (written to v_cache)
//   cos/sin: [P, D/2] fp32;  caches: [N, Hk, BS, D] bf16
//   slot_mapping: [T] int64 flat slot  (block = slot/BS, off = slot%BS)
// One wave per (token, head); lane l owns dims {l, l+D/2} when D==128
// (pairs 2 per lane via the rotate-half pairing), fp32 math.
// ---------------------------------------------------------------------------

__global__ void rope_kv_append_kernel(
    short* __restrict__ q, short* __restrict__ k, const short* __restrict__ v,
    const long* __restrict__ positions, const float* __restrict__ cos_t,
    const float* __restrict__ sin_t, short* __restrict__ k_cache,
    short* __restrict__ v_cache, const long* __restrict__ slots, int T, int Hq,
    int Hk, int D, int BS, long cache_head_stride, long cache_block_stride,
    int do_rope, long q_tstride, long k_tstride, long v_tstride) {
  // q/k/v may be row-strided VIEWS of a fused qkv projection (token-row
  // stride != H*D): all row addressing goes through *_tstride.
  // unit = one (token, head); heads 0..Hq-1 are q, Hq..Hq+Hk-1 are k,
  // Hq+Hk..Hq+2Hk-1 are v-copy-only
  int unit = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x & (WAVE - 1);
  int units = T * (Hq + 2 * Hk);
  if (unit >= units) return;
  int t = unit / (Hq + 2 * Hk);
  int h = unit % (Hq + 2 * Hk);
  long pos = positions[t];
  int half = D / 2;

  if (do_rope && h < Hq + Hk) {
    short* base = (h < Hq) ? q + (long)t * q_tstride + (long)h * D
                           : k + (long)t * k_tstride + (long)(h - Hq) * D;
    // lane l handles pair indices l, l+WAVE, ... over half
    for (int i = lane; i < half; i += WAVE) {
      float c = cos_t[pos * half + i];
      float s = sin_t[pos * half + i];
      float x1 = bf2f(base[i]);
      float x2 = bf2f(base[i + half]);
      base[i] = f2bf(x1 * c - x2 * s);
      base[i + half] = f2bf(x2 * c + x1 * s);
    }
  }
  if (h >= Hq) {
    // append to cache (k after rotation — same wave did the rotation above
    // for k units; v units copy straight through)
    bool is_v = h >= Hq + Hk;
    int kvh = is_v ? (h - Hq - Hk) : (h - Hq);
    const short* src = is_v ? v + (long)t * v_tstride + (long)kvh * D
                            : k + (long)t * k_tstride + (long)kvh * D;
    short* cache = is_v ? v_cache : k_cache;
    long slot = slots[t];
    long block = slot / BS, off = slot % BS;
    short* dst = cache + block * cache_block_stride + kvh * cache_head_stride +
                 off * D;
    for (int i = lane * 2; i < D; i += WAVE * 2) {
      // k was just rotated by THIS wave for k-units; safe (no cross-wave dep)
      *(int*)(dst + i) = *(const int*)(src + i);
    }
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static inline int grid_rows(int T) { return T < 2048 ? T : 2048; }

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  int T = x.numel() / x.size(-1), H = x.size(-1);
  TORCH_CHECK(H % (256 * 8) == 0 || H % 8 == 0, "H must be multiple of 8");
  hipLaunchKernelGGL((rmsnorm_kernel<256>), dim3(grid_rows(T)), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(),
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), (float)eps, T, H);
  HIP_CHECK_LAST();
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor w, double eps) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  int T = x.numel() / x.size(-1), H = x.size(-1);
  hipLaunchKernelGGL((fused_add_rmsnorm_kernel<256>), dim3(grid_rows(T)),
                     dim3(256), 0, c10::hip::getCurrentHIPStream(),
                     (short*)x.data_ptr(), (short*)residual.data_ptr(),
                     (const short*)w.data_ptr(), (float)eps, T, H);
  HIP_CHECK_LAST();
}

void layernorm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               torch::Tensor b, double eps) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  int T = x.numel() / x.size(-1), H = x.size(-1);
  TORCH_CHECK(H % 8 == 0);
  hipLaunchKernelGGL((layernorm_kernel<256>), dim3(grid_rows(T)), dim3(256),
                     0, c10::hip::getCurrentHIPStream(),
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), (const short*)b.data_ptr(),
                     (float)eps, T, H);
  HIP_CHECK_LAST();
}

void silu_mul(torch::Tensor out, torch::Tensor in) {
  TORCH_CHECK(in.scalar_type() == torch::kBFloat16 && in.is_contiguous());
  long T = in.size(0), I = in.size(1) / 2;
  TORCH_CHECK(I % 8 == 0);
  long total = T * I / 8;
  int blocks = (int)std::min<long>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(), (short*)out.data_ptr(),
                     (const short*)in.data_ptr(), T, I);
  HIP_CHECK_LAST();
}

static inline void check_qkv_layout(const torch::Tensor& t) {
  // [T, H, D] with contiguous (head, dim) inner layout; token-row stride
  // may be larger (a view of a fused qkv projection)
  TORCH_CHECK(t.dim() == 3 && t.stride(2) == 1 && t.stride(1) == t.size(2),
              "expected [T,H,D] with dense heads; got strides ", t.strides());
}

void rope_kv_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor positions, torch::Tensor cos_t,
                    torch::Tensor sin_t, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor slots) {
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32);
  check_qkv_layout(q);
  check_qkv_layout(k);
  check_qkv_layout(v);
  int T = q.size(0), Hq = q.size(1), D = q.size(2), Hk = k.size(1);
  int BS = k_cache.size(2);
  long head_stride = k_cache.stride(1), block_stride = k_cache.stride(0);
  int units = T * (Hq + 2 * Hk);
  int waves_per_block = 4;
  int blocks = (units + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(rope_kv_append_kernel, dim3(blocks),
                     dim3(waves_per_block * WAVE), 0,
                     c10::hip::getCurrentHIPStream(), (short*)q.data_ptr(),
                     (short*)k.data_ptr(), (const short*)v.data_ptr(),
                     (const long*)positions.data_ptr(),
                     (const float*)cos_t.data_ptr(),
                     (const float*)sin_t.data_ptr(),
                     (short*)k_cache.data_ptr(), (short*)v_cache.data_ptr(),
                     (const long*)slots.data_ptr(), T, Hq, Hk, D, BS,
                     head_stride, block_stride, 1, q.stride(0), k.stride(0),
                     v.stride(0));
  HIP_CHECK_LAST();
}

void kv_append(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
               torch::Tensor v_cache, torch::Tensor slots) {
  // no-rope append (GPT-2 path): reuse the same kernel with Hq = 0 and a
  // null q/rope — implemented by passing k as q with Hq=0.
  int T = k.size(0), Hk = k.size(1), D = k.size(2);
  int BS = k_cache.size(2);
  // positions unused for the copy path, but required by signature — pass
  // slots as positions (never dereferenced for copy-only units when Hq==0
  // ... it IS dereferenced (pos = positions[t]) but unused); safe.
  long head_stride = k_cache.stride(1), block_stride = k_cache.stride(0);
  int units = T * (0 + 2 * Hk);
  int blocks = (units + 3) / 4;
  hipLaunchKernelGGL(rope_kv_append_kernel, dim3(blocks), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(), (short*)k.data_ptr(),
                     (short*)k.data_ptr(), (const short*)v.data_ptr(),
                     (const long*)slots.data_ptr(), nullptr, nullptr,
                     (short*)k_cache.data_ptr(), (short*)v_cache.data_ptr(),
                     (const long*)slots.data_ptr(), T, 0, Hk, D, BS,
                     head_stride, block_stride, 0, 0, k.stride(0),
                     v.stride(0));
  HIP_CHECK_LAST();
}
