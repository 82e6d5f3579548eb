// Shared helpers for dts_amd CDNA4 kernels (gfx950 only).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

// 8 bf16 = 16 B, the coalescing sweet spot (guide G13).
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

DEV float bf2f(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

DEV short f2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  // round-to-nearest-even
  unsigned int lsb = (c.i >> 16) & 1;
  c.i += 0x7fff + lsb;
  return (short)(c.i >> 16);
}

// wave-wide sum over 64 lanes
DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

#define HIP_CHECK_LAST()                                                      \
  do {                                                                        \
    hipError_t e = hipGetLastError();                                         \
    if (e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
    }                                                                         \
  } while (0)
