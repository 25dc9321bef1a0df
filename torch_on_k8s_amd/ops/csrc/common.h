// Common device helpers for MI355X (gfx950 / CDNA4) kernels.
//
// All kernels in this package are written directly for CDNA4: wave64,
// MFMA matrix cores, 160 KiB LDS per CU, HBM3E. No CUDA compatibility
// shims, no hipify output.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define DEVINL __device__ __forceinline__

// CDNA wavefront is 64 lanes (not 32).
constexpr int WAVE = 64;

using bf16_t = __hip_bfloat16;

// ---- bf16 <-> f32 ------------------------------------------------------
DEVINL float bf2f(bf16_t b) { return __bfloat162float(b); }
DEVINL bf16_t f2bf(float f) { return __float2bfloat16(f); }

DEVINL float bfbits2f(uint16_t u) {
  union { float f; uint32_t u; } c;
  c.u = uint32_t(u) << 16;
  return c.f;
}

DEVINL uint16_t f2bfbits(float f) {
  bf16_t b = __float2bfloat16(f);
  return reinterpret_cast<uint16_t&>(b);
}

// 8 bf16 loaded as one 16-byte vector (coalescing sweet spot: 16 B/lane).
struct alignas(16) bf16x8 {
  uint16_t h[8];
};
static_assert(sizeof(bf16x8) == 16, "bf16x8 must be 16 bytes");

struct alignas(16) f32x4v {
  float v[4];
};

// ---- wave / block reductions ------------------------------------------
DEVINL float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

DEVINL float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block-wide sum over up to 1024 threads; `scratch` must hold >= blockDim/64
// floats. All threads receive the result.
DEVINL float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  return total;
}

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return _e;                                           \
  } while (0)
