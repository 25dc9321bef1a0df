// Packed-QKV split + rotary embedding in ONE pass for MI355X (gfx950).
//
// With q/k/v projections merged into one hipBLASLt GEMM (structural GEMM
// lever, docs/ROADMAP.md §1a) the projection output is
//   qkv [B, S, (Hq + 2*Hkv) * D]   heads ordered [q heads | k heads | v heads]
// This kernel splits it into contiguous q [B,S,Hq,D], k [B,S,Hkv,D],
// v [B,S,Hkv,D] (the flash-attention kernel's input layout) while
// applying RoPE to the q and k parts in the same pass — so packing adds
// only the v copy vs the unpacked path, and removes two rope launches
// plus the three GEMM output splits.
//
// Backward is the exact mirror: read dq/dk/dv, inverse-rotate dq/dk
// (sign = -1), write the packed dqkv the GEMM backward consumes.
//
// cos/sin: host-precomputed fp32 [table_rows, D/2] (no device trig —
// guide: memory-bound op must not become VALU-bound).
#include "common.h"

namespace {

constexpr int BLOCK = 256;

// Work item = one 8-wide vec of the FIRST half of one head row (its
// partner vec in the second half is handled by the same thread, as in
// rope_kernel). rows span tokens*heads_total.
__global__ void qkv_rope_kernel(const bf16x8* __restrict__ qkv,
                                bf16x8* __restrict__ q,
                                bf16x8* __restrict__ k,
                                bf16x8* __restrict__ v,
                                const float* __restrict__ tab_cos,
                                const float* __restrict__ tab_sin,
                                long tokens, int Hq, int Hkv, int hv /* (D/2)/8 */,
                                long table_rows, float sign) {
  const int heads_total = Hq + 2 * Hkv;
  const int dv8 = hv * 2;  // vecs per head (D/8)
  const long nwork = tokens * heads_total * hv;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nwork;
       i += (long)gridDim.x * BLOCK) {
    const long rowh = i / hv;                 // token*heads_total + h
    const int c = (int)(i - rowh * hv);       // vec index within half
    const long token = rowh / heads_total;
    const int h = (int)(rowh - token * heads_total);
    const bf16x8* src = qkv + (token * heads_total + h) * dv8;
    bf16x8 x1 = src[c];
    bf16x8 x2 = src[c + hv];
    bf16x8* dst;
    bool rot;
    if (h < Hq) {                 // q head
      dst = q + (token * Hq + h) * dv8;
      rot = true;
    } else if (h < Hq + Hkv) {    // k head
      dst = k + (token * Hkv + (h - Hq)) * dv8;
      rot = true;
    } else {                      // v head: plain copy
      dst = v + (token * Hkv + (h - Hq - Hkv)) * dv8;
      rot = false;
    }
    if (rot) {
      const long pos = token % table_rows;
      const float* cosr = tab_cos + pos * (hv * 8);
      const float* sinr = tab_sin + pos * (hv * 8);
      bf16x8 o1, o2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float cs = cosr[c * 8 + j];
        const float sn = sinr[c * 8 + j] * sign;
        const float a = bfbits2f(x1.h[j]);
        const float b = bfbits2f(x2.h[j]);
        o1.h[j] = f2bfbits(a * cs - b * sn);
        o2.h[j] = f2bfbits(b * cs + a * sn);
      }
      dst[c] = o1;
      dst[c + hv] = o2;
    } else {
      dst[c] = x1;
      dst[c + hv] = x2;
    }
  }
}

// Mirror: gather dq/dk/dv into packed dqkv, inverse-rotating dq/dk.
__global__ void qkv_rope_bwd_kernel(const bf16x8* __restrict__ dq,
                                    const bf16x8* __restrict__ dk,
                                    const bf16x8* __restrict__ dv,
                                    bf16x8* __restrict__ dqkv,
                                    const float* __restrict__ tab_cos,
                                    const float* __restrict__ tab_sin,
                                    long tokens, int Hq, int Hkv, int hv,
                                    long table_rows) {
  const int heads_total = Hq + 2 * Hkv;
  const int dv8 = hv * 2;
  const long nwork = tokens * heads_total * hv;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nwork;
       i += (long)gridDim.x * BLOCK) {
    const long rowh = i / hv;
    const int c = (int)(i - rowh * hv);
    const long token = rowh / heads_total;
    const int h = (int)(rowh - token * heads_total);
    const bf16x8* src;
    bool rot;
    if (h < Hq) {
      src = dq + (token * Hq + h) * dv8;
      rot = true;
    } else if (h < Hq + Hkv) {
      src = dk + (token * Hkv + (h - Hq)) * dv8;
      rot = true;
    } else {
      src = dv + (token * Hkv + (h - Hq - Hkv)) * dv8;
      rot = false;
    }
    bf16x8 x1 = src[c];
    bf16x8 x2 = src[c + hv];
    bf16x8* dst = dqkv + (token * heads_total + h) * dv8;
    if (rot) {
      const long pos = token % table_rows;
      const float* cosr = tab_cos + pos * (hv * 8);
      const float* sinr = tab_sin + pos * (hv * 8);
      bf16x8 o1, o2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float cs = cosr[c * 8 + j];
        const float sn = -sinr[c * 8 + j];  // inverse rotation
        const float a = bfbits2f(x1.h[j]);
        const float b = bfbits2f(x2.h[j]);
        o1.h[j] = f2bfbits(a * cs - b * sn);
        o2.h[j] = f2bfbits(b * cs + a * sn);
      }
      dst[c] = o1;
      dst[c + hv] = o2;
    } else {
      dst[c] = x1;
      dst[c + hv] = x2;
    }
  }
}

}  // namespace

extern "C" {

hipError_t tok_qkv_rope_fwd(const void* qkv, void* q, void* k, void* v,
                            const float* cos_tab, const float* sin_tab,
                            long tokens, int Hq, int Hkv, int D,
                            long table_rows, hipStream_t stream) {
  const int hv = (D / 2) / 8;
  const long nwork = tokens * (long)(Hq + 2 * Hkv) * hv;
  long grid = (nwork + BLOCK - 1) / BLOCK;
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  qkv_rope_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (const bf16x8*)qkv, (bf16x8*)q, (bf16x8*)k, (bf16x8*)v, cos_tab,
      sin_tab, tokens, Hq, Hkv, hv, table_rows, 1.0f);
  return hipGetLastError();
}

hipError_t tok_qkv_rope_bwd(const void* dq, const void* dk, const void* dv,
                            void* dqkv, const float* cos_tab,
                            const float* sin_tab, long tokens, int Hq,
                            int Hkv, int D, long table_rows,
                            hipStream_t stream) {
  const int hv = (D / 2) / 8;
  const long nwork = tokens * (long)(Hq + 2 * Hkv) * hv;
  long grid = (nwork + BLOCK - 1) / BLOCK;
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  qkv_rope_bwd_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (const bf16x8*)dq, (const bf16x8*)dk, (const bf16x8*)dv, (bf16x8*)dqkv,
      cos_tab, sin_tab, tokens, Hq, Hkv, hv, table_rows);
  return hipGetLastError();
}
}
