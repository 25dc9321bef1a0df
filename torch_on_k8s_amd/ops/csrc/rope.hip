// Rotary position embedding (Llama "rotate-half" convention) for MI355X.
//
// Memory-bound elementwise op. Per the CDNA guide (Appendix B): cos/sin are
// precomputed on the HOST into an fp32 [S, D/2] table — no on-device trig
// (device sinf/cosf turns this memory-bound op VALU-bound).
//
// x layout: [B, S, Hh, D] contiguous (rows = B*S*Hh of length D).
//   y[..., :D/2] = x1*cos - x2*sin
//   y[..., D/2:] = x2*cos + x1*sin
// backward is the inverse rotation: sign = -1.
#include "common.h"

namespace {

constexpr int BLOCK = 256;

// One thread handles one 8-element vec of the FIRST half plus its partner
// vec in the second half. rows_total = B*S*Hh, hv = (D/2)/8 vecs per half.
__global__ void rope_kernel(const bf16x8* __restrict__ x,
                            bf16x8* __restrict__ y,
                            const float* __restrict__ tab_cos,  // [table_rows, D/2]
                            const float* __restrict__ tab_sin,  // [table_rows, D/2]
                            long rows_total, int heads, int hv, long table_rows,
                            float sign) {
  const long nwork = rows_total * hv;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nwork;
       i += (long)gridDim.x * BLOCK) {
    const long row = i / hv;            // (b*S + s)*Hh + h
    const int c = (int)(i - row * hv);  // vec index within half
    const long pos = (row / heads) % table_rows;  // position within sequence
    const int D = hv * 16;              // full head dim
    const bf16x8* xr = x + row * (D / 8);
    bf16x8* yr = y + row * (D / 8);
    bf16x8 x1 = xr[c];
    bf16x8 x2 = xr[c + hv];
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
    yr[c] = o1;
    yr[c + hv] = o2;
  }
}

}  // namespace

extern "C" {

// x,y: bf16 [rows_total, D]; cos_tab/sin_tab: fp32 [table_rows, D/2];
// position = (row/heads) % table_rows.
hipError_t tok_rope(const void* x, void* y, const float* cos_tab,
                    const float* sin_tab, long rows_total, int heads, int D,
                    long table_rows, float sign, hipStream_t stream) {
  const int hv = (D / 2) / 8;
  const long nwork = rows_total * hv;
  long grid = (nwork + BLOCK - 1) / BLOCK;
  if (grid > 4096) grid = 4096;
  if (grid < 1) grid = 1;
  rope_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (const bf16x8*)x, (bf16x8*)y, cos_tab, sin_tab, rows_total, heads, hv,
      table_rows, sign);
  return hipGetLastError();
}
}
