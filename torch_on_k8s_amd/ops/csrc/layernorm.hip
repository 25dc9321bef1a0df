// LayerNorm forward/backward for MI355X (gfx950) — the GPT-family
// normalization (RMSNorm's mean-centered sibling; same memory-bound
// structure: bf16x8 vector I/O, fp32 accumulation, one block per row,
// column-parallel dweight/dbias with per-block register accumulation).
//   y = (x - mu) * rstd * w + b,  rstd = 1/sqrt(var + eps)
//   dx = rstd * (dyw - mean(dyw) - xhat * mean(dyw * xhat)),
//        dyw = dy * w,  xhat = (x - mu) * rstd
//   dw = sum_rows dy * xhat,  db = sum_rows dy
#include "common.h"

namespace {

constexpr int BLOCK = 256;

__global__ void ln_fwd_kernel(const bf16x8* __restrict__ x,
                              const bf16x8* __restrict__ w,
                              const bf16x8* __restrict__ b,
                              bf16x8* __restrict__ y,
                              float* __restrict__ mu_out,
                              float* __restrict__ rstd_out,
                              long nrows, int hc, float eps) {
  __shared__ float red[BLOCK / WAVE];
  const int H = hc * 8;
  for (long row = blockIdx.x; row < nrows; row += gridDim.x) {
    const bf16x8* xr = x + row * hc;
    float s1 = 0.f, s2 = 0.f;
    for (int c = threadIdx.x; c < hc; c += BLOCK) {
      bf16x8 v = xr[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bfbits2f(v.h[j]);
        s1 += f;
        s2 = fmaf(f, f, s2);
      }
    }
    s1 = block_reduce_sum(s1, red);
    __syncthreads();
    s2 = block_reduce_sum(s2, red);
    const float mu = s1 / (float)H;
    const float var = s2 / (float)H - mu * mu;
    const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mu_out[row] = mu;
      rstd_out[row] = rstd;
    }
    bf16x8* yr = y + row * hc;
    for (int c = threadIdx.x; c < hc; c += BLOCK) {
      bf16x8 xv = xr[c], wv = w[c], bv = b[c], ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = (bfbits2f(xv.h[j]) - mu) * rstd * bfbits2f(wv.h[j]) +
                        bfbits2f(bv.h[j]);
        ov.h[j] = f2bfbits(f);
      }
      yr[c] = ov;
    }
    __syncthreads();
  }
}

__global__ void ln_bwd_dx_kernel(const bf16x8* __restrict__ x,
                                 const bf16x8* __restrict__ w,
                                 const bf16x8* __restrict__ dy,
                                 const float* __restrict__ mu_in,
                                 const float* __restrict__ rstd_in,
                                 bf16x8* __restrict__ dx,
                                 long nrows, int hc) {
  __shared__ float red[BLOCK / WAVE];
  const int H = hc * 8;
  for (long row = blockIdx.x; row < nrows; row += gridDim.x) {
    const bf16x8* xr = x + row * hc;
    const bf16x8* dyr = dy + row * hc;
    const float mu = mu_in[row];
    const float rstd = rstd_in[row];
    float a1 = 0.f, a2 = 0.f;  // mean(dyw), mean(dyw * xhat)
    for (int c = threadIdx.x; c < hc; c += BLOCK) {
      bf16x8 xv = xr[c], wv = w[c], dv = dyr[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float dyw = bfbits2f(dv.h[j]) * bfbits2f(wv.h[j]);
        const float xh = (bfbits2f(xv.h[j]) - mu) * rstd;
        a1 += dyw;
        a2 = fmaf(dyw, xh, a2);
      }
    }
    a1 = block_reduce_sum(a1, red);
    __syncthreads();
    a2 = block_reduce_sum(a2, red);
    a1 /= (float)H;
    a2 /= (float)H;
    bf16x8* dxr = dx + row * hc;
    for (int c = threadIdx.x; c < hc; c += BLOCK) {
      bf16x8 xv = xr[c], wv = w[c], dv = dyr[c], ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float dyw = bfbits2f(dv.h[j]) * bfbits2f(wv.h[j]);
        const float xh = (bfbits2f(xv.h[j]) - mu) * rstd;
        ov.h[j] = f2bfbits(rstd * (dyw - a1 - xh * a2));
      }
      dxr[c] = ov;
    }
    __syncthreads();
  }
}

__global__ void ln_bwd_dwdb_kernel(const bf16x8* __restrict__ x,
                                   const bf16x8* __restrict__ dy,
                                   const float* __restrict__ mu_in,
                                   const float* __restrict__ rstd_in,
                                   float* __restrict__ dw,
                                   float* __restrict__ db,
                                   long nrows, int hc) {
  const int c = blockIdx.x * BLOCK + threadIdx.x;
  if (c >= hc) return;
  const long row0 = blockIdx.y;
  const long rstride = gridDim.y;
  float aw[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float ab[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (long row = row0; row < nrows; row += rstride) {
    const float mu = mu_in[row];
    const float rstd = rstd_in[row];
    bf16x8 xv = x[row * hc + c];
    bf16x8 dv = dy[row * hc + c];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float d = bfbits2f(dv.h[j]);
      aw[j] = fmaf(d, (bfbits2f(xv.h[j]) - mu) * rstd, aw[j]);
      ab[j] += d;
    }
  }
  float* ow = dw + (long)c * 8;
  float* ob = db + (long)c * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(ow + j, aw[j]);
    atomicAdd(ob + j, ab[j]);
  }
}

}  // namespace

extern "C" {

hipError_t tok_layernorm_fwd(const void* x, const void* w, const void* b,
                             void* y, float* mu, float* rstd, long nrows,
                             int H, float eps, hipStream_t stream) {
  const int hc = H / 8;
  int grid = (int)(nrows < 8192 ? nrows : 8192);
  if (grid < 1) grid = 1;
  ln_fwd_kernel<<<grid, BLOCK, 0, stream>>>(
      (const bf16x8*)x, (const bf16x8*)w, (const bf16x8*)b, (bf16x8*)y, mu,
      rstd, nrows, hc, eps);
  return hipGetLastError();
}

hipError_t tok_layernorm_bwd(const void* x, const void* w, const void* dy,
                             const float* mu, const float* rstd, void* dx,
                             float* dw, float* db, long nrows, int H,
                             hipStream_t stream) {
  const int hc = H / 8;
  int grid = (int)(nrows < 8192 ? nrows : 8192);
  if (grid < 1) grid = 1;
  ln_bwd_dx_kernel<<<grid, BLOCK, 0, stream>>>(
      (const bf16x8*)x, (const bf16x8*)w, (const bf16x8*)dy, mu, rstd,
      (bf16x8*)dx, nrows, hc);
  int cblocks = (hc + BLOCK - 1) / BLOCK;
  int rsplit = 2048 / (cblocks > 0 ? cblocks : 1);
  if (rsplit < 1) rsplit = 1;
  if ((long)rsplit > nrows) rsplit = (int)nrows;
  dim3 g(cblocks, rsplit);
  ln_bwd_dwdb_kernel<<<g, BLOCK, 0, stream>>>(
      (const bf16x8*)x, (const bf16x8*)dy, mu, rstd, dw, db, nrows, hc);
  return hipGetLastError();
}
}
