// Fused softmax cross-entropy over bf16 logits for MI355X.
//
// Eager CE on Llama-3 logits ([tokens, 128256]) upcasts to an fp32 copy
// and runs multi-pass log_softmax fwd+bwd (~20 GB of HBM traffic per
// step at the flagship shape). Fused: forward reads the bf16 logits once
// (online max/sum in fp32, one block per row), backward reads them once
// more and writes bf16 grads directly: ~6 GB total.
//   fwd:  loss[i] = lse_i - logit[i, label_i];  lse saved for backward
//   bwd:  dlogit[i, v] = (exp(logit - lse) - [v == label]) * gscale
#include "common.h"

namespace {

constexpr int BLOCK = 256;

__global__ void ce_fwd_kernel(const bf16x8* __restrict__ logits,
                              const int* __restrict__ labels,
                              float* __restrict__ loss,
                              float* __restrict__ lse, long rows, int vc) {
  __shared__ float red_m[BLOCK / WAVE];
  __shared__ float red_s[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16x8* lr = logits + row * vc;
    // online (max, sumexp) per thread
    float m = -INFINITY, s = 0.f;
    for (int c = threadIdx.x; c < vc; c += BLOCK) {
      bf16x8 v = *(const bf16x8*)&lr[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float x = bfbits2f(v.h[j]);
        if (x > m) {
          s = s * __expf(m - x) + 1.f;
          m = x;
        } else {
          s += __expf(x - m);
        }
      }
    }
    // wave combine
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float mo = __shfl_xor(m, off, WAVE);
      const float so = __shfl_xor(s, off, WAVE);
      const float mn = fmaxf(m, mo);
      // guard: a lane with no elements has m = -inf (exp(-inf - -inf)
      // would be NaN)
      const float sa = (m == -INFINITY) ? 0.f : s * __expf(m - mn);
      const float sb = (mo == -INFINITY) ? 0.f : so * __expf(mo - mn);
      s = sa + sb;
      m = mn;
    }
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & 63) == 0) {
      red_m[wid] = m;
      red_s[wid] = s;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float M = red_m[0], S = red_s[0];
#pragma unroll 4
      for (int i = 1; i < BLOCK / WAVE; ++i) {
        const float mn = fmaxf(M, red_m[i]);
        const float sa = (M == -INFINITY) ? 0.f : S * __expf(M - mn);
        const float sb = (red_m[i] == -INFINITY)
                             ? 0.f : red_s[i] * __expf(red_m[i] - mn);
        S = sa + sb;
        M = mn;
      }
      const float l = M + __logf(S);
      lse[row] = l;
      const int lab = labels[row];
      const uint16_t* h = (const uint16_t*)lr;
      loss[row] = l - bfbits2f(h[lab]);
    }
    __syncthreads();
  }
}

__global__ void ce_bwd_kernel(const bf16x8* __restrict__ logits,
                              const int* __restrict__ labels,
                              const float* __restrict__ lse,
                              const float* __restrict__ gscale,  // device scalar
                              bf16x8* __restrict__ dlogits, long rows, int vc) {
  const float gs = *gscale;
  const long nwork = rows * vc;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nwork;
       i += (long)gridDim.x * BLOCK) {
    const long row = i / vc;
    const int c = (int)(i - row * vc);
    const float l = lse[row];
    const int lab = labels[row];
    bf16x8 v = *(const bf16x8*)&logits[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = __expf(bfbits2f(v.h[j]) - l);
      if (c * 8 + j == lab) g -= 1.f;
      o.h[j] = f2bfbits(g * gs);
    }
    dlogits[i] = o;
  }
}

}  // namespace

extern "C" {

hipError_t tok_ce_fwd(const void* logits, const int* labels, float* loss,
                      float* lse, long rows, int V, hipStream_t stream) {
  const int vc = V / 8;
  int grid = (int)(rows < 4096 ? rows : 4096);
  if (grid < 1) grid = 1;
  ce_fwd_kernel<<<grid, BLOCK, 0, stream>>>(
      (const bf16x8*)logits, labels, loss, lse, rows, vc);
  return hipGetLastError();
}

hipError_t tok_ce_bwd(const void* logits, const int* labels, const float* lse,
                      const float* gscale, void* dlogits, long rows, int V,
                      hipStream_t stream) {
  const int vc = V / 8;
  long grid = (rows * vc + BLOCK - 1) / BLOCK;
  if (grid > 4096) grid = 4096;
  ce_bwd_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (const bf16x8*)logits, labels, lse, gscale, (bf16x8*)dlogits, rows, vc);
  return hipGetLastError();
}
}
