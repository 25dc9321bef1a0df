// Fused AdamW on flat parameter buckets for MI355X.
//
// The trainer keeps every parameter as a view into large flat bf16 buckets
// (MI355X-first: 288 GB HBM3E -> few, large buffers; the gradient bucket is
// also the RCCL all-reduce buffer). One kernel launch updates a whole
// bucket: bf16 params/grads, fp32 exp_avg/exp_avg_sq, fp32 math, decoupled
// weight decay, bias correction, and the DP 1/world_size gradient scale
// folded into the grad read (saves a separate scaling pass over HBM).
#include "common.h"

namespace {

constexpr int BLOCK = 256;

__global__ void adamw_kernel(bf16x8* __restrict__ p, const bf16x8* __restrict__ g,
                             f32x4v* __restrict__ m, f32x4v* __restrict__ v,
                             long nvec, float lr, float beta1, float beta2,
                             float eps, float wd, float bc1, float bc2,
                             float gscale, const int* __restrict__ step_dev) {
  if (step_dev != nullptr) {
    // hipGraph-replay mode: the step counter lives on-device (a host
    // scalar would be frozen into the captured graph), so the bias
    // corrections are computed here.
    const float t = (float)*step_dev;
    bc1 = 1.f / (1.f - __powf(beta1, t));
    bc2 = 1.f / (1.f - __powf(beta2, t));
  }
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nvec;
       i += (long)gridDim.x * BLOCK) {
    bf16x8 pv = p[i];
    bf16x8 gv = g[i];
    f32x4v m0 = m[i * 2], m1 = m[i * 2 + 1];
    f32x4v v0 = v[i * 2], v1 = v[i * 2 + 1];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float* mj = (j < 4) ? &m0.v[j] : &m1.v[j - 4];
      float* vj = (j < 4) ? &v0.v[j] : &v1.v[j - 4];
      const float grad = bfbits2f(gv.h[j]) * gscale;
      float pf = bfbits2f(pv.h[j]);
      pf -= lr * wd * pf;  // decoupled weight decay
      const float mn = beta1 * (*mj) + (1.f - beta1) * grad;
      const float vn = beta2 * (*vj) + (1.f - beta2) * grad * grad;
      *mj = mn;
      *vj = vn;
      const float mhat = mn * bc1;  // bc1 = 1/(1-beta1^t)
      const float vhat = vn * bc2;  // bc2 = 1/(1-beta2^t)
      pf -= lr * mhat / (sqrtf(vhat) + eps);
      pv.h[j] = f2bfbits(pf);
    }
    p[i] = pv;
    m[i * 2] = m0;
    m[i * 2 + 1] = m1;
    v[i * 2] = v0;
    v[i * 2 + 1] = v1;
  }
}

}  // namespace

extern "C" {

// All pointers flat+contiguous; n must be a multiple of 8 (the trainer pads
// buckets to 8 elements).
hipError_t tok_adamw(void* p, const void* g, float* m, float* v, long n,
                     float lr, float beta1, float beta2, float eps, float wd,
                     int step, float gscale, const int* step_dev,
                     hipStream_t stream) {
  const long nvec = n / 8;
  const float bc1 = 1.f / (1.f - powf(beta1, (float)step));
  const float bc2 = 1.f / (1.f - powf(beta2, (float)step));
  long grid = (nvec + BLOCK - 1) / BLOCK;
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  adamw_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (bf16x8*)p, (const bf16x8*)g, (f32x4v*)m, (f32x4v*)v, nvec, lr, beta1,
      beta2, eps, wd, bc1, bc2, gscale, step_dev);
  return hipGetLastError();
}
}
