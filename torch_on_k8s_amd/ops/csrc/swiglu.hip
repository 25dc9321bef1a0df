// Fused SwiGLU over the PACKED gate_up projection for MI355X (gfx950).
//
// The MLP computes down( silu(gate(x)) * up(x) ). With gate and up
// merged into ONE hipBLASLt GEMM (fewer, larger GEMMs — the xGMI/HBM
// -era structural lever, docs/ROADMAP.md §1a), its output is
// gu = [rows, 2I] with gate in [:, :I] and up in [:, I:].
//
// This kernel pair replaces torch's separate silu + mul (+ their three
// backward elementwise kernels) with one pass each way:
//   fwd: read gu (2 passes of I), write out (1)        — was 3R+2W
//   bwd: read dout+gu (3), write dgu (2)               — was 6R+3W
// At [32768, 14336] bf16 each saved pass is ~940 MB of HBM traffic.
// No intermediate silu(g) tensor is materialized or saved for backward
// (recomputed from gu in the bwd pass — cheaper than an HBM roundtrip).
//
// Capability parity: the reference schedules opaque containers
// (SURVEY.md §0); this is part of the MI355X-native data plane.
#include "common.h"

namespace {

constexpr int BLOCK = 256;

DEVINL float sigmoidf(float x) { return 1.0f / (1.0f + __expf(-x)); }

// out[r, c] = silu(g) * u,  g = gu[r, c], u = gu[r, I + c]
__global__ void swiglu_fwd_kernel(const bf16x8* __restrict__ gu,
                                  bf16x8* __restrict__ out,
                                  long rows, int iv /* I/8 */) {
  const long nwork = rows * iv;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nwork;
       i += (long)gridDim.x * BLOCK) {
    const long row = i / iv;
    const int c = (int)(i - row * iv);
    const bf16x8* gur = gu + row * (2 * iv);
    bf16x8 g = gur[c], u = gur[c + iv], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bfbits2f(g.h[j]);
      const float s = gf * sigmoidf(gf);
      o.h[j] = f2bfbits(s * bfbits2f(u.h[j]));
    }
    out[row * iv + c] = o;
  }
}

// dgu[r, c]     = dout * u * dsilu(g),  dsilu(g) = sig(g)*(1 + g*(1-sig(g)))
// dgu[r, I + c] = dout * silu(g)
__global__ void swiglu_bwd_kernel(const bf16x8* __restrict__ dout,
                                  const bf16x8* __restrict__ gu,
                                  bf16x8* __restrict__ dgu,
                                  long rows, int iv) {
  const long nwork = rows * iv;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nwork;
       i += (long)gridDim.x * BLOCK) {
    const long row = i / iv;
    const int c = (int)(i - row * iv);
    const bf16x8* gur = gu + row * (2 * iv);
    bf16x8 g = gur[c], u = gur[c + iv];
    bf16x8 do8 = dout[row * iv + c];
    bf16x8 dg, du;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bfbits2f(g.h[j]);
      const float sg = sigmoidf(gf);
      const float silu = gf * sg;
      const float dsilu = sg * (1.0f + gf * (1.0f - sg));
      const float d = bfbits2f(do8.h[j]);
      dg.h[j] = f2bfbits(d * bfbits2f(u.h[j]) * dsilu);
      du.h[j] = f2bfbits(d * silu);
    }
    bf16x8* dgur = dgu + row * (2 * iv);
    dgur[c] = dg;
    dgur[c + iv] = du;
  }
}

}  // namespace

extern "C" {

hipError_t tok_swiglu_fwd(const void* gu, void* out, long rows, int I,
                          hipStream_t stream) {
  const int iv = I / 8;
  const long nwork = rows * iv;
  long grid = (nwork + BLOCK - 1) / BLOCK;
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  swiglu_fwd_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (const bf16x8*)gu, (bf16x8*)out, rows, iv);
  return hipGetLastError();
}

hipError_t tok_swiglu_bwd(const void* dout, const void* gu, void* dgu,
                          long rows, int I, hipStream_t stream) {
  const int iv = I / 8;
  const long nwork = rows * iv;
  long grid = (nwork + BLOCK - 1) / BLOCK;
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  swiglu_bwd_kernel<<<(int)grid, BLOCK, 0, stream>>>(
      (const bf16x8*)dout, (const bf16x8*)gu, (bf16x8*)dgu, rows, iv);
  return hipGetLastError();
}
}
