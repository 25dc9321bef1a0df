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
// 2D grid: blockIdx.x covers columns, blockIdx.y strides rows — no
// 64-bit integer division in the hot loop (no HW divide on CDNA; the
// div-per-iteration grid-stride form measured only ~5 TB/s).
__global__ void swiglu_fwd_kernel(const bf16x8* __restrict__ gu,
                                  bf16x8* __restrict__ out,
                                  long rows, int iv /* I/8 */) {
  const int c = blockIdx.x * BLOCK + threadIdx.x;
  if (c >= iv) return;
  for (long row = blockIdx.y; row < rows; row += gridDim.y) {
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
  const int c = blockIdx.x * BLOCK + threadIdx.x;
  if (c >= iv) return;
  for (long row = blockIdx.y; row < rows; row += gridDim.y) {
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

static dim3 elem2d_grid(long rows, int cvecs) {
  const int cblocks = (cvecs + BLOCK - 1) / BLOCK;
  long rblocks = rows;
  // target >= 2048 blocks to fill 256 CUs across 8 XCDs, cap row dim
  long cap = 16384 / (cblocks > 0 ? cblocks : 1);
  if (cap < 1) cap = 1;
  if (rblocks > cap) rblocks = cap;
  if (rblocks > 65535) rblocks = 65535;
  return dim3(cblocks, (int)rblocks);
}

hipError_t tok_swiglu_fwd(const void* gu, void* out, long rows, int I,
                          hipStream_t stream) {
  const int iv = I / 8;
  swiglu_fwd_kernel<<<elem2d_grid(rows, iv), BLOCK, 0, stream>>>(
      (const bf16x8*)gu, (bf16x8*)out, rows, iv);
  return hipGetLastError();
}

hipError_t tok_swiglu_bwd(const void* dout, const void* gu, void* dgu,
                          long rows, int I, hipStream_t stream) {
  const int iv = I / 8;
  swiglu_bwd_kernel<<<elem2d_grid(rows, iv), BLOCK, 0, stream>>>(
      (const bf16x8*)dout, (const bf16x8*)gu, (bf16x8*)dgu, rows, iv);
  return hipGetLastError();
}
}
