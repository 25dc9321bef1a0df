// MFMA fragment-layout probe for gfx950.
//
// Verifies on hardware the lane->element mapping assumed by the flash
// attention kernel (guide cdna_hip_programming.md §3):
//   mfma_f32_16x16x32_bf16:
//     A[m][k]: m = lane&15, k = (lane>>4)*8 + j   (j = 0..7)
//     B[k][n]: k = (lane>>4)*8 + j, n = lane&15
//     C/D[m][n]: m = (lane>>4)*4 + r, n = lane&15 (r = 0..3)
// Tested against torch.matmul with ASYMMETRIC inputs (transpose-detecting,
// guide §5.4 rule 16).
#include "common.h"

namespace {

using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void mfma_probe_16x16x32_kernel(const __bf16* __restrict__ A,
                                           const __bf16* __restrict__ B,
                                           float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  bf16x8v a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(lane & 15) * 32 + ((lane >> 4) * 8 + j)];
    b[j] = B[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = c[r];
}

//  mfma_f32_32x32x16_bf16 (assumed, to verify):
//    A[m][k]: m = lane&31, k = (lane>>5)*8 + j     (8 elems/lane)
//    B[k][n]: k = (lane>>5)*8 + j, n = lane&31
//    C/D[m][n]: n = lane&31, m = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//    (guide §3: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5), 16 regs)
using f32x16 = __attribute__((ext_vector_type(16))) float;

__global__ void mfma_probe_32x32x16_kernel(const __bf16* __restrict__ A,
                                           const __bf16* __restrict__ B,
                                           float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  bf16x8v a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(lane & 31) * 16 + ((lane >> 5) * 8 + j)];
    b[j] = B[((lane >> 5) * 8 + j) * 32 + (lane & 31)];
  }
  f32x16 c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    D[row * 32 + (lane & 31)] = c[r];
  }
}

}  // namespace

extern "C" {

// A: [16,32] bf16 row-major, B: [32,16] bf16 row-major, D: [16,16] f32.
hipError_t tok_mfma_probe_16x16x32(const void* A, const void* B, float* D,
                                   hipStream_t stream) {
  mfma_probe_16x16x32_kernel<<<1, 64, 0, stream>>>(
      (const __bf16*)A, (const __bf16*)B, D);
  return hipGetLastError();
}

// A: [32,16] bf16 row-major, B: [16,32] bf16 row-major, D: [32,32] f32.
hipError_t tok_mfma_probe_32x32x16(const void* A, const void* B, float* D,
                                   hipStream_t stream) {
  mfma_probe_32x32x16_kernel<<<1, 64, 0, stream>>>(
      (const __bf16*)A, (const __bf16*)B, D);
  return hipGetLastError();
}
}
