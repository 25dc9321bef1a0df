// Head-dim transpose: [B, S, H, D] -> [B, H, D, S_pad] bf16.
//
// Why: the flash-attention kernels consume V (fwd) and Q/K/dO (bwd) as
// MFMA B-operands, which need d-major (transposed) LDS tiles. Building
// those tiles with in-kernel scalar scatter writes measured ~20 LDS
// bank-conflict cycles per MFMA (profiles/README.md). Instead this tiny
// memory-bound kernel transposes each head's [S, D] activation once per
// attention call; the attention kernels then stage transposed tiles
// with plain vectorized (b128) LDS writes.
//
// Structure per 64x64 tile (one 256-thread block):
//   coalesced global loads (8 lanes cover one s-row's 128 B)
//   -> swizzled row-major LDS tile (conflict-free b128 writes)
//   -> per-thread column gather (8 x u16 LDS reads, banks spread by d)
//   -> coalesced 16 B global stores (8 lanes cover one d-row's 128 B)
// S is padded to a multiple of 64 in the output; pad columns are zero.
#include "common.h"

namespace {

constexpr int TT = 64;       // tile side
constexpr int TNT = 256;     // threads per block

DEVINL int tswz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

__global__ __launch_bounds__(TNT) void transpose_head_kernel(
    const bf16_t* __restrict__ in,  // [B, S, H, D]
    bf16_t* __restrict__ out,       // [B, H, D, S_pad]
    int S, int H, int D, int S_pad) {
  __shared__ __attribute__((aligned(16))) char lds[TT * TT * 2];
  const int s0 = blockIdx.x * TT;
  const int d0 = blockIdx.y * TT;
  const int bh = blockIdx.z;      // b * H + h
  const int b = bh / H, h = bh - b * H;

  const long in_tok = (long)H * D;
  const bf16_t* ip = in + ((long)b * S) * in_tok + (long)h * D + d0;
  bf16_t* op = out + ((long)bh * D + d0) * S_pad + s0;

  // load 64 x 64 (s x d), two passes of 32 rows
#pragma unroll
  for (int p = 0; p < TT * (TT / 8) / TNT; ++p) {
    const int vi = threadIdx.x + p * TNT;
    const int row = vi / 8, cv = vi % 8;   // row = s-local
    uint4 val = {0, 0, 0, 0};
    if (s0 + row < S)
      val = *(const uint4*)(ip + (long)(s0 + row) * in_tok + cv * 8);
    *(uint4*)(lds + row * 128 + tswz(row, cv * 16)) = val;
  }
  __syncthreads();

  // emit 64 x 64 (d x s): thread t covers d-row d0+(t>>3), s chunk (t&7)*8
#pragma unroll
  for (int p = 0; p < TT * (TT / 8) / TNT; ++p) {
    const int u = threadIdx.x + p * TNT;
    const int dr = u >> 3;              // d-local (two passes: 0..31, 32..63)
    const int sc = (u & 7) * 8;         // s-local chunk start
    uint16_t vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int s = sc + j;
      vals[j] = *(const uint16_t*)(lds + s * 128 + tswz(s, dr * 2));
    }
    *(uint4*)(op + (long)dr * S_pad + sc) = *(const uint4*)vals;
  }
}

}  // namespace

extern "C" {

hipError_t tok_transpose_head(const void* in, void* out, int B, int S, int H,
                              int D, int S_pad, hipStream_t stream) {
  if (D % 64 != 0 || S_pad % 64 != 0) return hipErrorInvalidValue;
  dim3 grid(S_pad / TT, D / TT, B * H);
  transpose_head_kernel<<<grid, TNT, 0, stream>>>(
      (const bf16_t*)in, (bf16_t*)out, S, H, D, S_pad);
  return hipGetLastError();
}
}
