// Flash attention (causal, GQA) for MI355X (gfx950) — forward + backward.
//
// MI355X-first design (cdna_hip_programming.md §Appendix B):
//  * MFMA 16x16x32 bf16 tiles, fp32 accumulation in AGPR/VGPR file.
//  * wave64: a block = 4 waves; each wave owns a 16-row strip of a
//    64x64 score tile; online softmax (running m,l) entirely in
//    registers with 16-lane shuffle row-reductions.
//  * K / V / Q / dO tiles staged in LDS with the XOR swizzle
//    (byte ^= (row&7)<<4) — row-major [*][128] bf16 tiles read by
//    ds_read_b128 are otherwise an up-to-16-way bank conflict
//    (guide §6 Guideline 4).
//  * V (and dO/Q in backward) additionally staged TRANSPOSED so every
//    MFMA B-fragment read is one contiguous ds_read_b128.
//  * Fragment layouts (verified on hardware by mfma_probe):
//      A[m][k]: m=lane&15, k=(lane>>4)*8+j
//      B[k][n]: k=(lane>>4)*8+j, n=lane&15
//      C/D[m][n]: m=(lane>>4)*4+r, n=lane&15
//
// Backward uses the standard two-kernel flash scheme with recompute:
//   dkdv: grid over KV tiles; computes S^T directly (A=K,B=Q) so P^T
//         needs no lse/Dsum transpose; accumulates dK,dV in registers
//         across Q tiles and the GQA q-head group.
//   dq:   grid over Q tiles; recomputes S,P and accumulates dQ.
// Plus a tiny preprocess kernel: Dsum = rowsum(dO * O).
#include "common.h"

namespace {

using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 64;   // q rows per block
constexpr int BN = 64;   // kv rows per tile
constexpr int NW = 4;    // waves per block
constexpr int NTHREADS = NW * WAVE;
constexpr float NEG_INF = -INFINITY;

DEVINL int swz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

DEVINL bf16x8v as_frag(uint4 raw) {
  union { uint4 u; bf16x8v f; } c;
  c.u = raw;
  return c.f;
}

// ---- cooperative tile staging -----------------------------------------
// Row-major [64][D] bf16 tile, swizzled; rows beyond row_limit -> 0.
template <int D>
DEVINL void load_tile_rm(char* lds, const bf16_t* src, long row0,
                         long row_limit, long tok_stride) {
  constexpr int VPR = D / 8;
  constexpr int NV = 64 * VPR;
#pragma unroll 2
  for (int vi = threadIdx.x; vi < NV; vi += NTHREADS) {
    const int row = vi / VPR, cv = vi % VPR;
    uint4 val = {0, 0, 0, 0};
    if (row0 + row < row_limit)
      val = *(const uint4*)(src + (row0 + row) * tok_stride + cv * 8);
    *(uint4*)(lds + row * (D * 2) + swz(row, cv * 16)) = val;
  }
}

// Transposed [D][64] bf16 tile (row = d, col = src row), swizzled.
template <int D>
DEVINL void load_tile_tr(char* lds, const bf16_t* src, long row0,
                         long row_limit, long tok_stride) {
  constexpr int VPR = D / 8;
  constexpr int NV = 64 * VPR;
#pragma unroll 2
  for (int vi = threadIdx.x; vi < NV; vi += NTHREADS) {
    const int row = vi / VPR, cv = vi % VPR;
    uint4 val = {0, 0, 0, 0};
    if (row0 + row < row_limit)
      val = *(const uint4*)(src + (row0 + row) * tok_stride + cv * 8);
    const uint16_t* h = (const uint16_t*)&val;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = cv * 8 + j;
      *(uint16_t*)(lds + d * 128 + swz(d, row * 2)) = h[j];
    }
  }
}

// B-fragment read from a swizzled row-major LDS tile with row stride RB
// bytes: B[k][n] where n = row (16 rows starting at row0), k contiguous.
template <int RB>
DEVINL bf16x8v read_bfrag(const char* lds, int row, int col_elem) {
  return as_frag(
      *(const uint4*)(lds + row * RB + swz(row, col_elem * 2)));
}

// Load an A-fragment set (rows m = lane&15 within a 16-row strip) straight
// from global memory into registers; OOB rows -> 0.
template <int D>
DEVINL void load_afrags(bf16x8v* frag, const bf16_t* src, long row,
                        long row_limit, long tok_stride, int lane) {
  constexpr int DC = D / 32;
#pragma unroll
  for (int c = 0; c < DC; ++c) {
    uint4 raw = {0, 0, 0, 0};
    if (row < row_limit)
      raw = *(const uint4*)(src + row * tok_stride + c * 32 + (lane >> 4) * 8);
    frag[c] = as_frag(raw);
  }
}

// ========================================================================
// Forward
// ========================================================================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(NTHREADS) void attn_fwd_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
    float* __restrict__ lse, int B, int S, int Hq, int Hkv, float scale) {
  constexpr int DC = D / 32;  // MFMA k-chunks
  constexpr int DT = D / 16;  // output col tiles
  constexpr int KB = 64 * D * 2;
  // LDS carve: K tile | VT tile | P scratch (2 KiB per wave)
  __shared__ __attribute__((aligned(16))) char smem[KB + KB + NW * 2048];
  char* k_lds = smem;
  char* vt_lds = smem + KB;
  char* p_lds = smem + 2 * KB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int m0 = blockIdx.x * BM;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* vp = v + ((long)b * S * kv_tok) + (long)hkv * D;
  bf16_t* op = o + ((long)b * S * q_tok) + (long)hq * D;
  float* lsep = lse + ((long)b * Hq + hq) * S;

  // Q fragments for this wave's 16-row strip
  bf16x8v q_frag[DC];
  const int mrow = m0 + wid * 16 + (lane & 15);
  load_afrags<D>(q_frag, qp, mrow, S, q_tok, lane);

  f32x4 o_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) o_acc[t] = {0.f, 0.f, 0.f, 0.f};
  float m_run[4] = {NEG_INF, NEG_INF, NEG_INF, NEG_INF};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};

  const int n_end = CAUSAL ? min(S, m0 + BM) : S;
  for (int n0 = 0; n0 < n_end; n0 += BN) {
    load_tile_rm<D>(k_lds, kp, n0, S, kv_tok);
    load_tile_tr<D>(vt_lds, vp, n0, S, kv_tok);
    __syncthreads();

    const bool strip_live = !CAUSAL || (n0 <= m0 + wid * 16 + 15);
    if (strip_live) {
      // S strip: [16 rows x 64 cols] per wave
      f32x4 s_acc[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) s_acc[t] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int c = 0; c < DC; ++c) {
          bf16x8v bfr = read_bfrag<D * 2>(
              k_lds, t * 16 + (lane & 15), c * 32 + (lane >> 4) * 8);
          s_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[c], bfr, s_acc[t], 0, 0, 0);
        }

      // online softmax on the strip
      float p[4][4];     // [t][r]
      float mx[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) mx[r] = NEG_INF;
      const int row_base = m0 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int col = n0 + t * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = row_base + r;
          float s = s_acc[t][r] * scale;
          if ((CAUSAL && col > row) || col >= S || row >= S) s = NEG_INF;
          p[t][r] = s;
          mx[r] = fmaxf(mx[r], s);
        }
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          mx[r] = fmaxf(mx[r], __shfl_xor(mx[r], off, 64));

      float alpha[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float mn = fmaxf(m_run[r], mx[r]);
        alpha[r] = (m_run[r] == NEG_INF) ? 0.f : __expf(m_run[r] - mn);
        m_run[r] = mn;
      }
      float rowsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float pe =
              (p[t][r] == NEG_INF) ? 0.f : __expf(p[t][r] - m_run[r]);
          p[t][r] = pe;
          rowsum[r] += pe;
        }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          rowsum[r] += __shfl_xor(rowsum[r], off, 64);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        l_run[r] = l_run[r] * alpha[r] + rowsum[r];
#pragma unroll
      for (int t = 0; t < DT; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[t][r] *= alpha[r];

      // P -> LDS (per-wave scratch) to convert C-layout to A-layout
      char* pw = p_lds + wid * 2048;
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int rl = (lane >> 4) * 4 + r;
          const int cl = t * 16 + (lane & 15);
          *(uint16_t*)(pw + rl * 128 + swz(rl, cl * 2)) = f2bfbits(p[t][r]);
        }
      // PV: O += P * V
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8v pa = as_frag(*(const uint4*)(
            pw + (lane & 15) * 128 +
            swz(lane & 15, (kc * 32 + (lane >> 4) * 8) * 2)));
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v vb = read_bfrag<128>(
              vt_lds, t * 16 + (lane & 15), kc * 32 + (lane >> 4) * 8);
          o_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pa, vb, o_acc[t], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue: O = o_acc / l; LSE = m + log(l)
  const int row_base = m0 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row_base + r;
    if (row >= S) continue;
    const float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      const int col = t * 16 + (lane & 15);
      op[(long)row * q_tok + col] = f2bf(o_acc[t][r] * inv_l);
    }
    if ((lane & 15) == 0)
      lsep[row] = (l_run[r] > 0.f) ? m_run[r] + __logf(l_run[r]) : NEG_INF;
  }
}

// ========================================================================
// Backward preprocess: Dsum[b,h,m] = sum_d dO*O (fp32)
// ========================================================================
template <int D>
__global__ void attn_bwd_pre_kernel(const bf16_t* __restrict__ dout,
                                    const bf16_t* __restrict__ o,
                                    float* __restrict__ dsum, long rows,
                                    int Hq, int S) {
  // one thread per row; row index = (b*S + s)*Hq + h ; dsum is [B,Hq,S]
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < rows;
       i += (long)gridDim.x * blockDim.x) {
    const bf16_t* dp = dout + i * D;
    const bf16_t* opr = o + i * D;
    float acc = 0.f;
#pragma unroll 4
    for (int cv = 0; cv < D / 8; ++cv) {
      uint4 a = *(const uint4*)(dp + cv * 8);
      uint4 b = *(const uint4*)(opr + cv * 8);
      const uint16_t* ah = (const uint16_t*)&a;
      const uint16_t* bh = (const uint16_t*)&b;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc = fmaf(bfbits2f(ah[j]), bfbits2f(bh[j]), acc);
    }
    const long tok = i / Hq;        // b*S + s
    const long h = i - tok * Hq;
    const long b_ = tok / S;
    const long s_ = tok - b_ * S;
    dsum[(b_ * Hq + h) * S + s_] = acc;
  }
}

// ========================================================================
// Backward dK/dV: grid over KV tiles (per kv-head); loops q tiles and the
// GQA q-head group, accumulating dK/dV in registers.
// ========================================================================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(NTHREADS) void attn_bwd_dkdv_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dsum,
    bf16_t* __restrict__ dk, bf16_t* __restrict__ dv, int B, int S, int Hq,
    int Hkv, float scale) {
  constexpr int DC = D / 32;
  constexpr int DT = D / 16;
  constexpr int KB = 64 * D * 2;
  // Q | QT | dO | dOT | per-wave scratch
  __shared__ __attribute__((aligned(16))) char smem[4 * KB + NW * 2048];
  char* q_lds = smem;
  char* qt_lds = smem + KB;
  char* do_lds = smem + 2 * KB;
  char* dot_lds = smem + 3 * KB;
  char* p_lds = smem + 4 * KB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int n0 = blockIdx.x * BN;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int rep = Hq / Hkv;

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* vp = v + ((long)b * S * kv_tok) + (long)hkv * D;
  bf16_t* dkp = dk + ((long)b * S * kv_tok) + (long)hkv * D;
  bf16_t* dvp = dv + ((long)b * S * kv_tok) + (long)hkv * D;

  // this wave's 16 kv rows: A-fragments of K and V
  const int nrow = n0 + wid * 16 + (lane & 15);
  bf16x8v k_frag[DC], v_frag[DC];
  load_afrags<D>(k_frag, kp, nrow, S, kv_tok, lane);
  load_afrags<D>(v_frag, vp, nrow, S, kv_tok, lane);

  f32x4 dk_acc[DT], dv_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) {
    dk_acc[t] = {0.f, 0.f, 0.f, 0.f};
    dv_acc[t] = {0.f, 0.f, 0.f, 0.f};
  }

  char* pw = p_lds + wid * 2048;
  const int m_start = CAUSAL ? (n0 / BM) * BM : 0;

  for (int g = 0; g < rep; ++g) {
    const int hq = hkv * rep + g;
    const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
    const bf16_t* dop = dout + ((long)b * S * q_tok) + (long)hq * D;
    const float* lsep = lse + ((long)b * Hq + hq) * S;
    const float* dsp = dsum + ((long)b * Hq + hq) * S;

    for (int m0 = m_start; m0 < S; m0 += BM) {
      load_tile_rm<D>(q_lds, qp, m0, S, q_tok);
      load_tile_tr<D>(qt_lds, qp, m0, S, q_tok);
      load_tile_rm<D>(do_lds, dop, m0, S, q_tok);
      load_tile_tr<D>(dot_lds, dop, m0, S, q_tok);
      __syncthreads();

      // S^T strip: rows = kv n (this wave's 16), cols = q m (64)
      f32x4 st[4], dpt[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        st[t] = {0.f, 0.f, 0.f, 0.f};
        dpt[t] = {0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int c = 0; c < DC; ++c) {
          bf16x8v qb = read_bfrag<D * 2>(
              q_lds, t * 16 + (lane & 15), c * 32 + (lane >> 4) * 8);
          st[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              k_frag[c], qb, st[t], 0, 0, 0);
          bf16x8v dob = read_bfrag<D * 2>(
              do_lds, t * 16 + (lane & 15), c * 32 + (lane >> 4) * 8);
          dpt[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              v_frag[c], dob, dpt[t], 0, 0, 0);
        }

      // P^T = exp(scale*S^T - lse[col]); dS^T = P^T*(dP^T - Dsum[col])
      const int nrow_base = n0 + wid * 16 + (lane >> 4) * 4;
      float pt[4][4], dst[4][4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int mcol = m0 + t * 16 + (lane & 15);
        const bool mvalid = mcol < S;
        const float lse_m = mvalid ? lsep[mcol] : 0.f;
        const float ds_m = mvalid ? dsp[mcol] : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int nr = nrow_base + r;
          float pv = 0.f;
          if (mvalid && nr < S && (!CAUSAL || mcol >= nr) &&
              lse_m != NEG_INF)
            pv = __expf(st[t][r] * scale - lse_m);
          pt[t][r] = pv;
          dst[t][r] = pv * (dpt[t][r] - ds_m);
        }
      }

      // dV += P^T * dO  (A = P^T via scratch, B = dOT)
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int rl = (lane >> 4) * 4 + r;
          const int cl = t * 16 + (lane & 15);
          *(uint16_t*)(pw + rl * 128 + swz(rl, cl * 2)) = f2bfbits(pt[t][r]);
        }
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8v pa = as_frag(*(const uint4*)(
            pw + (lane & 15) * 128 +
            swz(lane & 15, (kc * 32 + (lane >> 4) * 8) * 2)));
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v db = read_bfrag<128>(
              dot_lds, t * 16 + (lane & 15), kc * 32 + (lane >> 4) * 8);
          dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pa, db, dv_acc[t], 0, 0, 0);
        }
      }

      // dK += dS^T * Q  (A = dS^T via scratch, B = QT); scale folded later
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int rl = (lane >> 4) * 4 + r;
          const int cl = t * 16 + (lane & 15);
          *(uint16_t*)(pw + rl * 128 + swz(rl, cl * 2)) = f2bfbits(dst[t][r]);
        }
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8v da = as_frag(*(const uint4*)(
            pw + (lane & 15) * 128 +
            swz(lane & 15, (kc * 32 + (lane >> 4) * 8) * 2)));
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v qb = read_bfrag<128>(
              qt_lds, t * 16 + (lane & 15), kc * 32 + (lane >> 4) * 8);
          dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              da, qb, dk_acc[t], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

  // epilogue: write dK (scaled), dV
  const int nrow_base = n0 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = nrow_base + r;
    if (row >= S) continue;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      const int col = t * 16 + (lane & 15);
      dkp[(long)row * kv_tok + col] = f2bf(dk_acc[t][r] * scale);
      dvp[(long)row * kv_tok + col] = f2bf(dv_acc[t][r]);
    }
  }
}

// ========================================================================
// Backward dQ: grid over Q tiles; recompute S,P; dQ += scale * dS * K
// ========================================================================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(NTHREADS) void attn_bwd_dq_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dsum,
    bf16_t* __restrict__ dq, int B, int S, int Hq, int Hkv, float scale) {
  constexpr int DC = D / 32;
  constexpr int DT = D / 16;
  constexpr int KB = 64 * D * 2;
  // K | KT | V | per-wave scratch
  __shared__ __attribute__((aligned(16))) char smem[3 * KB + NW * 2048];
  char* k_lds = smem;
  char* kt_lds = smem + KB;
  char* v_lds = smem + 2 * KB;
  char* p_lds = smem + 3 * KB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int m0 = blockIdx.x * BM;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* vp = v + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* dop = dout + ((long)b * S * q_tok) + (long)hq * D;
  const float* lsep = lse + ((long)b * Hq + hq) * S;
  const float* dsp = dsum + ((long)b * Hq + hq) * S;
  bf16_t* dqp = dq + ((long)b * S * q_tok) + (long)hq * D;

  const int mrow = m0 + wid * 16 + (lane & 15);
  bf16x8v q_frag[DC], do_frag[DC];
  load_afrags<D>(q_frag, qp, mrow, S, q_tok, lane);
  load_afrags<D>(do_frag, dop, mrow, S, q_tok, lane);

  // per-row lse/Dsum (C-layout rows)
  const int row_base = m0 + wid * 16 + (lane >> 4) * 4;
  float lse_r[4], ds_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row_base + r;
    lse_r[r] = (row < S) ? lsep[row] : NEG_INF;
    ds_r[r] = (row < S) ? dsp[row] : 0.f;
  }

  f32x4 dq_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) dq_acc[t] = {0.f, 0.f, 0.f, 0.f};

  char* pw = p_lds + wid * 2048;
  const int n_end = CAUSAL ? min(S, m0 + BM) : S;
  for (int n0 = 0; n0 < n_end; n0 += BN) {
    load_tile_rm<D>(k_lds, kp, n0, S, kv_tok);
    load_tile_tr<D>(kt_lds, kp, n0, S, kv_tok);
    load_tile_rm<D>(v_lds, vp, n0, S, kv_tok);
    __syncthreads();

    const bool strip_live = !CAUSAL || (n0 <= m0 + wid * 16 + 15);
    if (strip_live) {
      f32x4 s_acc[4], dp_acc[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        s_acc[t] = {0.f, 0.f, 0.f, 0.f};
        dp_acc[t] = {0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int c = 0; c < DC; ++c) {
          bf16x8v kb = read_bfrag<D * 2>(
              k_lds, t * 16 + (lane & 15), c * 32 + (lane >> 4) * 8);
          s_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[c], kb, s_acc[t], 0, 0, 0);
          bf16x8v vb = read_bfrag<D * 2>(
              v_lds, t * 16 + (lane & 15), c * 32 + (lane >> 4) * 8);
          dp_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              do_frag[c], vb, dp_acc[t], 0, 0, 0);
        }

      // dS = P*(dP - Dsum[row]); P = exp(scale*S - lse[row])
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int col = n0 + t * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = row_base + r;
          float pv = 0.f;
          if (row < S && col < S && (!CAUSAL || col <= row) &&
              lse_r[r] != NEG_INF)
            pv = __expf(s_acc[t][r] * scale - lse_r[r]);
          const float dsv = pv * (dp_acc[t][r] - ds_r[r]);
          const int rl = (lane >> 4) * 4 + r;
          const int cl = t * 16 + (lane & 15);
          *(uint16_t*)(pw + rl * 128 + swz(rl, cl * 2)) = f2bfbits(dsv);
        }
      }
      // dQ += dS * K (A = dS via scratch, B = KT)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8v da = as_frag(*(const uint4*)(
            pw + (lane & 15) * 128 +
            swz(lane & 15, (kc * 32 + (lane >> 4) * 8) * 2)));
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v ktb = read_bfrag<128>(
              kt_lds, t * 16 + (lane & 15), kc * 32 + (lane >> 4) * 8);
          dq_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              da, ktb, dq_acc[t], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row_base + r;
    if (row >= S) continue;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      const int col = t * 16 + (lane & 15);
      dqp[(long)row * q_tok + col] = f2bf(dq_acc[t][r] * scale);
    }
  }
}

}  // namespace

extern "C" {

hipError_t tok_attn_fwd(const void* q, const void* k, const void* v, void* o,
                        float* lse, int B, int S, int Hq, int Hkv, int D,
                        int causal, hipStream_t stream) {
  dim3 grid((S + BM - 1) / BM, Hq, B);
  const float scale = 1.f / sqrtf((float)D);
#define LAUNCH_FWD(DD, CC)                                                    \
  attn_fwd_kernel<DD, CC><<<grid, NTHREADS, 0, stream>>>(                     \
      (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)o, lse, \
      B, S, Hq, Hkv, scale)
  if (D == 128) { if (causal) LAUNCH_FWD(128, true); else LAUNCH_FWD(128, false); }
  else if (D == 64) { if (causal) LAUNCH_FWD(64, true); else LAUNCH_FWD(64, false); }
  else return hipErrorInvalidValue;
#undef LAUNCH_FWD
  return hipGetLastError();
}

hipError_t tok_attn_bwd(const void* q, const void* k, const void* v,
                        const void* o, const void* dout, const float* lse,
                        float* dsum_ws, void* dq, void* dk, void* dv, int B,
                        int S, int Hq, int Hkv, int D, int causal,
                        hipStream_t stream) {
  const float scale = 1.f / sqrtf((float)D);
  const long rows = (long)B * S * Hq;
  long pgrid = (rows + 255) / 256;
  if (pgrid > 4096) pgrid = 4096;
  if (D == 128)
    attn_bwd_pre_kernel<128><<<(int)pgrid, 256, 0, stream>>>(
        (const bf16_t*)dout, (const bf16_t*)o, dsum_ws, rows, Hq, S);
  else if (D == 64)
    attn_bwd_pre_kernel<64><<<(int)pgrid, 256, 0, stream>>>(
        (const bf16_t*)dout, (const bf16_t*)o, dsum_ws, rows, Hq, S);
  else
    return hipErrorInvalidValue;

  dim3 gkv((S + BN - 1) / BN, Hkv, B);
  dim3 gq((S + BM - 1) / BM, Hq, B);
#define LAUNCH_BWD(DD, CC)                                                    \
  do {                                                                        \
    attn_bwd_dkdv_kernel<DD, CC><<<gkv, NTHREADS, 0, stream>>>(               \
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,                 \
        (const bf16_t*)dout, lse, dsum_ws, (bf16_t*)dk, (bf16_t*)dv, B, S,   \
        Hq, Hkv, scale);                                                      \
    attn_bwd_dq_kernel<DD, CC><<<gq, NTHREADS, 0, stream>>>(                  \
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,                 \
        (const bf16_t*)dout, lse, dsum_ws, (bf16_t*)dq, B, S, Hq, Hkv,       \
        scale);                                                               \
  } while (0)
  if (D == 128) { if (causal) LAUNCH_BWD(128, true); else LAUNCH_BWD(128, false); }
  else { if (causal) LAUNCH_BWD(64, true); else LAUNCH_BWD(64, false); }
#undef LAUNCH_BWD
  return hipGetLastError();
}
}
