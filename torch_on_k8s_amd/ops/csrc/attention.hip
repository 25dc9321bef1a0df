// Flash attention (causal, GQA/MHA) for MI355X (gfx950) - fwd + bwd.
//
// Architecture (v3; progression + measurements in profiles/README.md):
//  * mfma_f32_32x32x16_bf16 tiles with SWAPPED operands: S^T = K*Q^T so
//    each lane owns ONE q row; online softmax is fully in-register (an
//    in-lane chain + one half-wave shuffle; no cross-lane reduce, no P
//    round trip through LDS). Fragment layouts verified on hardware by
//    mfma_probe.hip.
//  * P / dS^T are repacked from the MFMA C-layout to the next MFMA's
//    A-layout with v_cvt_pk_bf16_f32 pairs + permlane32_swap (guide T12):
//    C-layout element r of half h sits at k=(r&3)+8*(r>>2)+4h; one swap
//    fixes two A-fragment words.
//  * Q (fwd), K/V (dq) and K (dv/dk) live in registers as the MFMA
//    B-operand; tiles consumed d-major (V fwd, K^T/Q^T/dO^T bwd) are
//    read from PRE-TRANSPOSED [B,H,D,S_pad] planes (transpose.hip) so
//    all LDS staging is vectorized b128 writes - the in-kernel scalar
//    transpose scatter measured ~20 LDS bank-conflict cycles per MFMA.
//  * LDS tiles are double-buffered (one barrier per tile; the next
//    tile's write overlaps other waves' compute) and XOR-swizzled on the
//    FULL tile offset ((row&15)<<4: conflict-free 16-lane b128 groups;
//    bit 7 swaps row parity in-tile for 128-B rows).
//  * async-stage prefetch (T14), s_setprio around MFMA clusters (T5),
//    defer-max online softmax (T13), interior tiles skip masking.
//  * Backward: flash recompute scheme in three kernels - dV and dK split
//    (a combined kernel needs 472 regs = 1 wave/SIMD and measured
//    slower; split keeps 2 waves/SIMD), dQ over q tiles, plus a
//    Dsum = rowsum(dO*O) preprocess. All kernels: 0 scratch spill.
//
// Measured (B=2 S=4096 Hq=32 Hkv=8 D=128 causal, random data): fwd
// 358-371 TF (parity with torch's aotriton flash), bwd 204-206 TF incl.
// the three operand transposes (r2: uniform-flag interior fast path —
// the bwd kernels are VALU-overhead-bound, profiles/pmc_attention_r2.csv).
#include "common.h"

namespace {

using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BN = 64;    // kv rows per LDS tile (fwd/dq); q rows in dkdv
constexpr int NW = 8;     // waves per block
constexpr int NTHREADS = NW * WAVE;   // 512
constexpr int BM = NW * 16;           // 128 rows owned per block (1 strip/wave)
constexpr float NEG_INF = -INFINITY;

// Swizzled tile offset: (row*RB + col) ^ ((row&15)<<4) spreads a 16-lane
// ds_read_b128 group over 16 16-B slots (conflict-free when the group's
// rows are distinct mod 16, guide §6 G4). The XOR is applied to the FULL
// tile offset: for 128-B rows bit 7 of the mask swaps row parity, which
// keeps the mapping a bijection INSIDE the tile (XOR-ing only the
// byte-in-row would escape the last rows' allocation).
DEVINL int swz_off(int row, int row_bytes, int byte_in_row) {
  return (row * row_bytes + byte_in_row) ^ ((row & 15) << 4);
}

DEVINL bf16x8v as_frag(uint4 raw) {
  union { uint4 u; bf16x8v f; } c;
  c.u = raw;
  return c.f;
}

// ---- two-phase tile staging (async-STAGE split) -----------------------
// A [64][D] bf16 tile staged by all 512 threads: issue() starts the
// global loads into registers; write_rm()/write_tr() put them into LDS
// (row-major swizzled / transposed swizzled). OOB rows load zero.
template <int D, int NTH = NTHREADS>
struct TileStage {
  static constexpr int VPR = D / 8;              // 16B vecs per row
  static constexpr int VPT = 64 * VPR / NTH;     // vecs per thread
  uint4 vals[VPT];

  DEVINL void issue(const bf16_t* src, long row0, long row_limit,
                    long tok_stride) {
#pragma unroll
    for (int i = 0; i < VPT; ++i) {
      const int vi = threadIdx.x + i * NTH;
      const int row = vi / VPR, cv = vi % VPR;
      uint4 val = {0, 0, 0, 0};
      if (row0 + row < row_limit)
        val = *(const uint4*)(src + (row0 + row) * tok_stride + cv * 8);
      vals[i] = val;
    }
  }

  DEVINL void write_rm(char* lds) const {
#pragma unroll
    for (int i = 0; i < VPT; ++i) {
      const int vi = threadIdx.x + i * NTH;
      const int row = vi / VPR, cv = vi % VPR;
      *(uint4*)(lds + swz_off(row, D * 2, cv * 16)) = vals[i];
    }
  }

};

// Loads a [D][64] tile (rows = d, cols = 64 tokens) from a
// pre-transposed [D][S_pad] plane (transpose.hip) into swizzled LDS with
// plain b128 writes — replaces the in-kernel scalar transpose scatter
// that measured ~20 LDS bank-conflict cycles per MFMA.
template <int D, int NTH = NTHREADS>
struct TileStageT {
  static constexpr int VPR = 64 / 8;
  static constexpr int VPT = D * VPR / NTH;
  uint4 vals[VPT];

  DEVINL void issue(const bf16_t* plane, int col0, long S_pad) {
#pragma unroll
    for (int i = 0; i < VPT; ++i) {
      const int vi = threadIdx.x + i * NTH;
      const int row = vi / VPR, cv = vi % VPR;
      vals[i] = *(const uint4*)(plane + (long)row * S_pad + col0 + cv * 8);
    }
  }

  DEVINL void write(char* lds) const {
#pragma unroll
    for (int i = 0; i < VPT; ++i) {
      const int vi = threadIdx.x + i * NTH;
      const int row = vi / VPR, cv = vi % VPR;
      *(uint4*)(lds + swz_off(row, 128, cv * 16)) = vals[i];
    }
  }
};

// B-fragment read from a swizzled row-major LDS tile, row stride RB bytes.
template <int RB>
DEVINL bf16x8v read_bfrag(const char* lds, int row, int col_elem) {
  return as_frag(*(const uint4*)(lds + swz_off(row, RB, col_elem * 2)));
}

// A-fragment set (rows m = lane&15 of a 16-row strip) from global.
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
// Forward v3: grid (ceil(S/256), Hq, B); block = 8 waves, wave = 32 q rows.
//
// Swapped QK^T on mfma_f32_32x32x16_bf16: compute S^T = K·Q^T so each
// lane owns ONE q row (lane&31) and its P values stay in-register —
// row reduce is an in-lane chain + one half-wave exchange; no P LDS
// round trip. P is repacked to the PV A-fragment layout with
// v_cvt_pk_bf16_f32 pairs + permlane32_swap (guide T12):
//   C-layout kv of reg r (half h): (r&3) + 8*(r>>2) + 4*h
//   A-layout needs kv = h*8 + j in each 16-slot; one swap fixes two words.
// Q lives in registers as the MFMA B-operand (B[k][n]: n = lane&31 = its
// q row, k contiguous per half) — loaded once per wave.
// ========================================================================
using f32x16 = __attribute__((ext_vector_type(16))) float;

DEVINL uint32_t pack_bf16(float lo, float hi) {
  return (uint32_t)f2bfbits(lo) | ((uint32_t)f2bfbits(hi) << 16);
}

template <int D, bool CAUSAL>
__global__ __launch_bounds__(NTHREADS) void attn_fwd_kernel_v3(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ vt, bf16_t* __restrict__ o,
    float* __restrict__ lse, int B, int S, int Hq, int Hkv, int S_pad,
    float scale) {
  constexpr int QC = D / 16;    // Q B-fragments (k-slots of 16)
  constexpr int DT = D / 32;    // O col tiles of 32
  constexpr int BM3 = NW * 32;  // 256 q rows per block
  constexpr int KB = BN * D * 2;
  // double-buffered K|VT: one barrier per tile; tile t+1's LDS write
  // overlaps other waves' compute on tile t
  __shared__ __attribute__((aligned(16))) char smem[2 * (2 * KB)];

  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int wid = threadIdx.x >> 6;
  const int m0 = blockIdx.x * BM3;
  const int m0w = m0 + wid * 32;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* vtp = vt + ((long)b * Hkv + hkv) * D * (long)S_pad;
  bf16_t* op = o + ((long)b * S * q_tok) + (long)hq * D;
  float* lsep = lse + ((long)b * Hq + hq) * S;

  const int qrow = m0w + (lane & 31);
  bf16x8v q_reg[QC];
#pragma unroll
  for (int c = 0; c < QC; ++c) {
    uint4 raw = {0, 0, 0, 0};
    if (qrow < S)
      raw = *(const uint4*)(qp + (long)qrow * q_tok + c * 16 + hi * 8);
    q_reg[c] = as_frag(raw);
  }

  f32x16 o_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[t][r] = 0.f;
  float m_run = NEG_INF, l_run = 0.f;

  TileStage<D> k_st;
  TileStageT<D> v_st;
  const int n_end = CAUSAL ? min(S, m0 + BM3) : S;
  k_st.issue(kp, 0, S, kv_tok);
  v_st.issue(vtp, 0, S_pad);
  k_st.write_rm(smem);
  v_st.write(smem + KB);
  __syncthreads();
  int cur = 0;

  for (int n0 = 0; n0 < n_end; n0 += BN) {
    char* k_lds = smem + cur * (2 * KB);
    char* vt_lds = k_lds + KB;
    const bool more = n0 + BN < n_end;
    if (more) {
      k_st.issue(kp, n0 + BN, S, kv_tok);
      v_st.issue(vtp, n0 + BN, S_pad);
    }

    const bool strip_live = !CAUSAL || (n0 <= m0w + 31);
    if (strip_live) {
      // S^T strips (two 32-kv subtiles x 32 q cols)
      f32x16 st[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int r = 0; r < 16; ++r) st[ks][r] = 0.f;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int c = 0; c < QC; ++c) {
          bf16x8v ka = read_bfrag<D * 2>(
              k_lds, ks * 32 + (lane & 31), c * 16 + hi * 8);
          st[ks] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              ka, q_reg[c], st[ks], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);

      // mask + scale + in-lane row max; interior tiles (fully below the
      // diagonal and inside S) skip the per-element mask entirely
      const bool needs_mask = (n0 + BN > S) ||
                              (CAUSAL && (n0 + BN - 1 > m0w));
      float mx = NEG_INF;
      if (needs_mask) {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kvg = n0 + ks * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
            float s = st[ks][r] * scale;
            if ((CAUSAL && kvg > qrow) || kvg >= S || qrow >= S) s = NEG_INF;
            st[ks][r] = s;
            mx = fmaxf(mx, s);
          }
      } else {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float s = st[ks][r] * scale;
            st[ks][r] = s;
            mx = fmaxf(mx, s);
          }
      }
      mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
      // defer-max (guide T13): when every row's max grew by < THR keep
      // the old running max and skip the O rescale; P is then bounded by
      // e^THR which the fp32 accumulator tolerates. Nothing of this tile
      // is pending when the branch is taken (P*V of tile t completes
      // before tile t+1's decision), so the T13 ordering hazard cannot
      // occur in this structure.
      constexpr float DEFER_THR = 8.f;
      const bool rescale = !__all(mx <= m_run + DEFER_THR);
      float alpha = 1.f;
      if (rescale) {
        const float mn = fmaxf(m_run, mx);
        alpha = (m_run == NEG_INF) ? 0.f : __expf(m_run - mn);
        m_run = mn;
      }
      float rsum = 0.f;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float pe =
              (st[ks][r] == NEG_INF) ? 0.f : __expf(st[ks][r] - m_run);
          st[ks][r] = pe;
          rsum += pe;
        }
      rsum += __shfl_xor(rsum, 32, 64);
      l_run = l_run * alpha + rsum;

      // rescale O by the per-ROW alpha (rows are reg-mapped; alpha is
      // lane-mapped -> one bpermute gather per reg row); skipped
      // entirely on the defer path
      if (rescale) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int rowidx = (r & 3) + 8 * (r >> 2) + 4 * hi;
          const float ar = __shfl(alpha, rowidx, 64);
#pragma unroll
          for (int t = 0; t < DT; ++t) o_acc[t][r] *= ar;
        }
      }

      // P (f32, C-layout) -> bf16 A-fragments via cvt_pk + permlane swap
      bf16x8v pa[4];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        union { uint32_t w[4]; bf16x8v f; } f0, f1;
        {
          auto sA = __builtin_amdgcn_permlane32_swap(
              pack_bf16(st[ks][0], st[ks][1]),
              pack_bf16(st[ks][4], st[ks][5]), false, false);
          auto sB = __builtin_amdgcn_permlane32_swap(
              pack_bf16(st[ks][2], st[ks][3]),
              pack_bf16(st[ks][6], st[ks][7]), false, false);
          f0.w[0] = sA[0]; f0.w[1] = sB[0]; f0.w[2] = sA[1]; f0.w[3] = sB[1];
        }
        {
          auto sC = __builtin_amdgcn_permlane32_swap(
              pack_bf16(st[ks][8], st[ks][9]),
              pack_bf16(st[ks][12], st[ks][13]), false, false);
          auto sD = __builtin_amdgcn_permlane32_swap(
              pack_bf16(st[ks][10], st[ks][11]),
              pack_bf16(st[ks][14], st[ks][15]), false, false);
          f1.w[0] = sC[0]; f1.w[1] = sD[0]; f1.w[2] = sC[1]; f1.w[3] = sD[1];
        }
        pa[2 * ks] = f0.f;
        pa[2 * ks + 1] = f1.f;
      }

      // O += P * V
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 4; ++ks)
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v vb = read_bfrag<128>(
              vt_lds, t * 32 + (lane & 31), ks * 16 + hi * 8);
          o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa[ks], vb, o_acc[t], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    if (more) {
      char* nk = smem + (cur ^ 1) * (2 * KB);
      k_st.write_rm(nk);
      v_st.write(nk + KB);
    }
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: O = o_acc / l (per reg row); LSE per lane row
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int rowidx = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const float lv = __shfl(l_run, rowidx, 64);
    const float inv_l = (lv > 0.f) ? 1.f / lv : 0.f;
    const int row = m0w + rowidx;
    if (row >= S) continue;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      const int col = t * 32 + (lane & 31);
      op[(long)row * q_tok + col] = f2bf(o_acc[t][r] * inv_l);
    }
  }
  if (hi == 0 && qrow < S)
    lsep[qrow] = (l_run > 0.f) ? m_run + __logf(l_run) : NEG_INF;
}

// ========================================================================
// Backward preprocess: Dsum[b,h,m] = sum_d dO*O (fp32)
// ========================================================================
template <int D>
__global__ void attn_bwd_pre_kernel(const bf16_t* __restrict__ dout,
                                    const bf16_t* __restrict__ o,
                                    float* __restrict__ dsum, long rows,
                                    int Hq, int S) {
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
    const long tok = i / Hq;
    const long h = i - tok * Hq;
    const long b_ = tok / S;
    const long s_ = tok - b_ * S;
    dsum[(b_ * Hq + h) * S + s_] = acc;
  }
}

// ========================================================================
// Backward dV / dK v3: two kernels, grid (ceil(S/256), Hkv, B), 8 waves,
// wave = 32 kv rows. Splitting dV and dK keeps each kernel at 2
// waves/SIMD (a combined kernel needs dk+dv accumulators = 128 regs and
// drops to 1 wave/SIMD, which measured SLOWER than the 16x16 v2 —
// occupancy beats the saved recompute here). Swapped-operand structure
// as the forward: K (and V in the dK kernel) live in registers as MFMA
// B-operands; P^T / dS^T stay in-register and are repacked with
// cvt_pk+permlane32_swap.
// ========================================================================
DEVINL void repack_pa(const f32x16& pt, bf16x8v& pa0, bf16x8v& pa1) {
  union { uint32_t w[4]; bf16x8v f; } f0, f1;
  auto sA = __builtin_amdgcn_permlane32_swap(
      pack_bf16(pt[0], pt[1]), pack_bf16(pt[4], pt[5]), false, false);
  auto sB = __builtin_amdgcn_permlane32_swap(
      pack_bf16(pt[2], pt[3]), pack_bf16(pt[6], pt[7]), false, false);
  f0.w[0] = sA[0]; f0.w[1] = sB[0]; f0.w[2] = sA[1]; f0.w[3] = sB[1];
  auto sC = __builtin_amdgcn_permlane32_swap(
      pack_bf16(pt[8], pt[9]), pack_bf16(pt[12], pt[13]), false, false);
  auto sD = __builtin_amdgcn_permlane32_swap(
      pack_bf16(pt[10], pt[11]), pack_bf16(pt[14], pt[15]), false, false);
  f1.w[0] = sC[0]; f1.w[1] = sD[0]; f1.w[2] = sC[1]; f1.w[3] = sD[1];
  pa0 = f0.f;
  pa1 = f1.f;
}

// NWT (waves per block) A/B: 8 = one block/CU (lockstep phases);
// 4 = TWO resident blocks/CU whose barrier phases drift, so each SIMD
// hosts one wave of each block and staging overlaps the partner
// block's MFMA segments (guide's compute/load wave-pair regime).
template <int D, bool CAUSAL, int NWT = NW>
__global__ __launch_bounds__(NWT * WAVE) void attn_bwd_dv_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ dot_t, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dsum,
    bf16_t* __restrict__ dv, int B, int S, int Hq, int Hkv, int S_pad,
    float scale) {
  constexpr int QC = D / 16;
  constexpr int DT = D / 32;
  constexpr int NTH = NWT * WAVE;
  constexpr int BNK = NWT * 32;  // kv rows per block
  constexpr int KB = BN * D * 2;
  // double-buffered (Q rm | dOT): one barrier per q tile
  __shared__ __attribute__((aligned(16))) char smem[2 * (2 * KB)];

  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int wid = threadIdx.x >> 6;
  const int n0 = blockIdx.x * BNK;
  const int n0w = n0 + wid * 32;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int rep = Hq / Hkv;

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  bf16_t* dvp = dv + ((long)b * S * kv_tok) + (long)hkv * D;

  const int kvrow = n0w + (lane & 31);
  bf16x8v k_reg[QC];
#pragma unroll
  for (int c = 0; c < QC; ++c) {
    uint4 kr = {0, 0, 0, 0};
    if (kvrow < S)
      kr = *(const uint4*)(kp + (long)kvrow * kv_tok + c * 16 + hi * 8);
    k_reg[c] = as_frag(kr);
  }

  f32x16 dv_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dv_acc[t][r] = 0.f;

  const int m_start = CAUSAL ? (n0 / BN) * BN : 0;
  const long dplane = (long)D * S_pad;
  TileStage<D, NTH> q_st;
  TileStageT<D, NTH> do_st;
  const bf16_t* dot0 = dot_t + ((long)b * Hq + hkv * rep) * dplane;
  q_st.issue(q + ((long)b * S * q_tok) + (long)(hkv * rep) * D, m_start, S,
             q_tok);
  do_st.issue(dot0, m_start, S_pad);
  q_st.write_rm(smem);
  do_st.write(smem + KB);
  __syncthreads();
  int cur = 0;

  for (int g = 0; g < rep; ++g) {
    const int hq = hkv * rep + g;
    const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
    const bf16_t* dotp = dot_t + ((long)b * Hq + hq) * dplane;
    const float* lsep = lse + ((long)b * Hq + hq) * S;

    for (int m0 = m_start; m0 < S; m0 += BN) {
      char* q_lds = smem + cur * (2 * KB);
      char* dot_lds = q_lds + KB;
      const int m1 = m0 + BN;
      const bool more = m1 < S || g + 1 < rep;
      if (m1 < S) {
        q_st.issue(qp, m1, S, q_tok);
        do_st.issue(dotp, m1, S_pad);
      } else if (g + 1 < rep) {
        q_st.issue(qp + D, m_start, S, q_tok);
        do_st.issue(dotp + dplane, m_start, S_pad);
      }

#pragma unroll
      for (int qs = 0; qs < 2; ++qs) {
        f32x16 sv;
#pragma unroll
        for (int r = 0; r < 16; ++r) sv[r] = 0.f;
#pragma unroll
        for (int c = 0; c < QC; ++c) {
          bf16x8v qa = read_bfrag<D * 2>(
              q_lds, qs * 32 + (lane & 31), c * 16 + hi * 8);
          sv = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              qa, k_reg[c], sv, 0, 0, 0);
        }
        const int qg_lane = m0 + qs * 32 + (lane & 31);
        const float lse_lane = (qg_lane < S) ? lsep[qg_lane] : NEG_INF;
        f32x16 pt;
        // interior subtile: all q rows >= every kv row of the wave and
        // in-bounds -> the per-element mask chain is dead VALU work
        // (PMC r2: bwd is ~11 VALU/MFMA). One loop, uniform flag: the
        // exp stays common so the compiler shares registers.
        const bool need_mask = (CAUSAL && m0 + qs * 32 < n0w + 32) ||
                               (m0 + qs * 32 + 32 > S) || (n0w + 32 > S);
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int rowidx = (r & 3) + 8 * (r >> 2) + 4 * hi;
          const float lse_r = __shfl(lse_lane, rowidx, 64);
          float pv = __expf(sv[r] * scale - lse_r);
          if (need_mask) {
            const int qg = m0 + qs * 32 + rowidx;
            if (!(qg < S && kvrow < S && (!CAUSAL || qg >= kvrow) &&
                  lse_r != NEG_INF))
              pv = 0.f;
          }
          pt[r] = pv;
        }
        bf16x8v pa0, pa1;
        repack_pa(pt, pa0, pa1);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v b0 = read_bfrag<128>(
              dot_lds, t * 32 + (lane & 31), qs * 32 + hi * 8);
          dv_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa0, b0, dv_acc[t], 0, 0, 0);
          bf16x8v b1 = read_bfrag<128>(
              dot_lds, t * 32 + (lane & 31), qs * 32 + 16 + hi * 8);
          dv_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa1, b1, dv_acc[t], 0, 0, 0);
        }
      }
      if (more) {
        char* nq = smem + (cur ^ 1) * (2 * KB);
        q_st.write_rm(nq);
        do_st.write(nq + KB);
      }
      __syncthreads();
      cur ^= 1;
    }
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = n0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (row >= S) continue;
#pragma unroll
    for (int t = 0; t < DT; ++t)
      dvp[(long)row * kv_tok + t * 32 + (lane & 31)] = f2bf(dv_acc[t][r]);
  }
}

template <int D, bool CAUSAL, int NWT = NW>
__global__ __launch_bounds__(NWT * WAVE) void attn_bwd_dk_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const bf16_t* __restrict__ q_t, const float* __restrict__ lse,
    const float* __restrict__ dsum, bf16_t* __restrict__ dk, int B, int S,
    int Hq, int Hkv, int S_pad, float scale) {
  constexpr int QC = D / 16;
  constexpr int DT = D / 32;
  constexpr int NTH = NWT * WAVE;
  constexpr int BNK = NWT * 32;
  constexpr int KB = BN * D * 2;
  // Q rm | QT | dO rm | V block tile (staged once; keeping V in
  // registers alongside K pushed the kernel to 256 VGPR + scratch spill)
  __shared__ __attribute__((aligned(16))) char smem[3 * KB + BNK * D * 2];
  char* q_lds = smem;
  char* qt_lds = smem + KB;
  char* do_lds = smem + 2 * KB;
  char* v_lds = smem + 3 * KB;

  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int wid = threadIdx.x >> 6;
  const int n0 = blockIdx.x * BNK;
  const int n0w = n0 + wid * 32;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int rep = Hq / Hkv;

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* vp = v + ((long)b * S * kv_tok) + (long)hkv * D;
  bf16_t* dkp = dk + ((long)b * S * kv_tok) + (long)hkv * D;

  const int kvrow = n0w + (lane & 31);
  bf16x8v k_reg[QC];
#pragma unroll
  for (int c = 0; c < QC; ++c) {
    uint4 kr = {0, 0, 0, 0};
    if (kvrow < S)
      kr = *(const uint4*)(kp + (long)kvrow * kv_tok + c * 16 + hi * 8);
    k_reg[c] = as_frag(kr);
  }
  {  // stage the block's 256-row V tile (row-major, swizzled) once
    constexpr int VPR = D / 8;
#pragma unroll 4
    for (int vi = threadIdx.x; vi < BNK * VPR; vi += NTH) {
      const int row = vi / VPR, cv = vi % VPR;
      uint4 val = {0, 0, 0, 0};
      if (n0 + row < S)
        val = *(const uint4*)(vp + (long)(n0 + row) * kv_tok + cv * 8);
      *(uint4*)(v_lds + swz_off(row, D * 2, cv * 16)) = val;
    }
  }

  f32x16 dk_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dk_acc[t][r] = 0.f;

  const int m_start = CAUSAL ? (n0 / BN) * BN : 0;

  for (int g = 0; g < rep; ++g) {
    const int hq = hkv * rep + g;
    const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
    const bf16_t* dop = dout + ((long)b * S * q_tok) + (long)hq * D;
    const float* lsep = lse + ((long)b * Hq + hq) * S;
    const float* dsp = dsum + ((long)b * Hq + hq) * S;

    const bf16_t* qtp = q_t + ((long)b * Hq + hq) * (long)D * S_pad;
    for (int m0 = m_start; m0 < S; m0 += BN) {
      {  // synchronous staging: the prefetch ring costs ~16 VGPRs and
         // tips this kernel into scratch spill, which is worse
        TileStage<D, NTH> q_st, do_st;
        TileStageT<D, NTH> qt_st;
        q_st.issue(qp, m0, S, q_tok);
        do_st.issue(dop, m0, S, q_tok);
        qt_st.issue(qtp, m0, S_pad);
        q_st.write_rm(q_lds);
        qt_st.write(qt_lds);
        do_st.write_rm(do_lds);
      }
      __syncthreads();

#pragma unroll
      for (int qs = 0; qs < 2; ++qs) {
        f32x16 sv, dpv;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          sv[r] = 0.f;
          dpv[r] = 0.f;
        }
#pragma unroll
        for (int c = 0; c < QC; ++c) {
          bf16x8v qa = read_bfrag<D * 2>(
              q_lds, qs * 32 + (lane & 31), c * 16 + hi * 8);
          sv = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              qa, k_reg[c], sv, 0, 0, 0);
          bf16x8v doa = read_bfrag<D * 2>(
              do_lds, qs * 32 + (lane & 31), c * 16 + hi * 8);
          bf16x8v vb = read_bfrag<D * 2>(
              v_lds, wid * 32 + (lane & 31), c * 16 + hi * 8);
          dpv = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              doa, vb, dpv, 0, 0, 0);
        }
        const int qg_lane = m0 + qs * 32 + (lane & 31);
        const float lse_lane = (qg_lane < S) ? lsep[qg_lane] : NEG_INF;
        const float ds_lane = (qg_lane < S) ? dsp[qg_lane] : 0.f;
        f32x16 dst;
        // interior fast path, one loop + uniform flag (see dv kernel)
        const bool need_mask = (CAUSAL && m0 + qs * 32 < n0w + 32) ||
                               (m0 + qs * 32 + 32 > S) || (n0w + 32 > S);
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int rowidx = (r & 3) + 8 * (r >> 2) + 4 * hi;
          const float lse_r = __shfl(lse_lane, rowidx, 64);
          const float ds_r = __shfl(ds_lane, rowidx, 64);
          float pv = __expf(sv[r] * scale - lse_r);
          if (need_mask) {
            const int qg = m0 + qs * 32 + rowidx;
            if (!(qg < S && kvrow < S && (!CAUSAL || qg >= kvrow) &&
                  lse_r != NEG_INF))
              pv = 0.f;
          }
          dst[r] = pv * (dpv[r] - ds_r);
        }
        bf16x8v da0, da1;
        repack_pa(dst, da0, da1);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v b0 = read_bfrag<128>(
              qt_lds, t * 32 + (lane & 31), qs * 32 + hi * 8);
          dk_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              da0, b0, dk_acc[t], 0, 0, 0);
          bf16x8v b1 = read_bfrag<128>(
              qt_lds, t * 32 + (lane & 31), qs * 32 + 16 + hi * 8);
          dk_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              da1, b1, dk_acc[t], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = n0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (row >= S) continue;
#pragma unroll
    for (int t = 0; t < DT; ++t)
      dkp[(long)row * kv_tok + t * 32 + (lane & 31)] =
          f2bf(dk_acc[t][r] * scale);
  }
}

// ========================================================================
// Backward dQ v3: grid (ceil(S/256), Hq, B); 8 waves, wave = 32 q rows.
// Q/dO in registers (lane owns q row); lse/Dsum are per-lane scalars.
// ========================================================================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(NTHREADS) void attn_bwd_dq_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const bf16_t* __restrict__ k_t, const float* __restrict__ lse,
    const float* __restrict__ dsum, bf16_t* __restrict__ dq, int B, int S,
    int Hq, int Hkv, int S_pad, float scale) {
  constexpr int QC = D / 16;
  constexpr int DT = D / 32;
  constexpr int BM3 = NW * 32;
  constexpr int KB = BN * D * 2;
  // double-buffered (K rm | KT | V rm): one barrier per kv tile
  __shared__ __attribute__((aligned(16))) char smem[2 * (3 * KB)];

  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int wid = threadIdx.x >> 6;
  const int m0 = blockIdx.x * BM3;
  const int m0w = m0 + wid * 32;
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);

  const long q_tok = (long)Hq * D, kv_tok = (long)Hkv * D;
  const bf16_t* qp = q + ((long)b * S * q_tok) + (long)hq * D;
  const bf16_t* kp = k + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* vp = v + ((long)b * S * kv_tok) + (long)hkv * D;
  const bf16_t* dop = dout + ((long)b * S * q_tok) + (long)hq * D;
  const bf16_t* ktp = k_t + ((long)b * Hkv + hkv) * (long)D * S_pad;
  const float* lsep = lse + ((long)b * Hq + hq) * S;
  const float* dsp = dsum + ((long)b * Hq + hq) * S;
  bf16_t* dqp = dq + ((long)b * S * q_tok) + (long)hq * D;

  const int qrow = m0w + (lane & 31);
  bf16x8v q_reg[QC], do_reg[QC];
#pragma unroll
  for (int c = 0; c < QC; ++c) {
    uint4 qr = {0, 0, 0, 0}, dor = {0, 0, 0, 0};
    if (qrow < S) {
      qr = *(const uint4*)(qp + (long)qrow * q_tok + c * 16 + hi * 8);
      dor = *(const uint4*)(dop + (long)qrow * q_tok + c * 16 + hi * 8);
    }
    q_reg[c] = as_frag(qr);
    do_reg[c] = as_frag(dor);
  }
  const float lse_lane = (qrow < S) ? lsep[qrow] : NEG_INF;
  const float ds_lane = (qrow < S) ? dsp[qrow] : 0.f;

  f32x16 dq_acc[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[t][r] = 0.f;

  TileStage<D> k_st, v_st;
  TileStageT<D> kt_st;
  const int n_end = CAUSAL ? min(S, m0 + BM3) : S;
  k_st.issue(kp, 0, S, kv_tok);
  v_st.issue(vp, 0, S, kv_tok);
  kt_st.issue(ktp, 0, S_pad);
  k_st.write_rm(smem);
  kt_st.write(smem + KB);
  v_st.write_rm(smem + 2 * KB);
  __syncthreads();
  int cur = 0;

  for (int n0 = 0; n0 < n_end; n0 += BN) {
    char* k_lds = smem + cur * (3 * KB);
    char* kt_lds = k_lds + KB;
    char* v_lds = k_lds + 2 * KB;
    const bool more = n0 + BN < n_end;
    if (more) {
      k_st.issue(kp, n0 + BN, S, kv_tok);
      v_st.issue(vp, n0 + BN, S, kv_tok);
      kt_st.issue(ktp, n0 + BN, S_pad);
    }

    const bool strip_live = !CAUSAL || (n0 <= m0w + 31);
    if (strip_live) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {   // two 32-kv subtiles
        f32x16 sv, dpv;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          sv[r] = 0.f;
          dpv[r] = 0.f;
        }
#pragma unroll
        for (int c = 0; c < QC; ++c) {
          bf16x8v ka = read_bfrag<D * 2>(
              k_lds, ks * 32 + (lane & 31), c * 16 + hi * 8);
          sv = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              ka, q_reg[c], sv, 0, 0, 0);
          bf16x8v va = read_bfrag<D * 2>(
              v_lds, ks * 32 + (lane & 31), c * 16 + hi * 8);
          dpv = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              va, do_reg[c], dpv, 0, 0, 0);
        }

        f32x16 dst;
        // interior subtile: every (q, kv) pair of this wave is strictly
        // causal-valid and in-bounds -> the per-element mask chain is
        // dead VALU work (PMC r2: bwd is ~11 VALU/MFMA). One loop with
        // a uniform flag keeps the register footprint shared.
        const bool need_mask = (CAUSAL && n0 + ks * 32 + 32 > m0w) ||
                               (n0 + ks * 32 + 32 > S) || (m0w + 32 > S);
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float pv = __expf(sv[r] * scale - lse_lane);
          if (need_mask) {
            const int kvg = n0 + ks * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
            if (!(qrow < S && kvg < S && (!CAUSAL || kvg <= qrow) &&
                  lse_lane != NEG_INF))
              pv = 0.f;
          }
          dst[r] = pv * (dpv[r] - ds_lane);
        }

        bf16x8v da0, da1;
        {
          union { uint32_t w[4]; bf16x8v f; } f0, f1;
          auto sA = __builtin_amdgcn_permlane32_swap(
              pack_bf16(dst[0], dst[1]), pack_bf16(dst[4], dst[5]), false,
              false);
          auto sB = __builtin_amdgcn_permlane32_swap(
              pack_bf16(dst[2], dst[3]), pack_bf16(dst[6], dst[7]), false,
              false);
          f0.w[0] = sA[0]; f0.w[1] = sB[0]; f0.w[2] = sA[1]; f0.w[3] = sB[1];
          auto sC = __builtin_amdgcn_permlane32_swap(
              pack_bf16(dst[8], dst[9]), pack_bf16(dst[12], dst[13]), false,
              false);
          auto sD = __builtin_amdgcn_permlane32_swap(
              pack_bf16(dst[10], dst[11]), pack_bf16(dst[14], dst[15]),
              false, false);
          f1.w[0] = sC[0]; f1.w[1] = sD[0]; f1.w[2] = sC[1]; f1.w[3] = sD[1];
          da0 = f0.f;
          da1 = f1.f;
        }
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          bf16x8v b0 = read_bfrag<128>(
              kt_lds, t * 32 + (lane & 31), ks * 32 + hi * 8);
          dq_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              da0, b0, dq_acc[t], 0, 0, 0);
          bf16x8v b1 = read_bfrag<128>(
              kt_lds, t * 32 + (lane & 31), ks * 32 + 16 + hi * 8);
          dq_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              da1, b1, dq_acc[t], 0, 0, 0);
        }
      }
    }
    if (more) {
      char* nk = smem + (cur ^ 1) * (3 * KB);
      k_st.write_rm(nk);
      kt_st.write(nk + KB);
      v_st.write_rm(nk + 2 * KB);
    }
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = m0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (row >= S) continue;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      const int col = t * 32 + (lane & 31);
      dqp[(long)row * q_tok + col] = f2bf(dq_acc[t][r] * scale);
    }
  }
}

}  // namespace

extern "C" {

// vt: pre-transposed V plane [B, Hkv, D, S_pad] (tok_transpose_head)
hipError_t tok_attn_fwd(const void* q, const void* k, const void* vt, void* o,
                        float* lse, int B, int S, int Hq, int Hkv, int D,
                        int S_pad, int causal, hipStream_t stream) {
  dim3 grid((S + NW * 32 - 1) / (NW * 32), Hq, B);
  const float scale = 1.f / sqrtf((float)D);
#define LAUNCH_FWD(DD, CC)                                                    \
  attn_fwd_kernel_v3<DD, CC><<<grid, NTHREADS, 0, stream>>>(                  \
      (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)vt, (bf16_t*)o,     \
      lse, B, S, Hq, Hkv, S_pad, scale)
  if (D == 128) { if (causal) LAUNCH_FWD(128, true); else LAUNCH_FWD(128, false); }
  else if (D == 64) { if (causal) LAUNCH_FWD(64, true); else LAUNCH_FWD(64, false); }
  else return hipErrorInvalidValue;
#undef LAUNCH_FWD
  return hipGetLastError();
}

// q_t/k_t/dot_t: pre-transposed [.., D, S_pad] planes (tok_transpose_head)
hipError_t tok_attn_bwd(const void* q, const void* k, const void* v,
                        const void* o, const void* dout, const void* q_t,
                        const void* k_t, const void* dot_t, const float* lse,
                        float* dsum_ws, void* dq, void* dk, void* dv, int B,
                        int S, int Hq, int Hkv, int D, int S_pad, int causal,
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

  dim3 gkv((S + NW * 32 - 1) / (NW * 32), Hkv, B);
  dim3 gkv4((S + 4 * 32 - 1) / (4 * 32), Hkv, B);
  dim3 gq((S + NW * 32 - 1) / (NW * 32), Hq, B);
  // dv/dk block-shape A/B: TOK_BWD_NW4=1 runs 4-wave blocks (two
  // resident per CU -> cross-block phase drift on each SIMD)
  static const bool nw4 = [] {
    const char* e = getenv("TOK_BWD_NW4");
    return e && e[0] == '1';
  }();
#define LAUNCH_BWD(DD, CC)                                                    \
  do {                                                                        \
    if (nw4) {                                                                \
      attn_bwd_dv_kernel<DD, CC, 4><<<gkv4, 4 * WAVE, 0, stream>>>(           \
          (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)dot_t,           \
          (const bf16_t*)dout, lse, dsum_ws, (bf16_t*)dv, B, S, Hq, Hkv,     \
          S_pad, scale);                                                      \
      attn_bwd_dk_kernel<DD, CC, 4><<<gkv4, 4 * WAVE, 0, stream>>>(           \
          (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,               \
          (const bf16_t*)dout, (const bf16_t*)q_t, lse, dsum_ws,              \
          (bf16_t*)dk, B, S, Hq, Hkv, S_pad, scale);                          \
    } else {                                                                  \
      attn_bwd_dv_kernel<DD, CC><<<gkv, NTHREADS, 0, stream>>>(               \
          (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)dot_t,           \
          (const bf16_t*)dout, lse, dsum_ws, (bf16_t*)dv, B, S, Hq, Hkv,     \
          S_pad, scale);                                                      \
      attn_bwd_dk_kernel<DD, CC><<<gkv, NTHREADS, 0, stream>>>(               \
          (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,               \
          (const bf16_t*)dout, (const bf16_t*)q_t, lse, dsum_ws,              \
          (bf16_t*)dk, B, S, Hq, Hkv, S_pad, scale);                          \
    }                                                                         \
    attn_bwd_dq_kernel<DD, CC><<<gq, NTHREADS, 0, stream>>>(                  \
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,                 \
        (const bf16_t*)dout, (const bf16_t*)k_t, lse, dsum_ws, (bf16_t*)dq,  \
        B, S, Hq, Hkv, S_pad, scale);                                         \
  } while (0)
  if (D == 128) { if (causal) LAUNCH_BWD(128, true); else LAUNCH_BWD(128, false); }
  else { if (causal) LAUNCH_BWD(64, true); else LAUNCH_BWD(64, false); }
#undef LAUNCH_BWD
  return hipGetLastError();
}
}
