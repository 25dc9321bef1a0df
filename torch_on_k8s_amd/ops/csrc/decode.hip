// Single-token attention decode for MI355X (gfx950) — the serving path.
//
// Memory-bound KV-cache read (guide Appendix B "Attention decode"):
// one block per (batch, q-head); 8 waves split the T cached positions in
// strided chunks with independent online-softmax state; lanes own 2 head
// dims (D=128) so each K/V row is one coalesced 256 B wave-load; the 8
// partial (m, l, o) states merge through LDS at the end.
//   q: [B, Hq, D] (the new token, post-RoPE)
//   kcache/vcache: [B, Tmax, Hkv, D], first T rows valid (post-RoPE)
//   out: [B, Hq, D]
#include "common.h"

namespace {

constexpr int NW_DEC = 8;
constexpr int NTH_DEC = NW_DEC * WAVE;  // 512

template <int D>
__global__ __launch_bounds__(NTH_DEC) void attn_decode_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, bf16_t* __restrict__ out, int B, int T,
    const int* __restrict__ T_dev, int Tmax, int Hq, int Hkv, float scale) {
  if (T_dev != nullptr) T = *T_dev;  // hipGraph replay: T lives on-device
  constexpr int EPL = D / WAVE;  // elems per lane (2 for D=128, 1 for 64)
  // per-wave partial state: o slab [NW][D] f32 + (m, l) pairs
  __shared__ __attribute__((aligned(16))) float o_slab[NW_DEC][D];
  __shared__ float m_slab[NW_DEC], l_slab[NW_DEC];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int hq = blockIdx.x;
  const int b = blockIdx.y;
  const int hkv = hq / (Hq / Hkv);

  const long kv_tok = (long)Hkv * D;
  const bf16_t* qp = q + ((long)b * Hq + hq) * D;
  const bf16_t* kp = k + ((long)b * Tmax) * kv_tok + (long)hkv * D;
  const bf16_t* vp = v + ((long)b * Tmax) * kv_tok + (long)hkv * D;

  float qr[EPL];
#pragma unroll
  for (int e = 0; e < EPL; ++e) qr[e] = bf2f(qp[lane * EPL + e]);

  float m = (-INFINITY), l = 0.f, o[EPL];
#pragma unroll
  for (int e = 0; e < EPL; ++e) o[e] = 0.f;

  for (int t = wid; t < T; t += NW_DEC) {
    const bf16_t* kr = kp + (long)t * kv_tok;
    const bf16_t* vr = vp + (long)t * kv_tok;
    float s = 0.f;
#pragma unroll
    for (int e = 0; e < EPL; ++e)
      s = fmaf(qr[e], bf2f(kr[lane * EPL + e]), s);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) s += __shfl_xor(s, off, WAVE);
    s *= scale;
    const float mn = fmaxf(m, s);
    const float alpha = (m == (-INFINITY)) ? 0.f : __expf(m - mn);
    const float p = __expf(s - mn);
    m = mn;
    l = l * alpha + p;
#pragma unroll
    for (int e = 0; e < EPL; ++e)
      o[e] = o[e] * alpha + p * bf2f(vr[lane * EPL + e]);
  }

  // merge the 8 per-wave states
#pragma unroll
  for (int e = 0; e < EPL; ++e) o_slab[wid][lane * EPL + e] = o[e];
  if (lane == 0) {
    m_slab[wid] = m;
    l_slab[wid] = l;
  }
  __syncthreads();
  if (wid == 0) {
    float M = (-INFINITY);
#pragma unroll
    for (int w = 0; w < NW_DEC; ++w) M = fmaxf(M, m_slab[w]);
    float L = 0.f, acc[EPL];
#pragma unroll
    for (int e = 0; e < EPL; ++e) acc[e] = 0.f;
#pragma unroll
    for (int w = 0; w < NW_DEC; ++w) {
      const float sw =
          (m_slab[w] == (-INFINITY)) ? 0.f : __expf(m_slab[w] - M);
      L += l_slab[w] * sw;
#pragma unroll
      for (int e = 0; e < EPL; ++e)
        acc[e] = fmaf(o_slab[w][lane * EPL + e], sw, acc[e]);
    }
    const float inv = (L > 0.f) ? 1.f / L : 0.f;
    bf16_t* op = out + ((long)b * Hq + hq) * D;
#pragma unroll
    for (int e = 0; e < EPL; ++e) op[lane * EPL + e] = f2bf(acc[e] * inv);
  }
}

}  // namespace

extern "C" {

hipError_t tok_attn_decode(const void* q, const void* k, const void* v,
                           void* out, int B, int T, const int* T_dev,
                           int Tmax, int Hq, int Hkv, int D,
                           hipStream_t stream) {
  dim3 grid(Hq, B);
  const float scale = 1.f / sqrtf((float)D);
  if (D == 128)
    attn_decode_kernel<128><<<grid, NTH_DEC, 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)out,
        B, T, T_dev, Tmax, Hq, Hkv, scale);
  else if (D == 64)
    attn_decode_kernel<64><<<grid, NTH_DEC, 0, stream>>>(
        (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v, (bf16_t*)out,
        B, T, T_dev, Tmax, Hq, Hkv, scale);
  else
    return hipErrorInvalidValue;
  return hipGetLastError();
}
}
