// Python bindings for the MI355X-native kernels (torch extension).
//
// The kernels themselves live in pure handwritten HIP files (*.hip); this
// translation unit only does tensor checking, workspace allocation and
// stream plumbing. Stream access uses PyTorch-ROCm's native HIP stream
// API (at::hip::getCurrentHIPStreamMasqueradingAsCUDA — the real symbol
// exported by a ROCm build of ATen) so this TU contains no CUDA-named
// calls for hipify to rewrite.
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

extern "C" {
hipError_t tok_rmsnorm_fwd(const void* x, const void* w, void* y, float* invr,
                           long nrows, int H, float eps, hipStream_t stream);
int tok_rmsnorm_dw_rsplit(long nrows, int H);
hipError_t tok_rmsnorm_bwd(const void* x, const void* w, const void* dy,
                           const float* invr, void* dx, float* dw_f32,
                           float* dw_ws, long nrows, int H,
                           hipStream_t stream);
hipError_t tok_rmsnorm_res_fwd(const void* x, const void* res, const void* w,
                               void* xr, void* y, float* invr, long nrows,
                               int H, float eps, hipStream_t stream);
hipError_t tok_rmsnorm_res_bwd(const void* xr, const void* w, const void* dy,
                               const void* dxr_in, const float* invr,
                               void* dx, float* dw_f32, float* dw_ws,
                               long nrows, int H, hipStream_t stream);
hipError_t tok_swiglu_fwd(const void* gu, void* out, long rows, int I,
                          hipStream_t stream);
hipError_t tok_swiglu_bwd(const void* dout, const void* gu, void* dgu,
                          long rows, int I, hipStream_t stream);
hipError_t tok_qkv_rope_fwd(const void* qkv, void* q, void* k, void* v,
                            const float* cos_tab, const float* sin_tab,
                            long tokens, int Hq, int Hkv, int D,
                            long table_rows, hipStream_t stream);
hipError_t tok_qkv_rope_bwd(const void* dq, const void* dk, const void* dv,
                            void* dqkv, const float* cos_tab,
                            const float* sin_tab, long tokens, int Hq,
                            int Hkv, int D, long table_rows,
                            hipStream_t stream);
hipError_t tok_rope(const void* x, void* y, const float* cos_tab,
                    const float* sin_tab, long rows_total, int heads, int D,
                    long table_rows, float sign, hipStream_t stream);
hipError_t tok_adamw(void* p, const void* g, float* m, float* v, long n,
                     float lr, float beta1, float beta2, float eps, float wd,
                     int step, float gscale, const int* step_dev,
                     hipStream_t stream);
hipError_t tok_mfma_probe_16x16x32(const void* A, const void* B, float* D,
                                   hipStream_t stream);
hipError_t tok_mfma_probe_32x32x16(const void* A, const void* B, float* D,
                                   hipStream_t stream);
hipError_t tok_attn_fwd(const void* q, const void* k, const void* vt, void* o,
                        float* lse, int B, int S, int Hq, int Hkv, int D,
                        int S_pad, int causal, hipStream_t stream);
hipError_t tok_attn_bwd(const void* q, const void* k, const void* v,
                        const void* o, const void* dout, const void* q_t,
                        const void* k_t, const void* dot_t, const float* lse,
                        float* dsum_ws, void* dq, void* dk, void* dv, int B,
                        int S, int Hq, int Hkv, int D, int S_pad, int causal,
                        hipStream_t stream);
hipError_t tok_transpose_head(const void* in, void* out, int B, int S, int H,
                              int D, int S_pad, hipStream_t stream);
hipError_t tok_attn_decode(const void* q, const void* k, const void* v,
                           void* out, int B, int T, const int* T_dev,
                           int Tmax, int Hq, int Hkv, int D,
                           hipStream_t stream);
hipError_t tok_layernorm_fwd(const void* x, const void* w, const void* b,
                             void* y, float* mu, float* rstd, long nrows,
                             int H, float eps, hipStream_t stream);
hipError_t tok_layernorm_bwd(const void* x, const void* w, const void* dy,
                             const float* mu, const float* rstd, void* dx,
                             float* dw, float* db, long nrows, int H,
                             hipStream_t stream);
hipError_t tok_ce_fwd(const void* logits, const int* labels, float* loss,
                      float* lse, long rows, int V, hipStream_t stream);
hipError_t tok_ce_bwd(const void* logits, const int* labels, const float* lse,
                      const float* gscale, void* dlogits, long rows, int V,
                      hipStream_t stream);
}

namespace {

#define CHECK_BF16_CUDA(t)                                                    \
  TORCH_CHECK(t.is_cuda(), #t " must be on GPU");                             \
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, #t " must be bf16");          \
  TORCH_CHECK(t.is_contiguous(), #t " must be contiguous")

#define TOK_HIP_OK(expr)                                                      \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    TORCH_CHECK(_e == hipSuccess, "HIP kernel launch failed: ",               \
                hipGetErrorString(_e));                                       \
  } while (0)

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  const long H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden dim must be a multiple of 8");
  TORCH_CHECK(w.numel() == H, "weight shape mismatch");
  const long nrows = x.numel() / H;
  auto y = at::empty_like(x);
  auto invr = at::empty({nrows}, x.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                             invr.data_ptr<float>(), nrows, (int)H,
                             (float)eps, current_stream()));
  return {y, invr};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor x, at::Tensor w, at::Tensor dy,
                                    at::Tensor invr) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  CHECK_BF16_CUDA(dy);
  const long H = x.size(-1);
  const long nrows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  const int rsplit = tok_rmsnorm_dw_rsplit(nrows, (int)H);
  auto ws = at::empty({(long)rsplit * H}, x.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_rmsnorm_bwd(x.data_ptr(), w.data_ptr(), dy.data_ptr(),
                             invr.data_ptr<float>(), dx.data_ptr(),
                             dw.data_ptr<float>(), ws.data_ptr<float>(),
                             nrows, (int)H, current_stream()));
  return {dx, dw};
}

// Fused residual + RMSNorm: (x, res?, w) -> (y, xr, invr)
std::vector<at::Tensor> rmsnorm_res_fwd(at::Tensor x,
                                        c10::optional<at::Tensor> res,
                                        at::Tensor w, double eps) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  const long H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden dim must be a multiple of 8");
  const long nrows = x.numel() / H;
  const void* res_p = nullptr;
  if (res.has_value()) {
    CHECK_BF16_CUDA((*res));
    TORCH_CHECK(res->sizes() == x.sizes(), "residual shape mismatch");
    res_p = res->data_ptr();
  }
  auto xr = at::empty_like(x);
  auto y = at::empty_like(x);
  auto invr = at::empty({nrows}, x.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_rmsnorm_res_fwd(x.data_ptr(), res_p, w.data_ptr(),
                                 xr.data_ptr(), y.data_ptr(),
                                 invr.data_ptr<float>(), nrows, (int)H,
                                 (float)eps, current_stream()));
  return {y, xr, invr};
}

std::vector<at::Tensor> rmsnorm_res_bwd(at::Tensor xr, at::Tensor w,
                                        at::Tensor dy,
                                        c10::optional<at::Tensor> dxr,
                                        at::Tensor invr) {
  CHECK_BF16_CUDA(xr);
  CHECK_BF16_CUDA(dy);
  const long H = xr.size(-1);
  const long nrows = xr.numel() / H;
  const void* dxr_p = nullptr;
  if (dxr.has_value()) {
    CHECK_BF16_CUDA((*dxr));
    dxr_p = dxr->data_ptr();
  }
  auto dx = at::empty_like(xr);
  auto dw = at::zeros({H}, xr.options().dtype(at::kFloat));
  const int rsplit = tok_rmsnorm_dw_rsplit(nrows, (int)H);
  auto ws = at::empty({(long)rsplit * H}, xr.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_rmsnorm_res_bwd(xr.data_ptr(), w.data_ptr(), dy.data_ptr(),
                                 dxr_p, invr.data_ptr<float>(), dx.data_ptr(),
                                 dw.data_ptr<float>(), ws.data_ptr<float>(),
                                 nrows, (int)H, current_stream()));
  return {dx, dw};
}

// gu: [rows, 2I] packed gate|up -> out [rows, I] = silu(gate)*up
at::Tensor swiglu_fwd(at::Tensor gu) {
  CHECK_BF16_CUDA(gu);
  const long I2 = gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0, "packed gate_up dim must be a multiple of 16");
  const long I = I2 / 2;
  const long rows = gu.numel() / I2;
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = at::empty(sizes, gu.options());
  TOK_HIP_OK(tok_swiglu_fwd(gu.data_ptr(), out.data_ptr(), rows, (int)I,
                            current_stream()));
  return out;
}

at::Tensor swiglu_bwd(at::Tensor dout, at::Tensor gu) {
  CHECK_BF16_CUDA(dout);
  CHECK_BF16_CUDA(gu);
  const long I2 = gu.size(-1);
  const long I = I2 / 2;
  const long rows = gu.numel() / I2;
  TORCH_CHECK(dout.numel() == rows * I, "dout shape mismatch");
  auto dgu = at::empty_like(gu);
  TOK_HIP_OK(tok_swiglu_bwd(dout.data_ptr(), gu.data_ptr(), dgu.data_ptr(),
                            rows, (int)I, current_stream()));
  return dgu;
}

// qkv: [B, S, (Hq+2Hkv)*D] -> roped q [B,S,Hq,D], k [B,S,Hkv,D], v [B,S,Hkv,D]
std::vector<at::Tensor> qkv_rope_fwd(at::Tensor qkv, at::Tensor cos_tab,
                                     at::Tensor sin_tab, long Hq, long Hkv,
                                     long D) {
  CHECK_BF16_CUDA(qkv);
  TORCH_CHECK(cos_tab.scalar_type() == at::kFloat && cos_tab.is_contiguous());
  TORCH_CHECK(sin_tab.scalar_type() == at::kFloat && sin_tab.is_contiguous());
  TORCH_CHECK(D % 16 == 0, "head dim must be a multiple of 16");
  const long B = qkv.size(0), S = qkv.size(1);
  TORCH_CHECK(qkv.size(2) == (Hq + 2 * Hkv) * D, "packed qkv dim mismatch");
  const long tokens = B * S;
  const long table_rows = cos_tab.numel() / (D / 2);
  TORCH_CHECK(S % table_rows == 0 || table_rows % S == 0 || table_rows == S,
              "rope table rows must match sequence");
  auto q = at::empty({B, S, Hq, D}, qkv.options());
  auto k = at::empty({B, S, Hkv, D}, qkv.options());
  auto v = at::empty({B, S, Hkv, D}, qkv.options());
  TOK_HIP_OK(tok_qkv_rope_fwd(qkv.data_ptr(), q.data_ptr(), k.data_ptr(),
                              v.data_ptr(), cos_tab.data_ptr<float>(),
                              sin_tab.data_ptr<float>(), tokens, (int)Hq,
                              (int)Hkv, (int)D, table_rows,
                              current_stream()));
  return {q, k, v};
}

at::Tensor qkv_rope_bwd(at::Tensor dq, at::Tensor dk, at::Tensor dv,
                        at::Tensor cos_tab, at::Tensor sin_tab) {
  CHECK_BF16_CUDA(dq);
  CHECK_BF16_CUDA(dk);
  CHECK_BF16_CUDA(dv);
  const long B = dq.size(0), S = dq.size(1), Hq = dq.size(2), D = dq.size(3);
  const long Hkv = dk.size(2);
  const long tokens = B * S;
  const long table_rows = cos_tab.numel() / (D / 2);
  auto dqkv = at::empty({B, S, (Hq + 2 * Hkv) * D}, dq.options());
  TOK_HIP_OK(tok_qkv_rope_bwd(dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                              dqkv.data_ptr(), cos_tab.data_ptr<float>(),
                              sin_tab.data_ptr<float>(), tokens, (int)Hq,
                              (int)Hkv, (int)D, table_rows,
                              current_stream()));
  return dqkv;
}

at::Tensor rope(at::Tensor x, at::Tensor cos_tab, at::Tensor sin_tab,
                long heads, double sign) {
  CHECK_BF16_CUDA(x);
  TORCH_CHECK(cos_tab.scalar_type() == at::kFloat && cos_tab.is_contiguous());
  TORCH_CHECK(sin_tab.scalar_type() == at::kFloat && sin_tab.is_contiguous());
  const long D = x.size(-1);
  TORCH_CHECK(D % 16 == 0, "head dim must be a multiple of 16");
  const long rows_total = x.numel() / D;
  const long table_rows = cos_tab.numel() / (D / 2);
  TORCH_CHECK((rows_total / heads) % table_rows == 0,
              "rope table rows must divide token count");
  auto y = at::empty_like(x);
  TOK_HIP_OK(tok_rope(x.data_ptr(), y.data_ptr(), cos_tab.data_ptr<float>(),
                      sin_tab.data_ptr<float>(), rows_total, (int)heads,
                      (int)D, table_rows, (float)sign, current_stream()));
  return y;
}

void adamw_(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v, double lr,
            double beta1, double beta2, double eps, double wd, long step,
            double gscale, c10::optional<at::Tensor> step_dev) {
  CHECK_BF16_CUDA(p);
  CHECK_BF16_CUDA(g);
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  const long n = p.numel();
  TORCH_CHECK(n % 8 == 0, "bucket size must be a multiple of 8");
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  const int* sd = nullptr;
  if (step_dev.has_value()) {
    TORCH_CHECK(step_dev->scalar_type() == at::kInt && step_dev->is_cuda());
    sd = step_dev->data_ptr<int>();
  }
  TOK_HIP_OK(tok_adamw(p.data_ptr(), g.data_ptr(), m.data_ptr<float>(),
                       v.data_ptr<float>(), n, (float)lr, (float)beta1,
                       (float)beta2, (float)eps, (float)wd, (int)step,
                       (float)gscale, sd, current_stream()));
}

// q: [B,S,Hq,D], k/v: [B,S,Hkv,D] bf16 contiguous -> (o, lse[B,Hq,S] f32)
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 bool causal) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(k);
  CHECK_BF16_CUDA(v);
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  TORCH_CHECK(k.sizes() == v.sizes() && k.size(0) == B && k.size(1) == S);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  const int S_pad = (S + 63) / 64 * 64;
  auto vt = at::empty({B, Hkv, D, S_pad}, q.options());
  TOK_HIP_OK(tok_transpose_head(v.data_ptr(), vt.data_ptr(), B, S, Hkv, D,
                                S_pad, current_stream()));
  TOK_HIP_OK(tok_attn_fwd(q.data_ptr(), k.data_ptr(), vt.data_ptr(),
                          o.data_ptr(), lse.data_ptr<float>(), B, S, Hq, Hkv,
                          D, S_pad, causal ? 1 : 0, current_stream()));
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor o, at::Tensor lse, at::Tensor dout,
                                 bool causal) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(dout);
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto dsum = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  const int S_pad = (S + 63) / 64 * 64;
  auto qt = at::empty({B, Hq, D, S_pad}, q.options());
  auto kt = at::empty({B, Hkv, D, S_pad}, q.options());
  auto dot = at::empty({B, Hq, D, S_pad}, q.options());
  TOK_HIP_OK(tok_transpose_head(q.data_ptr(), qt.data_ptr(), B, S, Hq, D,
                                S_pad, current_stream()));
  TOK_HIP_OK(tok_transpose_head(k.data_ptr(), kt.data_ptr(), B, S, Hkv, D,
                                S_pad, current_stream()));
  TOK_HIP_OK(tok_transpose_head(dout.data_ptr(), dot.data_ptr(), B, S, Hq, D,
                                S_pad, current_stream()));
  TOK_HIP_OK(tok_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          o.data_ptr(), dout.data_ptr(), qt.data_ptr(),
                          kt.data_ptr(), dot.data_ptr(),
                          lse.data_ptr<float>(), dsum.data_ptr<float>(),
                          dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), B, S,
                          Hq, Hkv, D, S_pad, causal ? 1 : 0, current_stream()));
  return {dq, dk, dv};
}

// q: [B, Hq, D]; kcache/vcache: [B, Tmax, Hkv, D] (first T rows valid)
at::Tensor attn_decode(at::Tensor q, at::Tensor kcache, at::Tensor vcache,
                       long T, c10::optional<at::Tensor> T_dev) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(kcache);
  CHECK_BF16_CUDA(vcache);
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Tmax = kcache.size(1), Hkv = kcache.size(2);
  TORCH_CHECK(T_dev.has_value() || (T >= 1 && T <= Tmax));
  TORCH_CHECK(D == 64 || D == 128);
  const int* td = nullptr;
  if (T_dev.has_value()) {
    TORCH_CHECK(T_dev->scalar_type() == at::kInt && T_dev->is_cuda());
    td = T_dev->data_ptr<int>();
  }
  auto out = at::empty_like(q);
  TOK_HIP_OK(tok_attn_decode(q.data_ptr(), kcache.data_ptr(),
                             vcache.data_ptr(), out.data_ptr(), B, (int)T,
                             td, Tmax, Hq, Hkv, D, current_stream()));
  return out;
}

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, double eps) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  CHECK_BF16_CUDA(b);
  const long H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden dim must be a multiple of 8");
  const long nrows = x.numel() / H;
  auto y = at::empty_like(x);
  auto mu = at::empty({nrows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({nrows}, x.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_layernorm_fwd(x.data_ptr(), w.data_ptr(), b.data_ptr(),
                               y.data_ptr(), mu.data_ptr<float>(),
                               rstd.data_ptr<float>(), nrows, (int)H,
                               (float)eps, current_stream()));
  return {y, mu, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor x, at::Tensor w,
                                      at::Tensor dy, at::Tensor mu,
                                      at::Tensor rstd) {
  CHECK_BF16_CUDA(x);
  const long H = x.size(-1);
  const long nrows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  auto db = at::zeros({H}, x.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_layernorm_bwd(x.data_ptr(), w.data_ptr(), dy.data_ptr(),
                               mu.data_ptr<float>(), rstd.data_ptr<float>(),
                               dx.data_ptr(), dw.data_ptr<float>(),
                               db.data_ptr<float>(), nrows, (int)H,
                               current_stream()));
  return {dx, dw, db};
}

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor labels) {
  CHECK_BF16_CUDA(logits);
  TORCH_CHECK(labels.scalar_type() == at::kInt && labels.is_cuda());
  const long V = logits.size(-1);
  TORCH_CHECK(V % 8 == 0, "vocab must be a multiple of 8");
  const long rows = logits.numel() / V;
  TORCH_CHECK(labels.numel() == rows);
  auto loss = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({rows}, logits.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_ce_fwd(logits.data_ptr(), labels.data_ptr<int>(),
                        loss.data_ptr<float>(), lse.data_ptr<float>(), rows,
                        (int)V, current_stream()));
  return {loss, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor labels, at::Tensor lse,
                  at::Tensor gscale) {
  CHECK_BF16_CUDA(logits);
  TORCH_CHECK(gscale.scalar_type() == at::kFloat && gscale.is_cuda());
  const long V = logits.size(-1);
  const long rows = logits.numel() / V;
  auto dlogits = at::empty_like(logits);
  TOK_HIP_OK(tok_ce_bwd(logits.data_ptr(), labels.data_ptr<int>(),
                        lse.data_ptr<float>(), gscale.data_ptr<float>(),
                        dlogits.data_ptr(), rows, (int)V, current_stream()));
  return dlogits;
}

at::Tensor mfma_probe(at::Tensor A, at::Tensor B) {
  CHECK_BF16_CUDA(A);
  CHECK_BF16_CUDA(B);
  if (A.size(0) == 16) {
    TORCH_CHECK(A.sizes() == at::IntArrayRef({16, 32}) &&
                B.sizes() == at::IntArrayRef({32, 16}));
    auto D = at::empty({16, 16}, A.options().dtype(at::kFloat));
    TOK_HIP_OK(tok_mfma_probe_16x16x32(A.data_ptr(), B.data_ptr(),
                                       D.data_ptr<float>(), current_stream()));
    return D;
  }
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) &&
              B.sizes() == at::IntArrayRef({16, 32}));
  auto D = at::empty({32, 32}, A.options().dtype(at::kFloat));
  TOK_HIP_OK(tok_mfma_probe_32x32x16(A.data_ptr(), B.data_ptr(),
                                     D.data_ptr<float>(), current_stream()));
  return D;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("mfma_probe", &mfma_probe, "MFMA 16x16x32 bf16 layout probe");
  mod.def("attn_fwd", &attn_fwd, "Flash attention forward (bf16, gfx950)");
  mod.def("attn_bwd", &attn_bwd, "Flash attention backward (bf16, gfx950)");
  mod.def("ce_fwd", &ce_fwd, "Fused cross-entropy forward (bf16, gfx950)");
  mod.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (bf16, gfx950)");
  mod.def("attn_decode", &attn_decode,
          "Single-token KV-cache attention decode (bf16, gfx950)",
          py::arg("q"), py::arg("kcache"), py::arg("vcache"), py::arg("T"),
          py::arg("T_dev") = py::none());
  mod.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward (bf16, gfx950)");
  mod.def("ce_bwd", &ce_bwd, "Fused cross-entropy backward (bf16, gfx950)");
  mod.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (bf16, gfx950)");
  mod.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (bf16, gfx950)");
  mod.def("rmsnorm_res_fwd", &rmsnorm_res_fwd,
          "Fused residual-add + RMSNorm forward (bf16, gfx950)",
          py::arg("x"), py::arg("res"), py::arg("w"), py::arg("eps"));
  mod.def("rmsnorm_res_bwd", &rmsnorm_res_bwd,
          "Fused residual-add + RMSNorm backward (bf16, gfx950)",
          py::arg("xr"), py::arg("w"), py::arg("dy"), py::arg("dxr"),
          py::arg("invr"));
  mod.def("swiglu_fwd", &swiglu_fwd,
          "Fused SwiGLU over packed gate_up (bf16, gfx950)");
  mod.def("swiglu_bwd", &swiglu_bwd,
          "Fused SwiGLU backward (bf16, gfx950)");
  mod.def("qkv_rope_fwd", &qkv_rope_fwd,
          "Packed-QKV split + RoPE (bf16, gfx950)");
  mod.def("qkv_rope_bwd", &qkv_rope_bwd,
          "Packed-QKV gather + inverse RoPE (bf16, gfx950)");
  mod.def("rope", &rope, "Rotary embedding rotate-half (bf16, gfx950)");
  mod.def("adamw_", &adamw_, "Fused AdamW on a flat bucket (gfx950)",
          py::arg("p"), py::arg("g"), py::arg("m"), py::arg("v"),
          py::arg("lr"), py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
          py::arg("wd"), py::arg("step"), py::arg("gscale"),
          py::arg("step_dev") = py::none());
}
