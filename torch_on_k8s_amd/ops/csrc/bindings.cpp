// Python bindings for the MI355X-native kernels (torch extension).
//
// The kernels themselves live in pure handwritten HIP files (*.hip); this
// translation unit only does tensor checking, workspace allocation and
// stream plumbing.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

extern "C" {
hipError_t tok_rmsnorm_fwd(const void* x, const void* w, void* y, float* invr,
                           long nrows, int H, float eps, hipStream_t stream);
hipError_t tok_rmsnorm_bwd(const void* x, const void* w, const void* dy,
                           const float* invr, void* dx, float* dw_f32,
                           long nrows, int H, hipStream_t stream);
hipError_t tok_rope(const void* x, void* y, const float* cos_tab,
                    const float* sin_tab, long rows_total, int heads, int D,
                    long table_rows, float sign, hipStream_t stream);
hipError_t tok_adamw(void* p, const void* g, float* m, float* v, long n,
                     float lr, float beta1, float beta2, float eps, float wd,
                     int step, float gscale, hipStream_t stream);
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
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
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
  TOK_HIP_OK(tok_rmsnorm_bwd(x.data_ptr(), w.data_ptr(), dy.data_ptr(),
                             invr.data_ptr<float>(), dx.data_ptr(),
                             dw.data_ptr<float>(), nrows, (int)H,
                             current_stream()));
  return {dx, dw};
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
            double gscale) {
  CHECK_BF16_CUDA(p);
  CHECK_BF16_CUDA(g);
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  const long n = p.numel();
  TORCH_CHECK(n % 8 == 0, "bucket size must be a multiple of 8");
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  TOK_HIP_OK(tok_adamw(p.data_ptr(), g.data_ptr(), m.data_ptr<float>(),
                       v.data_ptr<float>(), n, (float)lr, (float)beta1,
                       (float)beta2, (float)eps, (float)wd, (int)step,
                       (float)gscale, current_stream()));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (bf16, gfx950)");
  mod.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (bf16, gfx950)");
  mod.def("rope", &rope, "Rotary embedding rotate-half (bf16, gfx950)");
  mod.def("adamw_", &adamw_, "Fused AdamW on a flat bucket (gfx950)");
}
