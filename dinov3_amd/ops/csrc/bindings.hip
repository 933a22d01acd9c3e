// PyTorch bindings for the dinov3_amd CDNA4 kernel library.
// Compiled by hipcc (PYTORCH_ROCM_ARCH=gfx950) into dinov3_amd/ops/_hip_ops.so.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define CHECK_INPUT(x) \
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be a contiguous HIP tensor")

// ---- extern launchers (defined in the .hip kernel TUs) ----
template <typename T>
void launch_layernorm_fwd(const T*, const T*, const T*, T*, float*, float*, long, int,
                          float, hipStream_t);
template <typename T>
void launch_layernorm_bwd(const T*, const T*, const T*, const float*, const float*, T*,
                          float*, float*, long, int, hipStream_t);
template <typename T>
void launch_rmsnorm_fwd(const T*, const T*, T*, float*, long, int, float, hipStream_t);
template <typename T>
void launch_rmsnorm_bwd(const T*, const T*, const T*, const float*, T*, float*, long, int,
                        hipStream_t);
template <typename T>
void launch_l2norm_fwd(const T*, T*, float*, long, int, float, hipStream_t);
template <typename T>
void launch_l2norm_bwd(const T*, const T*, const float*, T*, long, int, float, hipStream_t);
template <typename T>
void launch_bias_gelu_fwd(const T*, const T*, T*, long, int, hipStream_t);
template <typename T>
void launch_bias_gelu_bwd(const T*, const T*, const T*, T*, float*, long, int, hipStream_t);
template <typename T>
void launch_ls_axpy_fwd(const T*, const T*, const T*, T*, long, int, hipStream_t);
template <typename T>
void launch_ls_axpy_bwd(const T*, const T*, const T*, T*, float*, long, int, hipStream_t);
template <typename T>
void launch_ls_axpy_bias_fwd(const T*, const T*, const T*, const T*, T*, long, int,
                             hipStream_t);
template <typename T>
void launch_ls_axpy_bias_bwd(const T*, const T*, const T*, const T*, T*, float*, float*,
                             long, int, hipStream_t);
template <typename T>
void launch_ls_scatter_add(T*, const long*, const T*, const T*, const T*, const float*,
                           long, int, hipStream_t);
template <typename T>
void launch_ls_scatter_bwd(const T*, const long*, const T*, const T*, const T*,
                           const float*, T*, float*, float*, long, int, hipStream_t);
template <typename T>
void launch_row_gather(const T*, const long*, T*, long, int, hipStream_t);
template <typename T>
void launch_row_scatter_add(T*, const long*, const T*, const float*, long, int,
                            hipStream_t);
template <typename T>
void launch_row_gather_scaled(const T*, const long*, const float*, T*, long, int,
                              hipStream_t);
template <typename T>
void launch_swiglu_fwd(const T*, T*, long, int, hipStream_t);
template <typename T>
void launch_swiglu_bwd(const T*, const T*, T*, long, int, hipStream_t);
template <typename T>
void launch_rope_fwd(const T*, const float*, const float*, T*, long, int, int, int,
                     hipStream_t);
void launch_dino_ce_fwd(const __hip_bfloat16*, const float*, float*, float*, float*, int,
                        int, int, long, float, bool, hipStream_t);
void launch_dino_ce_bwd(const __hip_bfloat16*, const float*, const float*, const float*,
                        const float*, __hip_bfloat16*, int, int, int, long, float, bool,
                        hipStream_t);
void launch_ibot_ce_fwd(const __hip_bfloat16*, const float*, const float*, float*, float*,
                        float*, int, long, float, hipStream_t);
void launch_ibot_ce_bwd(const __hip_bfloat16*, const float*, const float*, const float*,
                        const float*, const float*, __hip_bfloat16*, int, long, float,
                        hipStream_t);
void launch_sinkhorn_exp(const __hip_bfloat16*, float*, float*, long, float, hipStream_t);
void launch_sinkhorn_fact_colsum(const __hip_bfloat16*, const float*, float*, int, long,
                                 float, hipStream_t);
void launch_sinkhorn_fact_rowsum(const __hip_bfloat16*, const float*, float*, int, long,
                                 float, hipStream_t);
void launch_ibot_ce_fact_fwd(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                             const float*, const float*, float*, float*, float*, int,
                             long, float, float, hipStream_t);
void launch_ibot_ce_fact_bwd(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                             const float*, const float*, const float*, const float*,
                             const float*, __hip_bfloat16*, int, long, float, float,
                             hipStream_t);
void launch_dino_ce_fact_fwd(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                             const float*, float*, float*, float*, int, int, int, long,
                             float, float, bool, hipStream_t);
void launch_dino_ce_fact_bwd(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                             const float*, const float*, const float*, const float*,
                             __hip_bfloat16*, int, int, int, long, float, float, bool,
                             hipStream_t);
void launch_sinkhorn_colsum(const float*, float*, int, long, const float*, hipStream_t);
void launch_sinkhorn_div_row(float*, const float*, int, long, float, const float*, bool,
                             hipStream_t);
void launch_fmha_rope_fwd(const __hip_bfloat16*, const float*, const float*,
                          __hip_bfloat16*, float*, int, int, int, int, int, float,
                          hipStream_t);
void launch_fmha_rope_bwd_pre(const __hip_bfloat16*, const __hip_bfloat16*, float*, int,
                              int, int, int, hipStream_t);
void launch_fmha_rope_bwd_dq(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                             const float*, const float*, const float*, __hip_bfloat16*,
                             int, int, int, int, int, float, hipStream_t);
void launch_fmha_rope_bwd_dkv(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                              const float*, const float*, const float*, __hip_bfloat16*,
                              int, int, int, int, int, float, hipStream_t);
void launch_patch_embed_fwd(const __hip_bfloat16*, const __hip_bfloat16*,
                            const __hip_bfloat16*, __hip_bfloat16*, int, int, int, int,
                            int, int, hipStream_t);
void launch_probe_mfma(const __hip_bfloat16*, const __hip_bfloat16*, float*, hipStream_t);
void launch_fmha_fwd(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                     __hip_bfloat16*, float*, int, int, int, float, hipStream_t);
void launch_fmha_bwd_pre(const __hip_bfloat16*, const __hip_bfloat16*, float*, long, int,
                         hipStream_t);
void launch_fmha_bwd_dq(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                        const __hip_bfloat16*, const float*, const float*, __hip_bfloat16*,
                        int, int, int, float, hipStream_t);
void launch_fmha_bwd_dkv(const __hip_bfloat16*, const __hip_bfloat16*,
                         const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                         const float*, __hip_bfloat16*, __hip_bfloat16*, int, int, int,
                         float, hipStream_t);
template <typename T>
void launch_multi_tensor_ema(T* const*, const T* const*, const long*, const int*,
                             const long*, int, float, hipStream_t);
template <typename T>
void launch_multi_tensor_adamw(T* const*, const T* const*, float* const*, float* const*,
                               float* const*, const long*, const int*, const long*, int,
                               float, float, float, float, float, float, float, float,
                               bool, hipStream_t);
template <typename T>
void launch_multi_tensor_l2norm_sq(const T* const*, const long*, const int*, const long*,
                                   int, float*, hipStream_t);
template <typename T, typename GT>
void launch_multi_tensor_adamw_planned(T* const*, const GT* const*, float* const*,
                                       float* const*, float* const*, const long*,
                                       const int*, const long*, int, const float*,
                                       const float*, const float*, const int*,
                                       const float*, float, float, float, float, float,
                                       float, float, float, bool, hipStream_t);
template <typename T>
void launch_multi_tensor_l2norm_planned(const T* const*, const long*, const int*,
                                        const long*, int, const int*, float*,
                                        hipStream_t);

namespace {

hipStream_t current_stream() { return at::hip::getCurrentHIPStream().stream(); }

#define DISPATCH_FLOAT_BF16(TYPE, NAME, ...)                                   \
  [&] {                                                                        \
    if (TYPE == at::ScalarType::Float) {                                       \
      using scalar_t = float;                                                  \
      return __VA_ARGS__();                                                    \
    } else if (TYPE == at::ScalarType::BFloat16) {                             \
      using scalar_t = __hip_bfloat16;                                         \
      return __VA_ARGS__();                                                    \
    } else {                                                                   \
      TORCH_CHECK(false, NAME ": unsupported dtype ", TYPE);                   \
    }                                                                          \
  }()

// ------------------------------- norms ---------------------------------

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                                         double eps) {
  CHECK_INPUT(x);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat));
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(x.scalar_type(), "layernorm_fwd", [&] {
    launch_layernorm_fwd<scalar_t>(
        (const scalar_t*)x.data_ptr(), (const scalar_t*)w.data_ptr(),
        (const scalar_t*)b.data_ptr(), (scalar_t*)y.data_ptr(),
        mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, D, (float)eps,
        current_stream());
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                         torch::Tensor mean, torch::Tensor rstd) {
  CHECK_INPUT(dy);
  CHECK_INPUT(x);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  // 8 shadow accumulators (LN_SHADOWS in norms.hip): per-address atomic
  // chains shrink 8x; the [8, D] partials are summed here
  auto dw = torch::zeros({32, D}, x.options().dtype(torch::kFloat));
  auto db = torch::zeros({32, D}, x.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(x.scalar_type(), "layernorm_bwd", [&] {
    launch_layernorm_bwd<scalar_t>(
        (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
        (const scalar_t*)w.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
        (scalar_t*)dx.data_ptr(), dw.data_ptr<float>(), db.data_ptr<float>(), rows, D,
        current_stream());
  });
  return {dx, dw.sum(0).to(x.scalar_type()), db.sum(0).to(x.scalar_type())};
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_INPUT(x);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(x.scalar_type(), "rmsnorm_fwd", [&] {
    launch_rmsnorm_fwd<scalar_t>((const scalar_t*)x.data_ptr(), (const scalar_t*)w.data_ptr(),
                                 (scalar_t*)y.data_ptr(), rstd.data_ptr<float>(), rows, D,
                                 (float)eps, current_stream());
  });
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                       torch::Tensor rstd) {
  CHECK_INPUT(dy);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({D}, x.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(x.scalar_type(), "rmsnorm_bwd", [&] {
    launch_rmsnorm_bwd<scalar_t>((const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
                                 (const scalar_t*)w.data_ptr(), rstd.data_ptr<float>(),
                                 (scalar_t*)dx.data_ptr(), dw.data_ptr<float>(), rows, D,
                                 current_stream());
  });
  return {dx, dw.to(x.scalar_type())};
}

std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x, double eps) {
  CHECK_INPUT(x);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto s = torch::empty({rows}, x.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(x.scalar_type(), "l2norm_fwd", [&] {
    launch_l2norm_fwd<scalar_t>((const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(),
                                s.data_ptr<float>(), rows, D, (float)eps, current_stream());
  });
  return {y, s};
}

torch::Tensor l2norm_bwd(torch::Tensor dy, torch::Tensor y, torch::Tensor s, double eps) {
  CHECK_INPUT(dy);
  const int D = y.size(-1);
  const long rows = y.numel() / D;
  auto dx = torch::empty_like(y);
  DISPATCH_FLOAT_BF16(y.scalar_type(), "l2norm_bwd", [&] {
    launch_l2norm_bwd<scalar_t>((const scalar_t*)dy.data_ptr(), (const scalar_t*)y.data_ptr(),
                                s.data_ptr<float>(), (scalar_t*)dx.data_ptr(), rows, D,
                                (float)eps, current_stream());
  });
  return dx;
}

// ----------------------------- elementwise ------------------------------

torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias) {
  CHECK_INPUT(x);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "bias_gelu: H must be a multiple of 8");
  const long rows = x.numel() / H;
  auto y = torch::empty_like(x);
  DISPATCH_FLOAT_BF16(x.scalar_type(), "bias_gelu_fwd", [&] {
    launch_bias_gelu_fwd<scalar_t>((const scalar_t*)x.data_ptr(),
                                   (const scalar_t*)bias.data_ptr(), (scalar_t*)y.data_ptr(),
                                   rows, H, current_stream());
  });
  return y;
}

std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias) {
  CHECK_INPUT(dy);
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dbias = torch::zeros({32, H}, x.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(x.scalar_type(), "bias_gelu_bwd", [&] {
    launch_bias_gelu_bwd<scalar_t>((const scalar_t*)dy.data_ptr(),
                                   (const scalar_t*)x.data_ptr(),
                                   (const scalar_t*)bias.data_ptr(), (scalar_t*)dx.data_ptr(),
                                   dbias.data_ptr<float>(), rows, H, current_stream());
  });
  return {dx, dbias.sum(0).to(x.scalar_type())};
}

torch::Tensor ls_axpy_fwd(torch::Tensor x, torch::Tensor res, torch::Tensor gamma) {
  CHECK_INPUT(x);
  CHECK_INPUT(res);
  const int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "ls_axpy: D must be a multiple of 8");
  const long rows = x.numel() / D;
  auto out = torch::empty_like(x);
  DISPATCH_FLOAT_BF16(x.scalar_type(), "ls_axpy_fwd", [&] {
    launch_ls_axpy_fwd<scalar_t>((const scalar_t*)x.data_ptr(),
                                 (const scalar_t*)res.data_ptr(),
                                 (const scalar_t*)gamma.data_ptr(),
                                 (scalar_t*)out.data_ptr(), rows, D, current_stream());
  });
  return out;
}

torch::Tensor ls_axpy_bias_fwd(torch::Tensor x, torch::Tensor res, torch::Tensor gamma,
                               torch::Tensor bias) {
  CHECK_INPUT(x);
  CHECK_INPUT(res);
  const int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "ls_axpy_bias: D must be a multiple of 8");
  const long rows = x.numel() / D;
  auto out = torch::empty_like(x);
  DISPATCH_FLOAT_BF16(x.scalar_type(), "ls_axpy_bias_fwd", [&] {
    launch_ls_axpy_bias_fwd<scalar_t>(
        (const scalar_t*)x.data_ptr(), (const scalar_t*)res.data_ptr(),
        (const scalar_t*)gamma.data_ptr(), (const scalar_t*)bias.data_ptr(),
        (scalar_t*)out.data_ptr(), rows, D, current_stream());
  });
  return out;
}

std::vector<torch::Tensor> ls_axpy_bwd(torch::Tensor dout, torch::Tensor res,
                                       torch::Tensor gamma) {
  CHECK_INPUT(dout);
  const int D = dout.size(-1);
  const long rows = dout.numel() / D;
  auto dres = torch::empty_like(dout);
  auto dgamma = torch::zeros({32, D}, dout.options().dtype(torch::kFloat));
  DISPATCH_FLOAT_BF16(dout.scalar_type(), "ls_axpy_bwd", [&] {
    launch_ls_axpy_bwd<scalar_t>((const scalar_t*)dout.data_ptr(),
                                 (const scalar_t*)res.data_ptr(),
                                 (const scalar_t*)gamma.data_ptr(),
                                 (scalar_t*)dres.data_ptr(), dgamma.data_ptr<float>(),
                                 rows, D, current_stream());
  });
  return {dres, dgamma.sum(0).to(dout.scalar_type())};
}

std::vector<torch::Tensor> ls_axpy_bias_bwd(torch::Tensor dout, torch::Tensor res,
                                            torch::Tensor gamma, torch::Tensor bias) {
  CHECK_INPUT(dout);
  const int D = dout.size(-1);
  const long rows = dout.numel() / D;
  auto dres = torch::empty_like(dout);
  auto fopt = dout.options().dtype(torch::kFloat);
  auto dgamma = torch::zeros({32, D}, fopt);
  auto dbias = torch::zeros({32, D}, fopt);
  DISPATCH_FLOAT_BF16(dout.scalar_type(), "ls_axpy_bias_bwd", [&] {
    launch_ls_axpy_bias_bwd<scalar_t>(
        (const scalar_t*)dout.data_ptr(), (const scalar_t*)res.data_ptr(),
        (const scalar_t*)gamma.data_ptr(), (const scalar_t*)bias.data_ptr(),
        (scalar_t*)dres.data_ptr(), dgamma.data_ptr<float>(), dbias.data_ptr<float>(),
        rows, D, current_stream());
  });
  return {dres, dgamma.sum(0).to(dout.scalar_type()), dbias.sum(0).to(dout.scalar_type())};
}

torch::Tensor row_gather(torch::Tensor src, torch::Tensor idx) {
  CHECK_INPUT(src);
  const int D = src.size(-1);
  TORCH_CHECK(D % 8 == 0, "row ops need D % 8 == 0");
  const long M = idx.numel();
  auto out = torch::empty({M, D}, src.options());
  DISPATCH_FLOAT_BF16(src.scalar_type(), "row_gather", [&] {
    launch_row_gather<scalar_t>((const scalar_t*)src.data_ptr(), idx.data_ptr<long>(),
                                (scalar_t*)out.data_ptr(), M, D, current_stream());
  });
  return out;
}

void row_scatter_add_(torch::Tensor dst, torch::Tensor idx, torch::Tensor src,
                      torch::Tensor scale) {
  CHECK_INPUT(dst);
  CHECK_INPUT(src);
  const int D = dst.size(-1);
  const long M = idx.numel();
  const float* sp = (scale.defined() && scale.numel() > 0) ? scale.data_ptr<float>() : nullptr;
  DISPATCH_FLOAT_BF16(dst.scalar_type(), "row_scatter_add", [&] {
    launch_row_scatter_add<scalar_t>((scalar_t*)dst.data_ptr(), idx.data_ptr<long>(),
                                     (const scalar_t*)src.data_ptr(), sp, M, D,
                                     current_stream());
  });
}

torch::Tensor row_gather_scaled(torch::Tensor src, torch::Tensor idx, torch::Tensor scale) {
  CHECK_INPUT(src);
  const int D = src.size(-1);
  const long M = idx.numel();
  auto out = torch::empty({M, D}, src.options());
  const float* sp = (scale.defined() && scale.numel() > 0) ? scale.data_ptr<float>() : nullptr;
  DISPATCH_FLOAT_BF16(src.scalar_type(), "row_gather_scaled", [&] {
    launch_row_gather_scaled<scalar_t>((const scalar_t*)src.data_ptr(),
                                       idx.data_ptr<long>(), sp, (scalar_t*)out.data_ptr(),
                                       M, D, current_stream());
  });
  return out;
}

void ls_scatter_add_(torch::Tensor dst, torch::Tensor idx, torch::Tensor src,
                     torch::Tensor gamma, torch::Tensor bias, torch::Tensor scale) {
  CHECK_INPUT(dst);
  CHECK_INPUT(src);
  const int D = dst.size(-1);
  TORCH_CHECK(D % 8 == 0, "ls_scatter: D must be a multiple of 8");
  const long M = idx.numel();
  DISPATCH_FLOAT_BF16(dst.scalar_type(), "ls_scatter_add_", [&] {
    launch_ls_scatter_add<scalar_t>(
        (scalar_t*)dst.data_ptr(), idx.data_ptr<long>(), (const scalar_t*)src.data_ptr(),
        gamma.defined() ? (const scalar_t*)gamma.data_ptr() : nullptr,
        bias.defined() ? (const scalar_t*)bias.data_ptr() : nullptr,
        scale.defined() && scale.numel() > 0 ? scale.data_ptr<float>() : nullptr, M, D,
        current_stream());
  });
}

std::vector<torch::Tensor> ls_scatter_bwd(torch::Tensor dy, torch::Tensor idx,
                                          torch::Tensor src, torch::Tensor gamma,
                                          torch::Tensor bias, torch::Tensor scale) {
  CHECK_INPUT(dy);
  CHECK_INPUT(src);
  const int D = dy.size(-1);
  const long M = idx.numel();
  auto dres = torch::empty_like(src);
  auto fopt = dy.options().dtype(torch::kFloat);
  auto dgamma = gamma.defined() ? torch::zeros({32, D}, fopt) : torch::Tensor();
  auto dbias = bias.defined() ? torch::zeros({32, D}, fopt) : torch::Tensor();
  DISPATCH_FLOAT_BF16(dy.scalar_type(), "ls_scatter_bwd", [&] {
    launch_ls_scatter_bwd<scalar_t>(
        (const scalar_t*)dy.data_ptr(), idx.data_ptr<long>(),
        (const scalar_t*)src.data_ptr(),
        gamma.defined() ? (const scalar_t*)gamma.data_ptr() : nullptr,
        bias.defined() ? (const scalar_t*)bias.data_ptr() : nullptr,
        scale.defined() && scale.numel() > 0 ? scale.data_ptr<float>() : nullptr,
        (scalar_t*)dres.data_ptr(),
        dgamma.defined() ? dgamma.data_ptr<float>() : nullptr,
        dbias.defined() ? dbias.data_ptr<float>() : nullptr, M, D, current_stream());
  });
  return {dres, dgamma.defined() ? dgamma.sum(0).to(dy.scalar_type()) : dgamma,
          dbias.defined() ? dbias.sum(0).to(dy.scalar_type()) : dbias};
}

torch::Tensor swiglu_fwd(torch::Tensor x12) {
  CHECK_INPUT(x12);
  const int H2 = x12.size(-1);
  const int H = H2 / 2;
  const long rows = x12.numel() / H2;
  auto sizes = x12.sizes().vec();
  sizes.back() = H;
  auto y = torch::empty(sizes, x12.options());
  DISPATCH_FLOAT_BF16(x12.scalar_type(), "swiglu_fwd", [&] {
    launch_swiglu_fwd<scalar_t>((const scalar_t*)x12.data_ptr(), (scalar_t*)y.data_ptr(),
                                rows, H, current_stream());
  });
  return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x12) {
  CHECK_INPUT(dy);
  const int H2 = x12.size(-1);
  const int H = H2 / 2;
  const long rows = x12.numel() / H2;
  auto dx12 = torch::empty_like(x12);
  DISPATCH_FLOAT_BF16(x12.scalar_type(), "swiglu_bwd", [&] {
    launch_swiglu_bwd<scalar_t>((const scalar_t*)dy.data_ptr(),
                                (const scalar_t*)x12.data_ptr(), (scalar_t*)dx12.data_ptr(),
                                rows, H, current_stream());
  });
  return dx12;
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor sin_t, torch::Tensor cos_t,
                       int64_t prefix) {
  CHECK_INPUT(x);
  TORCH_CHECK(x.dim() == 4, "rope_fwd expects [B, H, N, hd]");
  TORCH_CHECK(sin_t.scalar_type() == at::ScalarType::Float, "rope tables must be fp32");
  const long BH = x.size(0) * x.size(1);
  const int N = x.size(2);
  const int hd = x.size(3);
  const int P = N - (int)prefix;
  TORCH_CHECK(sin_t.size(0) == P && sin_t.size(1) == hd, "rope table shape mismatch");
  auto y = torch::empty_like(x);
  DISPATCH_FLOAT_BF16(x.scalar_type(), "rope_fwd", [&] {
    launch_rope_fwd<scalar_t>((const scalar_t*)x.data_ptr(), sin_t.data_ptr<float>(),
                              cos_t.data_ptr<float>(), (scalar_t*)y.data_ptr(), BH, N, P, hd,
                              current_stream());
  });
  return y;
}

// ------------------------------ proto CE --------------------------------

std::vector<torch::Tensor> dino_ce_fwd(torch::Tensor x, torch::Tensor t, double temp,
                                       bool ignore_diag) {
  CHECK_INPUT(x);
  CHECK_INPUT(t);
  TORCH_CHECK(x.dim() == 3 && t.dim() == 3, "dino_ce expects [S,B,K] and [T,B,K]");
  TORCH_CHECK(x.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(t.scalar_type() == at::ScalarType::Float);
  const int S = x.size(0), B = x.size(1);
  const int T = t.size(0);
  const long K = x.size(2);
  auto lse = torch::empty({S * B}, x.options().dtype(torch::kFloat));
  auto st = torch::empty({S * B}, x.options().dtype(torch::kFloat));
  auto loss = torch::zeros({}, x.options().dtype(torch::kFloat));
  launch_dino_ce_fwd((const __hip_bfloat16*)x.data_ptr(), t.data_ptr<float>(),
                     lse.data_ptr<float>(), st.data_ptr<float>(), loss.data_ptr<float>(),
                     S, T, B, K, (float)(1.0 / temp), ignore_diag, current_stream());
  return {loss, lse, st};
}

torch::Tensor dino_ce_bwd(torch::Tensor g, torch::Tensor x, torch::Tensor t,
                          torch::Tensor lse, torch::Tensor st, double temp,
                          bool ignore_diag) {
  const int S = x.size(0), B = x.size(1);
  const int T = t.size(0);
  const long K = x.size(2);
  auto dx = torch::empty_like(x);
  launch_dino_ce_bwd((const __hip_bfloat16*)x.data_ptr(), t.data_ptr<float>(),
                     lse.data_ptr<float>(), st.data_ptr<float>(), g.data_ptr<float>(),
                     (__hip_bfloat16*)dx.data_ptr(), S, T, B, K, (float)(1.0 / temp),
                     ignore_diag, current_stream());
  return dx;
}

std::vector<torch::Tensor> ibot_ce_fwd(torch::Tensor x, torch::Tensor t, torch::Tensor w,
                                       double temp) {
  CHECK_INPUT(x);
  CHECK_INPUT(t);
  const int M = x.size(0);
  const long K = x.size(1);
  auto lse = torch::empty({M}, x.options().dtype(torch::kFloat));
  auto st = torch::empty({M}, x.options().dtype(torch::kFloat));
  auto loss = torch::zeros({}, x.options().dtype(torch::kFloat));
  launch_ibot_ce_fwd((const __hip_bfloat16*)x.data_ptr(), t.data_ptr<float>(),
                     w.data_ptr<float>(), lse.data_ptr<float>(), st.data_ptr<float>(),
                     loss.data_ptr<float>(), M, K, (float)(1.0 / temp), current_stream());
  return {loss, lse, st};
}

torch::Tensor ibot_ce_bwd(torch::Tensor g, torch::Tensor x, torch::Tensor t,
                          torch::Tensor w, torch::Tensor lse, torch::Tensor st,
                          double temp) {
  const int M = x.size(0);
  const long K = x.size(1);
  auto dx = torch::empty_like(x);
  launch_ibot_ce_bwd((const __hip_bfloat16*)x.data_ptr(), t.data_ptr<float>(),
                     w.data_ptr<float>(), lse.data_ptr<float>(), st.data_ptr<float>(),
                     g.data_ptr<float>(), (__hip_bfloat16*)dx.data_ptr(), M, K,
                     (float)(1.0 / temp), current_stream());
  return dx;
}

torch::Tensor sinkhorn_fact_colsum(torch::Tensor x, torch::Tensor u, double temp) {
  CHECK_INPUT(x);
  const int M = x.size(0);
  const long K = x.size(1);
  auto A = torch::empty({K}, x.options().dtype(torch::kFloat));
  const float* up = (u.defined() && u.numel() > 0) ? u.data_ptr<float>() : nullptr;
  launch_sinkhorn_fact_colsum((const __hip_bfloat16*)x.data_ptr(), up,
                              A.data_ptr<float>(), M, K, (float)(1.0 / temp),
                              current_stream());
  return A;
}

torch::Tensor sinkhorn_fact_rowsum(torch::Tensor x, torch::Tensor v, double temp) {
  CHECK_INPUT(x);
  const int M = x.size(0);
  const long K = x.size(1);
  auto u = torch::empty({M}, x.options().dtype(torch::kFloat));
  launch_sinkhorn_fact_rowsum((const __hip_bfloat16*)x.data_ptr(), v.data_ptr<float>(),
                              u.data_ptr<float>(), M, K, (float)(1.0 / temp),
                              current_stream());
  return u;
}

std::vector<torch::Tensor> ibot_ce_fact_fwd(torch::Tensor x, torch::Tensor xt,
                                            torch::Tensor u, torch::Tensor v,
                                            torch::Tensor w, double temp, double tt) {
  CHECK_INPUT(x);
  CHECK_INPUT(xt);
  const int M = x.size(0);
  const long K = x.size(1);
  auto lse = torch::empty({M}, x.options().dtype(torch::kFloat));
  auto st = torch::empty({M}, x.options().dtype(torch::kFloat));
  auto loss = torch::zeros({}, x.options().dtype(torch::kFloat));
  launch_ibot_ce_fact_fwd((const __hip_bfloat16*)x.data_ptr(),
                          (const __hip_bfloat16*)xt.data_ptr(), u.data_ptr<float>(),
                          v.data_ptr<float>(), w.data_ptr<float>(), lse.data_ptr<float>(),
                          st.data_ptr<float>(), loss.data_ptr<float>(), M, K,
                          (float)(1.0 / temp), (float)(1.0 / tt), current_stream());
  return {loss, lse, st};
}

torch::Tensor ibot_ce_fact_bwd(torch::Tensor g, torch::Tensor x, torch::Tensor xt,
                               torch::Tensor u, torch::Tensor v, torch::Tensor w,
                               torch::Tensor lse, torch::Tensor st, double temp,
                               double tt) {
  const int M = x.size(0);
  const long K = x.size(1);
  auto dx = torch::empty_like(x);
  launch_ibot_ce_fact_bwd((const __hip_bfloat16*)x.data_ptr(),
                          (const __hip_bfloat16*)xt.data_ptr(), u.data_ptr<float>(),
                          v.data_ptr<float>(), w.data_ptr<float>(), lse.data_ptr<float>(),
                          st.data_ptr<float>(), g.data_ptr<float>(),
                          (__hip_bfloat16*)dx.data_ptr(), M, K, (float)(1.0 / temp),
                          (float)(1.0 / tt), current_stream());
  return dx;
}

std::vector<torch::Tensor> dino_ce_fact_fwd(torch::Tensor x, torch::Tensor xt,
                                            torch::Tensor u, torch::Tensor v, double temp,
                                            double tt, bool ignore_diag) {
  CHECK_INPUT(x);
  CHECK_INPUT(xt);
  const int S = x.size(0), B = x.size(1);
  const int T = xt.size(0);
  const long K = x.size(2);
  auto lse = torch::empty({S * B}, x.options().dtype(torch::kFloat));
  auto st = torch::empty({S * B}, x.options().dtype(torch::kFloat));
  auto loss = torch::zeros({}, x.options().dtype(torch::kFloat));
  launch_dino_ce_fact_fwd((const __hip_bfloat16*)x.data_ptr(),
                          (const __hip_bfloat16*)xt.data_ptr(), u.data_ptr<float>(),
                          v.data_ptr<float>(), lse.data_ptr<float>(), st.data_ptr<float>(),
                          loss.data_ptr<float>(), S, T, B, K, (float)(1.0 / temp),
                          (float)(1.0 / tt), ignore_diag, current_stream());
  return {loss, lse, st};
}

torch::Tensor dino_ce_fact_bwd(torch::Tensor g, torch::Tensor x, torch::Tensor xt,
                               torch::Tensor u, torch::Tensor v, torch::Tensor lse,
                               torch::Tensor st, double temp, double tt,
                               bool ignore_diag) {
  const int S = x.size(0), B = x.size(1);
  const int T = xt.size(0);
  const long K = x.size(2);
  auto dx = torch::empty_like(x);
  launch_dino_ce_fact_bwd((const __hip_bfloat16*)x.data_ptr(),
                          (const __hip_bfloat16*)xt.data_ptr(), u.data_ptr<float>(),
                          v.data_ptr<float>(), lse.data_ptr<float>(), st.data_ptr<float>(),
                          g.data_ptr<float>(), (__hip_bfloat16*)dx.data_ptr(), S, T, B, K,
                          (float)(1.0 / temp), (float)(1.0 / tt), ignore_diag,
                          current_stream());
  return dx;
}

std::vector<torch::Tensor> sinkhorn_exp(torch::Tensor x, double temp) {
  CHECK_INPUT(x);
  auto Q = torch::empty(x.sizes(), x.options().dtype(torch::kFloat));
  auto total = torch::zeros({}, x.options().dtype(torch::kFloat));
  launch_sinkhorn_exp((const __hip_bfloat16*)x.data_ptr(), Q.data_ptr<float>(),
                      total.data_ptr<float>(), x.numel(), (float)(1.0 / temp),
                      current_stream());
  return {Q, total};
}

torch::Tensor sinkhorn_colsum(torch::Tensor Q, torch::Tensor divisor) {
  const int M = Q.size(0);
  const long K = Q.size(1);
  auto out = torch::empty({K}, Q.options());
  launch_sinkhorn_colsum(Q.data_ptr<float>(), out.data_ptr<float>(), M, K,
                         divisor.data_ptr<float>(), current_stream());
  return out;
}

void sinkhorn_div_row(torch::Tensor Q, torch::Tensor col, double k_div, torch::Tensor B,
                      bool scale_back) {
  const int M = Q.size(0);
  const long K = Q.size(1);
  launch_sinkhorn_div_row(Q.data_ptr<float>(), col.data_ptr<float>(), M, K, (float)k_div,
                          B.data_ptr<float>(), scale_back, current_stream());
}

// -------------------------------- fmha ----------------------------------

torch::Tensor probe_mfma(torch::Tensor A, torch::Tensor B) {
  CHECK_INPUT(A);
  CHECK_INPUT(B);
  TORCH_CHECK(A.scalar_type() == at::ScalarType::BFloat16);
  auto C = torch::empty({32, 32}, A.options().dtype(torch::kFloat));
  launch_probe_mfma((const __hip_bfloat16*)A.data_ptr(), (const __hip_bfloat16*)B.data_ptr(),
                    C.data_ptr<float>(), current_stream());
  return C;
}

std::vector<torch::Tensor> fmha_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v) {
  CHECK_INPUT(q);
  CHECK_INPUT(k);
  CHECK_INPUT(v);
  TORCH_CHECK(q.scalar_type() == at::ScalarType::BFloat16, "fmha: bf16 only");
  TORCH_CHECK(q.dim() == 4, "fmha expects [B, H, N, hd]");
  const int B = q.size(0), H = q.size(1), N = q.size(2), HD = q.size(3);
  TORCH_CHECK(HD == 64 || HD == 128, "fmha: head_dim must be 64 or 128, got ", HD);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, N}, q.options().dtype(torch::kFloat));
  const float scale = 1.0f / std::sqrt((float)HD);
  launch_fmha_fwd((const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                  (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)o.data_ptr(),
                  lse.data_ptr<float>(), B * H, N, HD, scale, current_stream());
  return {o, lse};
}

std::vector<torch::Tensor> fmha_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o, torch::Tensor lse) {
  CHECK_INPUT(dout);
  const int B = q.size(0), H = q.size(1), N = q.size(2), HD = q.size(3);
  const float scale = 1.0f / std::sqrt((float)HD);
  auto D = torch::empty({B, H, N}, q.options().dtype(torch::kFloat));
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto stream = current_stream();
  launch_fmha_bwd_pre((const __hip_bfloat16*)dout.data_ptr(),
                      (const __hip_bfloat16*)o.data_ptr(), D.data_ptr<float>(),
                      (long)B * H * N, HD, stream);
  launch_fmha_bwd_dq((const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(),
                     (const __hip_bfloat16*)dout.data_ptr(), lse.data_ptr<float>(),
                     D.data_ptr<float>(), (__hip_bfloat16*)dq.data_ptr(), B * H, N, HD,
                     scale, stream);
  launch_fmha_bwd_dkv((const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                      (const __hip_bfloat16*)v.data_ptr(),
                      (const __hip_bfloat16*)dout.data_ptr(), lse.data_ptr<float>(),
                      D.data_ptr<float>(), (__hip_bfloat16*)dk.data_ptr(),
                      (__hip_bfloat16*)dv.data_ptr(), B * H, N, HD, scale, stream);
  return {dq, dk, dv};
}

// ----------------------------- patch embed ------------------------------

torch::Tensor patch_embed_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                              int64_t patch) {
  CHECK_INPUT(x);
  CHECK_INPUT(w);
  TORCH_CHECK(x.dim() == 4 && x.scalar_type() == at::ScalarType::BFloat16);
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int D = w.size(0);
  TORCH_CHECK(C == 3, "patch_embed_fwd kernel supports C=3 (any patch size)");
  TORCH_CHECK(H % patch == 0 && W % patch == 0, "H/W must be multiples of patch");
  const long rows = (long)B * (H / patch) * (W / patch);
  auto out = torch::empty({(long)B, rows / B, (long)D}, x.options());
  launch_patch_embed_fwd((const __hip_bfloat16*)x.data_ptr(),
                         (const __hip_bfloat16*)w.data_ptr(),
                         (const __hip_bfloat16*)bias.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(), B, C, H, W, (int)patch, D,
                         current_stream());
  return out;
}

// ----------------------------- fmha + rope ------------------------------

std::vector<torch::Tensor> fmha_rope_fwd_out(torch::Tensor qkv, torch::Tensor sin_t,
                                             torch::Tensor cos_t, int64_t prefix,
                                             torch::Tensor out_o) {
  CHECK_INPUT(qkv);
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3, "qkv must be [B, N, 3, H, hd]");
  TORCH_CHECK(qkv.scalar_type() == at::ScalarType::BFloat16);
  const int B = qkv.size(0), N = qkv.size(1), H = qkv.size(3), HD = qkv.size(4);
  TORCH_CHECK(HD == 64 || HD == 128, "head_dim must be 64 or 128");
  const bool has_rope = sin_t.defined() && sin_t.numel() > 0;
  const int P = has_rope ? (int)sin_t.size(0) : N - (int)prefix;
  if (has_rope) {
    TORCH_CHECK(sin_t.scalar_type() == at::ScalarType::Float && sin_t.is_contiguous());
    TORCH_CHECK(P == N - prefix, "rope table rows must equal N - prefix");
  }
  torch::Tensor o = out_o.defined() && out_o.numel() > 0 ? out_o
                                                        : torch::empty({B, N, H, HD}, qkv.options());
  TORCH_CHECK(o.is_contiguous() && o.numel() == (long)B * N * H * HD);
  auto lse = torch::empty({B, H, N}, qkv.options().dtype(torch::kFloat));
  const float scale = 1.0f / std::sqrt((float)HD);
  launch_fmha_rope_fwd((const __hip_bfloat16*)qkv.data_ptr(),
                       has_rope ? sin_t.data_ptr<float>() : nullptr,
                       has_rope ? cos_t.data_ptr<float>() : nullptr,
                       (__hip_bfloat16*)o.data_ptr(), lse.data_ptr<float>(), B, H, N, P,
                       HD, scale, current_stream());
  return {o, lse};
}

torch::Tensor fmha_rope_bwd_out(torch::Tensor dout, torch::Tensor qkv, torch::Tensor o,
                                torch::Tensor lse, torch::Tensor sin_t,
                                torch::Tensor cos_t, int64_t prefix,
                                torch::Tensor out_dqkv) {
  CHECK_INPUT(dout);
  const int B = qkv.size(0), N = qkv.size(1), H = qkv.size(3), HD = qkv.size(4);
  const bool has_rope = sin_t.defined() && sin_t.numel() > 0;
  const int P = N - (int)prefix;
  const float scale = 1.0f / std::sqrt((float)HD);
  auto D = torch::empty({B, H, N}, qkv.options().dtype(torch::kFloat));
  torch::Tensor dqkv = out_dqkv.defined() && out_dqkv.numel() > 0 ? out_dqkv
                                                                  : torch::empty_like(qkv);
  auto stream = current_stream();
  launch_fmha_rope_bwd_pre((const __hip_bfloat16*)dout.data_ptr(),
                           (const __hip_bfloat16*)o.data_ptr(), D.data_ptr<float>(), B, H,
                           N, HD, stream);
  launch_fmha_rope_bwd_dq((const __hip_bfloat16*)qkv.data_ptr(),
                          (const __hip_bfloat16*)dout.data_ptr(),
                          has_rope ? sin_t.data_ptr<float>() : nullptr,
                          has_rope ? cos_t.data_ptr<float>() : nullptr,
                          lse.data_ptr<float>(), D.data_ptr<float>(),
                          (__hip_bfloat16*)dqkv.data_ptr(), B, H, N, P, HD, scale, stream);
  launch_fmha_rope_bwd_dkv((const __hip_bfloat16*)qkv.data_ptr(),
                           (const __hip_bfloat16*)dout.data_ptr(),
                           has_rope ? sin_t.data_ptr<float>() : nullptr,
                           has_rope ? cos_t.data_ptr<float>() : nullptr,
                           lse.data_ptr<float>(), D.data_ptr<float>(),
                           (__hip_bfloat16*)dqkv.data_ptr(), B, H, N, P, HD, scale,
                           stream);
  return dqkv;
}


// planned single-launch variants: tables are prebuilt device tensors owned by
// the Python-side MultiTensorPlan (no per-call table construction/upload).

void multi_tensor_ema_planned(torch::Tensor ptrs, torch::Tensor sizes, torch::Tensor ct,
                              torch::Tensor co, int64_t n_tensors, double m,
                              bool is_bf16) {
  const int n_chunks = (int)ct.numel();
  long* pp = ptrs.data_ptr<long>();
  if (is_bf16) {
    launch_multi_tensor_ema<__hip_bfloat16>(
        (__hip_bfloat16* const*)pp, (const __hip_bfloat16* const*)(pp + n_tensors),
        sizes.data_ptr<long>(), ct.data_ptr<int>(), co.data_ptr<long>(), n_chunks,
        (float)m, current_stream());
  } else {
    launch_multi_tensor_ema<float>(
        (float* const*)pp, (const float* const*)(pp + n_tensors), sizes.data_ptr<long>(),
        ct.data_ptr<int>(), co.data_ptr<long>(), n_chunks, (float)m, current_stream());
  }
}

void multi_tensor_adamw_planned(torch::Tensor ptrs, torch::Tensor sizes, torch::Tensor ct,
                                torch::Tensor co, int64_t n_tensors, torch::Tensor lr_mult,
                                torch::Tensor wd_mult, torch::Tensor is_last,
                                torch::Tensor sub_id, torch::Tensor clip, double lr,
                                double last_lr, double wd, double beta1, double beta2,
                                double eps, double bc1, double bc2, bool has_master,
                                bool is_bf16, bool grad_is_f32) {
  const int n_chunks = (int)ct.numel();
  long* pp = ptrs.data_ptr<long>();
  const long n = n_tensors;
  if (is_bf16 && grad_is_f32) {
    launch_multi_tensor_adamw_planned<__hip_bfloat16, float>(
        (__hip_bfloat16* const*)pp, (const float* const*)(pp + n),
        (float* const*)(pp + 2 * n), (float* const*)(pp + 3 * n),
        has_master ? (float* const*)(pp + 4 * n) : nullptr, sizes.data_ptr<long>(),
        ct.data_ptr<int>(), co.data_ptr<long>(), n_chunks, lr_mult.data_ptr<float>(),
        wd_mult.data_ptr<float>(), is_last.data_ptr<float>(), sub_id.data_ptr<int>(),
        clip.data_ptr<float>(), (float)lr, (float)last_lr, (float)wd, (float)beta1,
        (float)beta2, (float)eps, (float)bc1, (float)bc2, has_master, current_stream());
  } else if (is_bf16) {
    launch_multi_tensor_adamw_planned<__hip_bfloat16, __hip_bfloat16>(
        (__hip_bfloat16* const*)pp, (const __hip_bfloat16* const*)(pp + n),
        (float* const*)(pp + 2 * n), (float* const*)(pp + 3 * n),
        has_master ? (float* const*)(pp + 4 * n) : nullptr, sizes.data_ptr<long>(),
        ct.data_ptr<int>(), co.data_ptr<long>(), n_chunks, lr_mult.data_ptr<float>(),
        wd_mult.data_ptr<float>(), is_last.data_ptr<float>(), sub_id.data_ptr<int>(),
        clip.data_ptr<float>(), (float)lr, (float)last_lr, (float)wd, (float)beta1,
        (float)beta2, (float)eps, (float)bc1, (float)bc2, has_master, current_stream());
  } else {
    launch_multi_tensor_adamw_planned<float, float>(
        (float* const*)pp, (const float* const*)(pp + n), (float* const*)(pp + 2 * n),
        (float* const*)(pp + 3 * n), has_master ? (float* const*)(pp + 4 * n) : nullptr,
        sizes.data_ptr<long>(), ct.data_ptr<int>(), co.data_ptr<long>(), n_chunks,
        lr_mult.data_ptr<float>(), wd_mult.data_ptr<float>(), is_last.data_ptr<float>(),
        sub_id.data_ptr<int>(), clip.data_ptr<float>(), (float)lr, (float)last_lr,
        (float)wd, (float)beta1, (float)beta2, (float)eps, (float)bc1, (float)bc2,
        has_master, current_stream());
  }
}

torch::Tensor multi_tensor_l2norm_planned(torch::Tensor ptrs, torch::Tensor sizes,
                                          torch::Tensor ct, torch::Tensor co,
                                          torch::Tensor sub_id, int64_t n_submodels,
                                          bool is_bf16) {
  const int n_chunks = (int)ct.numel();
  auto out = torch::zeros({n_submodels}, torch::dtype(torch::kFloat).device(ptrs.device()));
  long* pp = ptrs.data_ptr<long>();
  if (is_bf16) {
    launch_multi_tensor_l2norm_planned<__hip_bfloat16>(
        (const __hip_bfloat16* const*)pp, sizes.data_ptr<long>(), ct.data_ptr<int>(),
        co.data_ptr<long>(), n_chunks, sub_id.data_ptr<int>(), out.data_ptr<float>(),
        current_stream());
  } else {
    launch_multi_tensor_l2norm_planned<float>(
        (const float* const*)pp, sizes.data_ptr<long>(), ct.data_ptr<int>(),
        co.data_ptr<long>(), n_chunks, sub_id.data_ptr<int>(), out.data_ptr<float>(),
        current_stream());
  }
  return out;
}

// ---------------------------- multi-tensor ------------------------------

struct MTTables {
  torch::Tensor device_buf;  // holds sizes + chunk tables + ptr arrays
  long* sizes;
  int* chunk_tensor;
  long* chunk_offset;
  long* ptr_arrays;  // base of pointer storage (n_lists * n_tensors int64)
  int n_chunks;
  int n_tensors;
};

constexpr long kMTChunk = 65536;

MTTables build_tables(const std::vector<std::vector<torch::Tensor>>& lists,
                      torch::Device device) {
  const int n_tensors = (int)lists[0].size();
  const int n_lists = (int)lists.size();
  std::vector<long> sizes(n_tensors);
  std::vector<int> chunk_tensor;
  std::vector<long> chunk_offset;
  for (int t = 0; t < n_tensors; ++t) {
    sizes[t] = lists[0][t].numel();
    for (long off = 0; off < sizes[t]; off += kMTChunk) {
      chunk_tensor.push_back(t);
      chunk_offset.push_back(off);
    }
  }
  const int n_chunks = (int)chunk_tensor.size();
  // layout (int64 slots): [sizes n_tensors][chunk_tensor n_chunks (as i64)]
  //                       [chunk_offset n_chunks][ptrs n_lists*n_tensors]
  const long total = n_tensors + n_chunks + n_chunks + (long)n_lists * n_tensors;
  auto host = torch::empty({total}, torch::dtype(torch::kLong).pinned_memory(true));
  long* hp = host.data_ptr<long>();
  long* p_sizes = hp;
  long* p_ct = hp + n_tensors;
  long* p_co = p_ct + n_chunks;
  long* p_ptr = p_co + n_chunks;
  for (int t = 0; t < n_tensors; ++t) p_sizes[t] = sizes[t];
  for (int c = 0; c < n_chunks; ++c) {
    p_ct[c] = chunk_tensor[c];
    p_co[c] = chunk_offset[c];
  }
  for (int l = 0; l < n_lists; ++l)
    for (int t = 0; t < n_tensors; ++t)
      p_ptr[(long)l * n_tensors + t] = (long)(uintptr_t)lists[l][t].data_ptr();
  auto dev = host.to(device, /*non_blocking=*/true);
  MTTables out;
  out.device_buf = dev;
  long* dp = dev.data_ptr<long>();
  out.sizes = dp;
  out.chunk_tensor = nullptr;  // stored as i64; kernel wants int — handled below
  out.chunk_offset = dp + n_tensors + n_chunks;
  out.ptr_arrays = dp + n_tensors + 2 * (long)n_chunks;
  out.n_chunks = n_chunks;
  out.n_tensors = n_tensors;
  return out;
}

// The chunk_tensor table must be int32 for the kernels; build a separate one.
torch::Tensor build_chunk_tensor_i32(const std::vector<long>& sizes, torch::Device device,
                                     int* n_chunks_out) {
  std::vector<int> ct;
  for (int t = 0; t < (int)sizes.size(); ++t)
    for (long off = 0; off < sizes[t]; off += kMTChunk) ct.push_back(t);
  *n_chunks_out = (int)ct.size();
  auto host = torch::from_blob(ct.data(), {(long)ct.size()}, torch::kInt).clone();
  return host.to(device);
}

void multi_tensor_ema(std::vector<torch::Tensor> t_list, std::vector<torch::Tensor> s_list,
                      double m) {
  TORCH_CHECK(!t_list.empty() && t_list.size() == s_list.size());
  auto device = t_list[0].device();
  auto tables = build_tables({t_list, s_list}, device);
  std::vector<long> sizes(tables.n_tensors);
  for (int t = 0; t < tables.n_tensors; ++t) sizes[t] = t_list[t].numel();
  int n_chunks;
  auto ct32 = build_chunk_tensor_i32(sizes, device, &n_chunks);
  auto stream = current_stream();
  DISPATCH_FLOAT_BF16(t_list[0].scalar_type(), "multi_tensor_ema", [&] {
    launch_multi_tensor_ema<scalar_t>(
        (scalar_t* const*)(tables.ptr_arrays),
        (const scalar_t* const*)(tables.ptr_arrays + tables.n_tensors),
        tables.sizes, ct32.data_ptr<int>(), tables.chunk_offset, n_chunks, (float)m, stream);
  });
}

void multi_tensor_adamw(std::vector<torch::Tensor> p, std::vector<torch::Tensor> g,
                        std::vector<torch::Tensor> m, std::vector<torch::Tensor> v,
                        std::vector<torch::Tensor> w, double lr, double beta1, double beta2,
                        double eps, double weight_decay, double bc1, double bc2,
                        double grad_scale) {
  TORCH_CHECK(!p.empty());
  const bool has_master = !w.empty();
  auto device = p[0].device();
  std::vector<std::vector<torch::Tensor>> lists = {p, g, m, v};
  if (has_master) lists.push_back(w);
  auto tables = build_tables(lists, device);
  std::vector<long> sizes(tables.n_tensors);
  for (int t = 0; t < tables.n_tensors; ++t) sizes[t] = p[t].numel();
  int n_chunks;
  auto ct32 = build_chunk_tensor_i32(sizes, device, &n_chunks);
  auto stream = current_stream();
  const int nt = tables.n_tensors;
  DISPATCH_FLOAT_BF16(p[0].scalar_type(), "multi_tensor_adamw", [&] {
    launch_multi_tensor_adamw<scalar_t>(
        (scalar_t* const*)(tables.ptr_arrays),
        (const scalar_t* const*)(tables.ptr_arrays + nt),
        (float* const*)(tables.ptr_arrays + 2L * nt),
        (float* const*)(tables.ptr_arrays + 3L * nt),
        has_master ? (float* const*)(tables.ptr_arrays + 4L * nt) : nullptr,
        tables.sizes, ct32.data_ptr<int>(), tables.chunk_offset, n_chunks, (float)lr,
        (float)beta1, (float)beta2, (float)eps, (float)weight_decay, (float)bc1, (float)bc2,
        (float)grad_scale, has_master, stream);
  });
}

torch::Tensor multi_tensor_l2norm_sq(std::vector<torch::Tensor> g) {
  TORCH_CHECK(!g.empty());
  auto device = g[0].device();
  auto tables = build_tables({g}, device);
  std::vector<long> sizes(tables.n_tensors);
  for (int t = 0; t < tables.n_tensors; ++t) sizes[t] = g[t].numel();
  int n_chunks;
  auto ct32 = build_chunk_tensor_i32(sizes, device, &n_chunks);
  auto out = torch::zeros({}, torch::dtype(torch::kFloat).device(device));
  DISPATCH_FLOAT_BF16(g[0].scalar_type(), "multi_tensor_l2norm_sq", [&] {
    launch_multi_tensor_l2norm_sq<scalar_t>(
        (const scalar_t* const*)(tables.ptr_arrays), tables.sizes, ct32.data_ptr<int>(),
        tables.chunk_offset, n_chunks, out.data_ptr<float>(), current_stream());
  });
  return out;
}

}  // namespace

// hipBLASLt fused-epilogue entry points (blaslt_ext.hip)
std::vector<torch::Tensor> blaslt_gemm_bias_gelu_fwd(torch::Tensor x, torch::Tensor w,
                                                     torch::Tensor bias);
std::vector<torch::Tensor> blaslt_gemm_dgelu_bgrad(torch::Tensor dy, torch::Tensor w2,
                                                   torch::Tensor pre);
torch::Tensor blaslt_gemm_bias(torch::Tensor x, torch::Tensor w, torch::Tensor bias);
int64_t blaslt_probe_epilogue(int64_t epi, int64_t aux_type, int64_t bias_type,
                              int64_t m, int64_t n, int64_t k);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("blaslt_gemm_bias_gelu_fwd", &blaslt_gemm_bias_gelu_fwd);
  mod.def("blaslt_gemm_dgelu_bgrad", &blaslt_gemm_dgelu_bgrad);
  mod.def("blaslt_gemm_bias", &blaslt_gemm_bias);
  mod.def("blaslt_probe_epilogue", &blaslt_probe_epilogue);
  mod.def("layernorm_fwd", &layernorm_fwd);
  mod.def("layernorm_bwd", &layernorm_bwd);
  mod.def("rmsnorm_fwd", &rmsnorm_fwd);
  mod.def("rmsnorm_bwd", &rmsnorm_bwd);
  mod.def("l2norm_fwd", &l2norm_fwd);
  mod.def("l2norm_bwd", &l2norm_bwd);
  mod.def("bias_gelu_fwd", &bias_gelu_fwd);
  mod.def("bias_gelu_bwd", &bias_gelu_bwd);
  mod.def("row_gather", &row_gather);
  mod.def("row_scatter_add_", &row_scatter_add_);
  mod.def("row_gather_scaled", &row_gather_scaled);
  mod.def("ls_axpy_fwd", &ls_axpy_fwd);
  mod.def("ls_axpy_bias_fwd", &ls_axpy_bias_fwd);
  mod.def("ls_axpy_bias_bwd", &ls_axpy_bias_bwd);
  mod.def("ls_scatter_add_", &ls_scatter_add_);
  mod.def("ls_scatter_bwd", &ls_scatter_bwd);
  mod.def("ls_axpy_bwd", &ls_axpy_bwd);
  mod.def("swiglu_fwd", &swiglu_fwd);
  mod.def("swiglu_bwd", &swiglu_bwd);
  mod.def("rope_fwd", &rope_fwd);
  mod.def("probe_mfma", &probe_mfma);
  mod.def("patch_embed_fwd", &patch_embed_fwd);
  mod.def("dino_ce_fwd", &dino_ce_fwd);
  mod.def("dino_ce_bwd", &dino_ce_bwd);
  mod.def("ibot_ce_fwd", &ibot_ce_fwd);
  mod.def("ibot_ce_bwd", &ibot_ce_bwd);
  mod.def("sinkhorn_exp", &sinkhorn_exp);
  mod.def("sinkhorn_fact_colsum", &sinkhorn_fact_colsum);
  mod.def("sinkhorn_fact_rowsum", &sinkhorn_fact_rowsum);
  mod.def("ibot_ce_fact_fwd", &ibot_ce_fact_fwd);
  mod.def("ibot_ce_fact_bwd", &ibot_ce_fact_bwd);
  mod.def("dino_ce_fact_fwd", &dino_ce_fact_fwd);
  mod.def("dino_ce_fact_bwd", &dino_ce_fact_bwd);
  mod.def("sinkhorn_colsum", &sinkhorn_colsum);
  mod.def("sinkhorn_div_row", &sinkhorn_div_row);
  mod.def("fmha_fwd", &fmha_fwd);
  mod.def("fmha_rope_fwd", [](torch::Tensor qkv, torch::Tensor sin_t, torch::Tensor cos_t,
                              int64_t prefix) {
    return fmha_rope_fwd_out(qkv, sin_t, cos_t, prefix, torch::Tensor());
  });
  mod.def("fmha_rope_fwd_out", &fmha_rope_fwd_out);
  mod.def("fmha_rope_bwd", [](torch::Tensor dout, torch::Tensor qkv, torch::Tensor o,
                              torch::Tensor lse, torch::Tensor sin_t, torch::Tensor cos_t,
                              int64_t prefix) {
    return fmha_rope_bwd_out(dout, qkv, o, lse, sin_t, cos_t, prefix, torch::Tensor());
  });
  mod.def("fmha_rope_bwd_out", &fmha_rope_bwd_out);
  mod.def("fmha_bwd", &fmha_bwd);
  mod.def("multi_tensor_ema", &multi_tensor_ema);
  mod.def("multi_tensor_ema_planned", &multi_tensor_ema_planned);
  mod.def("multi_tensor_adamw_planned", &multi_tensor_adamw_planned);
  mod.def("multi_tensor_l2norm_planned", &multi_tensor_l2norm_planned);
  mod.def("multi_tensor_adamw", &multi_tensor_adamw);
  mod.def("multi_tensor_l2norm_sq", &multi_tensor_l2norm_sq);
}
