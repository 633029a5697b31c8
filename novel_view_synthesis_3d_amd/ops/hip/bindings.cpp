// torch op registration for the nvs3d gfx950 HIP kernels.
#include <torch/extension.h>

std::vector<torch::Tensor> gn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta,
                                  c10::optional<torch::Tensor> film,
                                  int64_t groups, double eps, bool silu,
                                  double p_drop,
                                  c10::optional<torch::Tensor> drop_seed);
std::vector<torch::Tensor> gn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  c10::optional<torch::Tensor> film,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  int64_t groups, bool silu,
                                  double p_drop,
                                  c10::optional<torch::Tensor> drop_seed);
torch::Tensor rays_posenc(torch::Tensor R, torch::Tensor t, torch::Tensor Kinv,
                          c10::optional<torch::Tensor> mask,
                          int64_t H, int64_t W, torch::ScalarType out_dtype);
void fused_adam(torch::Tensor ptrs, torch::Tensor chunk_tensor,
                torch::Tensor chunk_off, torch::Tensor numels,
                double lr, double b1, double b2, double eps, int64_t step);
torch::Tensor conv3x3_fwd(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> residual,
                          double out_scale);
std::vector<torch::Tensor> conv3x3_wgrad(torch::Tensor x, torch::Tensor dy,
                                         bool with_bias);
std::vector<torch::Tensor> linear_wgrad(torch::Tensor dy, torch::Tensor x,
                                        bool with_bias);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool kv_swap);
torch::Tensor attn_delta(torch::Tensor dout, torch::Tensor o);
std::vector<torch::Tensor> attn_bwd_fused(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, torch::Tensor dout,
                                          torch::Tensor lse,
                                          torch::Tensor delta, bool kv_swap);
torch::Tensor im2col3x3(torch::Tensor x, int64_t stride, int64_t nplanes,
                        int64_t m0, int64_t m1,
                        c10::optional<torch::Tensor> out_buf);
torch::Tensor attn_p_from_lse(torch::Tensor s, torch::Tensor lse,
                              double scale);
torch::Tensor attn_ds(torch::Tensor p, torch::Tensor dp, torch::Tensor delta,
                      double scale);
torch::Tensor up2x_fwd(torch::Tensor x);
torch::Tensor up2x_bwd(torch::Tensor dout);
torch::Tensor pool2x_fwd(torch::Tensor x);
torch::Tensor pool2x_bwd(torch::Tensor dout);
torch::Tensor add_scale(torch::Tensor a, torch::Tensor b, double scale);

torch::Tensor rays_posenc_py(torch::Tensor R, torch::Tensor t,
                             torch::Tensor Kinv,
                             c10::optional<torch::Tensor> mask,
                             int64_t H, int64_t W, torch::Tensor dtype_like) {
  return rays_posenc(R, t, Kinv, mask, H, W, dtype_like.scalar_type());
}

TORCH_LIBRARY(nvs3d, m) {
  m.def("gn_fwd(Tensor x, Tensor gamma, Tensor beta, Tensor? film, "
        "int groups, float eps, bool silu, float p_drop, "
        "Tensor? drop_seed) -> Tensor[]");
  m.def("gn_bwd(Tensor dy, Tensor x, Tensor gamma, Tensor beta, "
        "Tensor? film, Tensor mean, Tensor rstd, "
        "int groups, bool silu, float p_drop, Tensor? drop_seed) -> Tensor[]");
  m.def("rays_posenc(Tensor R, Tensor t, Tensor Kinv, Tensor? mask, "
        "int H, int W, Tensor dtype_like) -> Tensor");
  m.def("fused_adam(Tensor ptrs, Tensor chunk_tensor, Tensor chunk_off, "
        "Tensor numels, float lr, float b1, float b2, float eps, "
        "int step) -> ()");
  m.def("conv3x3_fwd(Tensor x, Tensor w, Tensor? bias, Tensor? residual, float out_scale) -> Tensor");
  m.def("conv3x3_wgrad(Tensor x, Tensor dy, bool with_bias) -> Tensor[]");
  m.def("linear_wgrad(Tensor dy, Tensor x, bool with_bias) -> Tensor[]");
  m.def("attn_fwd(Tensor q, Tensor k, Tensor v, bool kv_swap=False) -> Tensor[]");
  m.def("attn_delta(Tensor dout, Tensor o) -> Tensor");
  m.def("attn_bwd_fused(Tensor q, Tensor k, Tensor v, Tensor dout, "
        "Tensor lse, Tensor delta, bool kv_swap=False) -> Tensor[]");
  m.def("im2col3x3(Tensor x, int stride, int nplanes, int m0, int m1, Tensor? out_buf) -> Tensor");
  m.def("attn_p_from_lse(Tensor s, Tensor lse, float scale) -> Tensor");
  m.def("attn_ds(Tensor p, Tensor dp, Tensor delta, float scale) -> Tensor");
  m.def("up2x_fwd(Tensor x) -> Tensor");
  m.def("up2x_bwd(Tensor dout) -> Tensor");
  m.def("pool2x_fwd(Tensor x) -> Tensor");
  m.def("pool2x_bwd(Tensor dout) -> Tensor");
  m.def("add_scale(Tensor a, Tensor b, float scale) -> Tensor");
}

TORCH_LIBRARY_IMPL(nvs3d, CUDA, m) {
  m.impl("gn_fwd", gn_fwd);
  m.impl("gn_bwd", gn_bwd);
  m.impl("rays_posenc", rays_posenc_py);
  m.impl("fused_adam", fused_adam);
  m.impl("conv3x3_fwd", conv3x3_fwd);
  m.impl("conv3x3_wgrad", conv3x3_wgrad);
  m.impl("linear_wgrad", linear_wgrad);
  m.impl("attn_fwd", attn_fwd);
  m.impl("attn_delta", attn_delta);
  m.impl("attn_bwd_fused", attn_bwd_fused);
  m.impl("im2col3x3", im2col3x3);
  m.impl("attn_p_from_lse", attn_p_from_lse);
  m.impl("attn_ds", attn_ds);
  m.impl("up2x_fwd", up2x_fwd);
  m.impl("up2x_bwd", up2x_bwd);
  m.impl("pool2x_fwd", pool2x_fwd);
  m.impl("pool2x_bwd", pool2x_bwd);
  m.impl("add_scale", add_scale);
}
