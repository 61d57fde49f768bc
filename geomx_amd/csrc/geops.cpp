// Torch bindings for the geomx_amd gfx950 kernel library (kernels.hip).
//
// Thin checked wrappers: dtype/contiguity/device checks, stream plumbing
// (kernels run on the current torch stream so they order correctly with
// autograd/backward and RCCL collectives), and workspace management for
// the Bi-Sparse pack pipeline.

#include <torch/extension.h>

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#include <cstdint>

// C-ABI launchers from kernels.hip
extern "C" {
void geops_quantize_2bit(const float*, float*, uint32_t*, long long, float,
                         hipStream_t);
void geops_dequantize_2bit(const uint32_t*, float*, long long, float,
                           hipStream_t);
void geops_bsc_momentum(const float*, float*, float*, float, long long,
                        hipStream_t);
void geops_bsc_pack(const float*, float*, float*, float*, int*, long long*,
                    float, long long, long long, float, bool, hipStream_t);
void geops_bsc_pull_pack(const float*, float*, int*, long long*, long long,
                         long long, float, hipStream_t);
int geops_bsc_fused(const float*, float*, float*, float*, int*, const float*,
                    unsigned long long*, float, long long, long long, float,
                    hipStream_t);
void geops_bsc_unpack(const float*, const int*, float*, long long, bool,
                      hipStream_t);
void geops_dgt_contribution(const float*, float*, long long, int, int,
                            hipStream_t);
void geops_quantize_4bit(const float*, float*, uint8_t*, float*, long long,
                         int, int, bool, hipStream_t);
void geops_dequantize_4bit(const uint8_t*, const float*, float*, long long,
                           int, hipStream_t);
void geops_sgd_update(float*, const float*, float, float, float, long long,
                      hipStream_t);
void geops_sgd_mom_update(float*, const float*, float*, float, float, float,
                          float, long long, hipStream_t);
void geops_adam_update(float*, const float*, float*, float*, float, float,
                       float, float, float, float, long long, hipStream_t);
void geops_dcasgd_update(float*, const float*, float*, float*, float, float,
                         float, float, float, long long, bool, hipStream_t);
void geops_rmsprop_update(float*, const float*, float*, float, float, float,
                          float, float, long long, hipStream_t);
void geops_adagrad_update(float*, const float*, float*, float, float, float,
                          float, long long, hipStream_t);
void geops_signsgd_update(float*, const float*, float, float, float,
                          long long, hipStream_t);
void geops_signum_update(float*, const float*, float*, float, float, float,
                         float, long long, hipStream_t);
void geops_relu_maxpool2_fwd(const unsigned short*, unsigned short*, uint8_t*,
                             long long, int, int, int, hipStream_t);
void geops_relu_maxpool2_bwd(const unsigned short*, const uint8_t*,
                             unsigned short*, long long, int, int, int,
                             hipStream_t);
int geops_conv5_nhwc(const unsigned short*, const unsigned short*,
                     const float*, unsigned short*, int, int, int, int, int,
                     int, int, int, hipStream_t);
int geops_conv5_pool_nhwc(const unsigned short*, const unsigned short*,
                          const float*, unsigned short*, uint8_t*, int, int,
                          int, int, int, int, int, hipStream_t);
void geops_pad_ch3to4_nhwc(const void*, unsigned short*, long long, int,
                           hipStream_t);
int geops_conv5_wrw_nhwc(const unsigned short*, const unsigned short*,
                         float*, int, int, int, int, int, int, int, int,
                         hipStream_t);
int geops_conv5_wrw16_nhwc(const unsigned short*, const unsigned short*,
                           float*, int, int, int, int, int, int, int, int,
                           hipStream_t);
int geops_conv5_wrw4_nhwc(const unsigned short*, const unsigned short*,
                          float*, int, int, int, int, int, int, int, int,
                          hipStream_t);
void geops_tr16_probe(const unsigned short*, unsigned short*, hipStream_t);
void geops_tr16_probe2(const unsigned short*, const int*, unsigned short*,
                       hipStream_t);
int geops_wrw2_dump(const unsigned short*, const unsigned short*,
                    unsigned short*, int, int, int, int, int, int,
                    hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

// surface async launch errors at the call site instead of a later,
// unrelated synchronize
void launch_check(const char* what) {
  const hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, what, ": kernel launch failed: ",
              hipGetErrorString(e));
}

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(((uintptr_t)t.data_ptr() & 15) == 0, name,
              " must be 16-byte aligned");
}

void quantize_2bit(torch::Tensor grad, torch::Tensor residual,
                   torch::Tensor out, double thr) {
  check_f32(grad, "grad");
  check_f32(residual, "residual");
  TORCH_CHECK(out.scalar_type() == torch::kInt32 && out.is_contiguous());
  const long long n = grad.numel();
  TORCH_CHECK(residual.numel() == n);
  TORCH_CHECK(out.numel() == (n + 15) / 16);
  geops_quantize_2bit(grad.data_ptr<float>(), residual.data_ptr<float>(),
                      (uint32_t*)out.data_ptr<int32_t>(), n, (float)thr,
                      cur_stream());
  launch_check("quantize_2bit");
}

void dequantize_2bit(torch::Tensor packed, torch::Tensor out, double thr) {
  check_f32(out, "out");
  TORCH_CHECK(packed.scalar_type() == torch::kInt32 && packed.is_contiguous());
  const long long n = out.numel();
  TORCH_CHECK(packed.numel() >= (n + 15) / 16);
  geops_dequantize_2bit((const uint32_t*)packed.data_ptr<int32_t>(),
                        out.data_ptr<float>(), n, (float)thr, cur_stream());
  launch_check("dequantize_2bit");
}

void bsc_momentum(torch::Tensor g, torch::Tensor u, torch::Tensor v,
                  double mu) {
  check_f32(g, "g"); check_f32(u, "u"); check_f32(v, "v");
  const long long n = g.numel();
  TORCH_CHECK(u.numel() == n && v.numel() == n);
  geops_bsc_momentum(g.data_ptr<float>(), u.data_ptr<float>(),
                     v.data_ptr<float>(), (float)mu, n, cur_stream());
  launch_check("bsc_momentum");
}

torch::Tensor make_workspace(const torch::Tensor& like) {
  return torch::empty({2049}, torch::TensorOptions()
                                  .dtype(torch::kInt64)
                                  .device(like.device()));
}

void bsc_pack(torch::Tensor v, torch::Tensor u, torch::Tensor vals,
              torch::Tensor idx, double boundary, double placeholder) {
  check_f32(v, "v"); check_f32(u, "u"); check_f32(vals, "vals");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous());
  TORCH_CHECK(vals.numel() == idx.numel());
  auto ws = make_workspace(v);
  geops_bsc_pack(v.data_ptr<float>(), v.data_ptr<float>(),
                 u.data_ptr<float>(), vals.data_ptr<float>(),
                 idx.data_ptr<int32_t>(), (long long*)ws.data_ptr<int64_t>(),
                 (float)boundary, v.numel(), vals.numel(),
                 (float)placeholder, /*zero_uv=*/true, cur_stream());
  launch_check("bsc_pack");
}

bool bsc_compress_fused(torch::Tensor g, torch::Tensor u, torch::Tensor v,
                        torch::Tensor vals, torch::Tensor idx,
                        torch::Tensor boundary, double mu,
                        double placeholder) {
  check_f32(g, "g"); check_f32(u, "u"); check_f32(v, "v");
  check_f32(vals, "vals"); check_f32(boundary, "boundary");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous());
  TORCH_CHECK(vals.numel() == idx.numel());
  const long long n = g.numel();
  TORCH_CHECK(u.numel() == n && v.numel() == n && boundary.numel() == 1);
  auto ws = make_workspace(g);
  const int rc = geops_bsc_fused(
      g.data_ptr<float>(), u.data_ptr<float>(), v.data_ptr<float>(),
      vals.data_ptr<float>(), idx.data_ptr<int32_t>(),
      boundary.data_ptr<float>(), (unsigned long long*)ws.data_ptr<int64_t>(),
      (float)mu, n, vals.numel(), (float)placeholder, cur_stream());
  launch_check("bsc_compress_fused");
  return rc == 0;
}

void bsc_pull_pack(torch::Tensor x, torch::Tensor vals, torch::Tensor idx,
                   double placeholder) {
  check_f32(x, "x"); check_f32(vals, "vals");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous());
  auto ws = make_workspace(x);
  geops_bsc_pull_pack(x.data_ptr<float>(), vals.data_ptr<float>(),
                      idx.data_ptr<int32_t>(), (long long*)ws.data_ptr<int64_t>(),
                      x.numel(), vals.numel(), (float)placeholder,
                      cur_stream());
  launch_check("bsc_pull_pack");
}

void bsc_unpack(torch::Tensor vals, torch::Tensor idx, torch::Tensor out,
                bool accumulate) {
  check_f32(vals, "vals"); check_f32(out, "out");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous());
  if (!accumulate) {
    hipMemsetAsync(out.data_ptr<float>(), 0, out.numel() * sizeof(float),
                   cur_stream());
  }
  geops_bsc_unpack(vals.data_ptr<float>(), idx.data_ptr<int32_t>(),
                   out.data_ptr<float>(), vals.numel(), accumulate,
                   cur_stream());
  launch_check("bsc_unpack");
}

void dgt_contribution(torch::Tensor g, torch::Tensor out, int64_t chunk) {
  check_f32(g, "g"); check_f32(out, "out");
  const long long n = g.numel();
  const int nchunks = (int)((n + chunk - 1) / chunk);
  TORCH_CHECK(out.numel() == nchunks);
  geops_dgt_contribution(g.data_ptr<float>(), out.data_ptr<float>(), n,
                         (int)chunk, nchunks, cur_stream());
  launch_check("dgt_contribution");
}

void quantize_4bit(torch::Tensor x, torch::Tensor residual,
                   torch::Tensor packed, torch::Tensor minmax,
                   int64_t chunk) {
  check_f32(x, "x");
  TORCH_CHECK(packed.scalar_type() == torch::kUInt8 && packed.is_contiguous());
  check_f32(minmax, "minmax");
  const long long n = x.numel();
  const int nchunks = (int)((n + chunk - 1) / chunk);
  TORCH_CHECK(packed.numel() == (n + 1) / 2);
  TORCH_CHECK(minmax.numel() == 2 * nchunks);
  TORCH_CHECK((chunk & (chunk - 1)) == 0 && chunk >= 4,
              "GPU quantize_4bit needs power-of-two chunk >= 4");
  const bool has_res = residual.defined() && residual.numel() > 0;
  if (has_res) check_f32(residual, "residual");
  geops_quantize_4bit(x.data_ptr<float>(),
                      has_res ? residual.data_ptr<float>() : nullptr,
                      packed.data_ptr<uint8_t>(), minmax.data_ptr<float>(),
                      n, (int)chunk, nchunks, has_res, cur_stream());
  launch_check("quantize_4bit");
}

void dequantize_4bit(torch::Tensor packed, torch::Tensor minmax,
                     torch::Tensor out, int64_t chunk) {
  check_f32(out, "out"); check_f32(minmax, "minmax");
  TORCH_CHECK(packed.scalar_type() == torch::kUInt8 && packed.is_contiguous());
  TORCH_CHECK((chunk & (chunk - 1)) == 0 && chunk >= 4,
              "GPU dequantize_4bit needs power-of-two chunk >= 4");
  geops_dequantize_4bit(packed.data_ptr<uint8_t>(), minmax.data_ptr<float>(),
                        out.data_ptr<float>(), out.numel(), (int)chunk,
                        cur_stream());
  launch_check("dequantize_4bit");
}

void relu_maxpool2_fwd(torch::Tensor in, torch::Tensor out, torch::Tensor idx,
                       int64_t N, int64_t C, int64_t Hi, int64_t Wi) {
  TORCH_CHECK(in.is_cuda() && out.is_cuda() && idx.is_cuda());
  TORCH_CHECK(in.scalar_type() == torch::kBFloat16 &&
              out.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(idx.scalar_type() == torch::kUInt8);
  TORCH_CHECK(C % 8 == 0 && Hi % 2 == 0 && Wi % 2 == 0,
              "relu_maxpool2 needs C%8==0 and even H,W");
  const int Ho = (int)(Hi / 2), Wo = (int)(Wi / 2);
  const long long n_vec = (long long)N * Ho * Wo * (C / 8);
  geops_relu_maxpool2_fwd((const unsigned short*)in.data_ptr(),
                          (unsigned short*)out.data_ptr(),
                          idx.data_ptr<uint8_t>(), n_vec, Ho, Wo, (int)C,
                          cur_stream());
  launch_check("relu_maxpool2_fwd");
}

void relu_maxpool2_bwd(torch::Tensor grad_out, torch::Tensor idx,
                       torch::Tensor grad_in, int64_t N, int64_t C,
                       int64_t Hi, int64_t Wi) {
  TORCH_CHECK(grad_out.is_cuda() && idx.is_cuda() && grad_in.is_cuda());
  TORCH_CHECK(grad_out.scalar_type() == torch::kBFloat16 &&
              grad_in.scalar_type() == torch::kBFloat16);
  const long long n_vec_in = (long long)N * Hi * Wi * (C / 8);
  geops_relu_maxpool2_bwd((const unsigned short*)grad_out.data_ptr(),
                          idx.data_ptr<uint8_t>(),
                          (unsigned short*)grad_in.data_ptr(), n_vec_in,
                          (int)Hi, (int)Wi, (int)C, cur_stream());
  launch_check("relu_maxpool2_bwd");
}

void conv5_nhwc(torch::Tensor in, torch::Tensor w_frags, torch::Tensor bias,
                torch::Tensor out, int64_t N, int64_t Hi, int64_t Wi,
                int64_t Ho, int64_t Wo, int64_t CI, int64_t CO,
                int64_t pad) {
  TORCH_CHECK(in.is_cuda() && w_frags.is_cuda() && out.is_cuda());
  TORCH_CHECK(in.scalar_type() == torch::kBFloat16 &&
              w_frags.scalar_type() == torch::kBFloat16 &&
              out.scalar_type() == torch::kBFloat16);
  const bool has_bias = bias.defined() && bias.numel() > 0;
  if (has_bias)
    TORCH_CHECK(bias.scalar_type() == torch::kFloat32 && bias.is_cuda());
  const int rc = geops_conv5_nhwc(
      (const unsigned short*)in.data_ptr(),
      (const unsigned short*)w_frags.data_ptr(),
      has_bias ? bias.data_ptr<float>() : nullptr,
      (unsigned short*)out.data_ptr(), (int)N, (int)Hi, (int)Wi, (int)Ho,
      (int)Wo, (int)CI, (int)CO, (int)pad, cur_stream());
  TORCH_CHECK(rc == 0, "conv5_nhwc: unsupported geometry CI=", CI,
              " CO=", CO, " pad=", pad);
  launch_check("conv5_nhwc");
}

void conv5_pool_nhwc(torch::Tensor in, torch::Tensor w_frags,
                     torch::Tensor bias, torch::Tensor out,
                     torch::Tensor mask, int64_t N, int64_t Hi, int64_t Wi,
                     int64_t Ho, int64_t Wo, int64_t CI, int64_t CO) {
  TORCH_CHECK(in.is_cuda() && w_frags.is_cuda() && out.is_cuda() &&
              mask.is_cuda());
  TORCH_CHECK(in.scalar_type() == torch::kBFloat16 &&
              out.scalar_type() == torch::kBFloat16 &&
              mask.scalar_type() == torch::kUInt8);
  const bool has_bias = bias.numel() > 0;
  const int rc = geops_conv5_pool_nhwc(
      (const unsigned short*)in.data_ptr(),
      (const unsigned short*)w_frags.data_ptr(),
      has_bias ? bias.data_ptr<float>() : nullptr,
      (unsigned short*)out.data_ptr(), mask.data_ptr<uint8_t>(), (int)N,
      (int)Hi, (int)Wi, (int)Ho, (int)Wo, (int)CI, (int)CO, cur_stream());
  TORCH_CHECK(rc == 0, "conv5_pool_nhwc: unsupported geometry CI=", CI,
              " CO=", CO);
  launch_check("conv5_pool_nhwc");
}

void conv5_wrw_nhwc(torch::Tensor in, torch::Tensor gout, torch::Tensor part,
                    int64_t N, int64_t Hi, int64_t Wi, int64_t Ho,
                    int64_t Wo, int64_t CI, int64_t CO, int64_t n_wg) {
  TORCH_CHECK(in.is_cuda() && gout.is_cuda() && part.is_cuda());
  TORCH_CHECK(in.scalar_type() == torch::kBFloat16 &&
              gout.scalar_type() == torch::kBFloat16 &&
              part.scalar_type() == torch::kFloat32);
  // CI=16 shapes go to the v2 kernel (LDS-DMA staging + tr16 reads);
  // GEOPS_WRW_V1=1 forces the v1 path for A/B runs
  static const bool force_v1 = [] {
    const char* e = getenv("GEOPS_WRW_V1");
    return e && e[0] == '1';
  }();
  int rc = -1;
  if (CI == 4 && CO == 16 && !force_v1)
    rc = geops_conv5_wrw4_nhwc(
        (const unsigned short*)in.data_ptr(),
        (const unsigned short*)gout.data_ptr(), part.data_ptr<float>(),
        (int)N, (int)Hi, (int)Wi, (int)Ho, (int)Wo, (int)CI, (int)CO,
        (int)n_wg, cur_stream());
  if (CI == 16 && !force_v1)
    rc = geops_conv5_wrw16_nhwc(
        (const unsigned short*)in.data_ptr(),
        (const unsigned short*)gout.data_ptr(), part.data_ptr<float>(),
        (int)N, (int)Hi, (int)Wi, (int)Ho, (int)Wo, (int)CI, (int)CO,
        (int)n_wg, cur_stream());
  if (rc < 0)
    rc = geops_conv5_wrw_nhwc(
        (const unsigned short*)in.data_ptr(),
        (const unsigned short*)gout.data_ptr(), part.data_ptr<float>(),
        (int)N, (int)Hi, (int)Wi, (int)Ho, (int)Wo, (int)CI, (int)CO,
        (int)n_wg, cur_stream());
  TORCH_CHECK(rc > 0, "conv5_wrw_nhwc: unsupported geometry CI=", CI,
              " CO=", CO, " Wo=", Wo);
  launch_check("conv5_wrw_nhwc");
}

void pad_ch3to4_nhwc(torch::Tensor in, torch::Tensor out, int64_t npix) {
  TORCH_CHECK(in.is_cuda() && out.is_cuda());
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16);
  const bool fp32 = in.scalar_type() == torch::kFloat32;
  TORCH_CHECK(fp32 || in.scalar_type() == torch::kBFloat16);
  geops_pad_ch3to4_nhwc(in.data_ptr(), (unsigned short*)out.data_ptr(),
                        npix, fp32 ? 1 : 0, cur_stream());
  launch_check("pad_ch3to4_nhwc");
}

void sgd_update(torch::Tensor w, torch::Tensor g, double lr, double wd,
                double rescale) {
  check_f32(w, "w"); check_f32(g, "g");
  TORCH_CHECK(w.numel() == g.numel());
  geops_sgd_update(w.data_ptr<float>(), g.data_ptr<float>(), (float)lr,
                   (float)wd, (float)rescale, w.numel(), cur_stream());
  launch_check("sgd_update");
}

void sgd_mom_update(torch::Tensor w, torch::Tensor g, torch::Tensor mom,
                    double lr, double momentum, double wd, double rescale) {
  check_f32(w, "w"); check_f32(g, "g"); check_f32(mom, "mom");
  geops_sgd_mom_update(w.data_ptr<float>(), g.data_ptr<float>(),
                       mom.data_ptr<float>(), (float)lr, (float)momentum,
                       (float)wd, (float)rescale, w.numel(), cur_stream());
  launch_check("sgd_mom_update");
}

void adam_update(torch::Tensor w, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, int64_t t, double lr, double beta1,
                 double beta2, double eps, double wd, double rescale) {
  check_f32(w, "w"); check_f32(g, "g"); check_f32(m, "m"); check_f32(v, "v");
  const double lr_t =
      lr * std::sqrt(1.0 - std::pow(beta2, (double)t)) /
      (1.0 - std::pow(beta1, (double)t));
  geops_adam_update(w.data_ptr<float>(), g.data_ptr<float>(),
                    m.data_ptr<float>(), v.data_ptr<float>(), (float)lr_t,
                    (float)beta1, (float)beta2, (float)eps, (float)wd,
                    (float)rescale, w.numel(), cur_stream());
  launch_check("adam_update");
}

void dcasgd_update(torch::Tensor w, torch::Tensor g, torch::Tensor prev_w,
                   torch::Tensor mom, double lr, double lamda,
                   double momentum, double wd, double rescale) {
  check_f32(w, "w"); check_f32(g, "g"); check_f32(prev_w, "prev_w");
  const bool has_mom = mom.defined() && mom.numel() > 0;
  if (has_mom) check_f32(mom, "mom");
  geops_dcasgd_update(w.data_ptr<float>(), g.data_ptr<float>(),
                      prev_w.data_ptr<float>(),
                      has_mom ? mom.data_ptr<float>() : nullptr, (float)lr,
                      (float)lamda, (float)momentum, (float)wd,
                      (float)rescale, w.numel(), has_mom, cur_stream());
  launch_check("dcasgd_update");
}

void rmsprop_update(torch::Tensor w, torch::Tensor g, torch::Tensor n,
                    double lr, double rho, double eps, double wd,
                    double rescale) {
  check_f32(w, "w"); check_f32(g, "g"); check_f32(n, "n");
  geops_rmsprop_update(w.data_ptr<float>(), g.data_ptr<float>(),
                       n.data_ptr<float>(), (float)lr, (float)rho,
                       (float)eps, (float)wd, (float)rescale, w.numel(),
                       cur_stream());
  launch_check("rmsprop_update");
}

void adagrad_update(torch::Tensor w, torch::Tensor g, torch::Tensor h,
                    double lr, double eps, double wd, double rescale) {
  check_f32(w, "w"); check_f32(g, "g"); check_f32(h, "h");
  geops_adagrad_update(w.data_ptr<float>(), g.data_ptr<float>(),
                       h.data_ptr<float>(), (float)lr, (float)eps,
                       (float)wd, (float)rescale, w.numel(), cur_stream());
  launch_check("adagrad_update");
}

void signsgd_update(torch::Tensor w, torch::Tensor g, double lr, double wd,
                    double rescale) {
  check_f32(w, "w"); check_f32(g, "g");
  geops_signsgd_update(w.data_ptr<float>(), g.data_ptr<float>(), (float)lr,
                       (float)wd, (float)rescale, w.numel(), cur_stream());
  launch_check("signsgd_update");
}

void signum_update(torch::Tensor w, torch::Tensor g, torch::Tensor mom,
                   double lr, double momentum, double wd, double rescale) {
  check_f32(w, "w"); check_f32(g, "g"); check_f32(mom, "mom");
  geops_signum_update(w.data_ptr<float>(), g.data_ptr<float>(),
                      mom.data_ptr<float>(), (float)lr, (float)momentum,
                      (float)wd, (float)rescale, w.numel(), cur_stream());
  launch_check("signum_update");
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "geomx_amd gfx950 kernel library";
  m.def("quantize_2bit", &quantize_2bit);
  m.def("dequantize_2bit", &dequantize_2bit);
  m.def("bsc_momentum", &bsc_momentum);
  m.def("bsc_pack", &bsc_pack);
  m.def("bsc_pull_pack", &bsc_pull_pack);
  m.def("bsc_compress_fused", &bsc_compress_fused);
  m.def("bsc_unpack", &bsc_unpack);
  m.def("dgt_contribution", &dgt_contribution);
  m.def("quantize_4bit", &quantize_4bit);
  m.def("dequantize_4bit", &dequantize_4bit);
  m.def("conv5_nhwc", &conv5_nhwc);
  m.def("pad_ch3to4_nhwc", &pad_ch3to4_nhwc);
  m.def("conv5_wrw_nhwc", &conv5_wrw_nhwc);
  m.def("conv5_pool_nhwc", &conv5_pool_nhwc);
  m.def("wrw2_dump", [](torch::Tensor in, torch::Tensor gout,
                        torch::Tensor dump, int64_t N, int64_t Hi,
                        int64_t Wi, int64_t Ho, int64_t Wo, int64_t CO) {
    const int sz = geops_wrw2_dump(
        (const unsigned short*)in.data_ptr(),
        (const unsigned short*)gout.data_ptr(),
        (unsigned short*)dump.data_ptr(), (int)N, (int)Hi, (int)Wi,
        (int)Ho, (int)Wo, (int)CO, cur_stream());
    launch_check("wrw2_dump");
    return (int64_t)sz;
  });
  m.def("tr16_probe2", [](torch::Tensor in, torch::Tensor addr,
                          torch::Tensor out) {
    TORCH_CHECK(in.is_cuda() && addr.is_cuda() && out.is_cuda());
    TORCH_CHECK(in.numel() == 512 && addr.numel() == 64 &&
                out.numel() == 256);
    TORCH_CHECK(addr.scalar_type() == torch::kInt32);
    geops_tr16_probe2((const unsigned short*)in.data_ptr(),
                      addr.data_ptr<int32_t>(),
                      (unsigned short*)out.data_ptr(), cur_stream());
    launch_check("tr16_probe2");
  });
  m.def("tr16_probe", [](torch::Tensor in, torch::Tensor out) {
    TORCH_CHECK(in.is_cuda() && out.is_cuda());
    TORCH_CHECK(in.numel() == 64 && out.numel() == 256);
    geops_tr16_probe((const unsigned short*)in.data_ptr(),
                     (unsigned short*)out.data_ptr(), cur_stream());
    launch_check("tr16_probe");
  });
  m.def("relu_maxpool2_fwd", &relu_maxpool2_fwd);
  m.def("relu_maxpool2_bwd", &relu_maxpool2_bwd);
  m.def("sgd_update", &sgd_update);
  m.def("sgd_mom_update", &sgd_mom_update);
  m.def("adam_update", &adam_update);
  m.def("dcasgd_update", &dcasgd_update);
  m.def("rmsprop_update", &rmsprop_update);
  m.def("adagrad_update", &adagrad_update);
  m.def("signsgd_update", &signsgd_update);
  m.def("signum_update", &signum_update);
}
