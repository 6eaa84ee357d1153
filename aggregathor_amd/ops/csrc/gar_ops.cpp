// Torch glue for the gfx950 GAR kernels (gar_kernels.hip).
//
// Native-HIP torch extension: compiled directly with hipcc against the
// PyTorch-ROCm C++ API (c10::hip stream accessors) -- no CUDA compatibility
// layer, no hipify. All launches go to the rank's current HIP stream, no
// host synchronization anywhere: a full Krum/Bulyan aggregation is 3-4
// dependent kernels and is hipGraph-capturable.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include "gar_kernels.h"

namespace {

void check_input(const torch::Tensor& g) {
  TORCH_CHECK(g.is_cuda(), "GAR ops: expected a GPU tensor");
  TORCH_CHECK(g.scalar_type() == torch::kFloat32,
              "GAR ops: expected fp32 gradients, got ", g.scalar_type());
  TORCH_CHECK(g.dim() == 2, "GAR ops: expected stacked [n, d] gradients");
  TORCH_CHECK(g.is_contiguous(), "GAR ops: gradients must be contiguous");
  TORCH_CHECK(g.size(0) >= 2 && g.size(0) <= gar::kMaxNSelect,
              "GAR ops: need 2 <= n <= ", gar::kMaxNSelect, ", got n = ",
              g.size(0));
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

torch::Tensor dist_matrix(const torch::Tensor& g) {
  const int n = (int)g.size(0);
  const long d = (long)g.size(1);
  auto opts = g.options();
  auto partials = torch::empty({gar::sqdist_partials_elems(n, d)}, opts);
  auto dist = torch::empty({n, n}, opts);
  gar::sqdist(g.data_ptr<float>(), n, d, partials.data_ptr<float>(),
              dist.data_ptr<float>(), current_stream());
  return dist;
}

}  // namespace

torch::Tensor pairwise_sqdist(torch::Tensor g) {
  check_input(g);
  return dist_matrix(g);
}

torch::Tensor krum(torch::Tensor g, long f, long m) {
  check_input(g);
  const int n = (int)g.size(0);
  const long d = (long)g.size(1);
  TORCH_CHECK(n - f - 2 >= 1, "krum requires n - f - 2 >= 1");
  TORCH_CHECK(m >= 1 && m <= n, "krum selection count m out of range");
  auto dist = dist_matrix(g);
  auto sel = torch::empty({m}, g.options().dtype(torch::kInt32));
  gar::krum_select(dist.data_ptr<float>(), n, (int)f, (int)m,
                   sel.data_ptr<int>(), current_stream());
  auto out = torch::empty({d}, g.options());
  gar::selection_average(g.data_ptr<float>(), n, d, sel.data_ptr<int>(),
                         (int)m, out.data_ptr<float>(), current_stream());
  return out;
}

torch::Tensor bulyan(torch::Tensor g, long f, long m) {
  check_input(g);
  const int n = (int)g.size(0);
  const long d = (long)g.size(1);
  const long t = n - 2 * f - 2;
  const long b = t - 2 * f;
  TORCH_CHECK(t >= 1 && b >= 1, "bulyan requires n >= 4f + 3 (n=", n,
              ", f=", f, ")");
  TORCH_CHECK(n <= gar::kMaxNCoord,
              "bulyan GPU kernel supports n <= ", gar::kMaxNCoord);
  auto dist = dist_matrix(g);
  auto flags = torch::empty({gar::bulyan_flags_bytes(n, (int)f)},
                            g.options().dtype(torch::kUInt8));
  gar::bulyan_select(dist.data_ptr<float>(), n, (int)f, (int)m,
                     flags.data_ptr<unsigned char>(), current_stream());
  auto out = torch::empty({d}, g.options());
  gar::bulyan_final(g.data_ptr<float>(), n, d, (int)f, (int)m,
                    flags.data_ptr<unsigned char>(), out.data_ptr<float>(),
                    current_stream());
  return out;
}

torch::Tensor median(torch::Tensor g) {
  check_input(g);
  TORCH_CHECK(g.size(0) <= gar::kMaxNCoord,
              "median GPU kernel supports n <= ", gar::kMaxNCoord);
  auto out = torch::empty({g.size(1)}, g.options());
  gar::median(g.data_ptr<float>(), (int)g.size(0), (long)g.size(1),
              out.data_ptr<float>(), current_stream());
  return out;
}

torch::Tensor averaged_median(torch::Tensor g, long beta) {
  check_input(g);
  TORCH_CHECK(g.size(0) <= gar::kMaxNCoord,
              "averaged_median GPU kernel supports n <= ", gar::kMaxNCoord);
  TORCH_CHECK(beta >= 1 && beta <= g.size(0), "beta out of range");
  auto out = torch::empty({g.size(1)}, g.options());
  gar::averaged_median(g.data_ptr<float>(), (int)g.size(0), (long)g.size(1),
                       (int)beta, out.data_ptr<float>(), current_stream());
  return out;
}

torch::Tensor average_nan(torch::Tensor g) {
  check_input(g);
  TORCH_CHECK(g.size(0) <= gar::kMaxNCoord,
              "average_nan GPU kernel supports n <= ", gar::kMaxNCoord);
  auto out = torch::empty({g.size(1)}, g.options());
  gar::average_nan(g.data_ptr<float>(), (int)g.size(0), (long)g.size(1),
                   out.data_ptr<float>(), current_stream());
  return out;
}

torch::Tensor selection_average(torch::Tensor g, torch::Tensor sel) {
  check_input(g);
  TORCH_CHECK(sel.is_cuda() && sel.scalar_type() == torch::kInt32 &&
                  sel.is_contiguous(),
              "sel must be a contiguous int32 GPU tensor");
  auto out = torch::empty({g.size(1)}, g.options());
  gar::selection_average(g.data_ptr<float>(), (int)g.size(0),
                         (long)g.size(1), sel.data_ptr<int>(),
                         (int)sel.size(0), out.data_ptr<float>(),
                         current_stream());
  return out;
}

// ---------------------------------------------------------------------------
// Fused NHWC BatchNorm (training). x: [N, C, H, W] channels_last (bf16 or
// fp32); weight/bias/running stats fp32. Returns (y, save_mean,
// save_invstd).

static void check_bn_input(const torch::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn: expected 4D GPU tensor");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn: expected channels_last input");
  TORCH_CHECK(x.size(1) % 4 == 0, "bn: C must be a multiple of 4");
  TORCH_CHECK(x.scalar_type() == torch::kFloat32 ||
                  x.scalar_type() == torch::kBFloat16,
              "bn: fp32 or bf16 input");
}

std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x, torch::Tensor weight,
                                        torch::Tensor bias,
                                        torch::Tensor running_mean,
                                        torch::Tensor running_var,
                                        double momentum, double eps) {
  check_bn_input(x);
  const long m = x.size(0) * x.size(2) * x.size(3);
  const int c = (int)x.size(1);
  const int dtype = x.scalar_type() == torch::kBFloat16 ? 1 : 0;
  auto f32 = x.options().dtype(torch::kFloat32);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({c}, f32);
  auto invstd = torch::empty({c}, f32);
  auto partials = torch::empty({gar::bn_partials_elems(c)}, f32);
  gar::bn_fwd(x.data_ptr(), y.data_ptr(), m, c, dtype, (float)eps,
              (float)momentum, weight.data_ptr<float>(),
              bias.data_ptr<float>(), running_mean.data_ptr<float>(),
              running_var.data_ptr<float>(), mean.data_ptr<float>(),
              invstd.data_ptr<float>(), partials.data_ptr<float>(),
              current_stream());
  return {y, mean, invstd};
}

std::vector<torch::Tensor> bn_bwd_train(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor weight,
                                        torch::Tensor mean,
                                        torch::Tensor invstd) {
  check_bn_input(x);
  TORCH_CHECK(dy.sizes() == x.sizes() && dy.scalar_type() == x.scalar_type(),
              "bn_bwd: dy/x mismatch");
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const long m = x.size(0) * x.size(2) * x.size(3);
  const int c = (int)x.size(1);
  const int dtype = x.scalar_type() == torch::kBFloat16 ? 1 : 0;
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dx = torch::empty_like(x);
  auto dweight = torch::empty({c}, f32);
  auto dbias = torch::empty({c}, f32);
  auto partials = torch::empty({gar::bn_partials_elems(c)}, f32);
  auto consts = torch::empty({3L * c}, f32);
  gar::bn_bwd(dyc.data_ptr(), x.data_ptr(), dx.data_ptr(), m, c, dtype,
              weight.data_ptr<float>(), mean.data_ptr<float>(),
              invstd.data_ptr<float>(), dweight.data_ptr<float>(),
              dbias.data_ptr<float>(), partials.data_ptr<float>(),
              consts.data_ptr<float>(), current_stream());
  return {dx, dweight, dbias};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("bn_fwd_train", &bn_fwd_train,
          "Fused NHWC BatchNorm training forward (gfx950)");
  mod.def("bn_bwd_train", &bn_bwd_train,
          "Fused NHWC BatchNorm training backward (gfx950)");
  mod.def("pairwise_sqdist", &pairwise_sqdist,
          "All-pairs squared L2 distances [n,n] (diag=+inf), gfx950 kernels");
  mod.def("krum", &krum, "Multi-Krum GAR (gfx950 kernels)");
  mod.def("bulyan", &bulyan, "Bulyan over Multi-Krum GAR (gfx950 kernels)");
  mod.def("median", &median, "Coordinate-wise median GAR (gfx950)");
  mod.def("averaged_median", &averaged_median,
          "Averaged-median GAR (gfx950)");
  mod.def("average_nan", &average_nan, "NaN-skipping mean GAR (gfx950)");
  mod.def("selection_average", &selection_average,
          "Mean of selected rows (gfx950)");
}
