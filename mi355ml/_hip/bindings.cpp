// Torch bindings for the mi355ml CDNA4 kernels (kernels.hip).
// Partial slabs are reduced here with deterministic torch ops (tree order),
// so results are bitwise reproducible run to run.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

extern "C" {
void launch_col_moments(const float*, long, int, double*, double*, double*,
                        int, hipStream_t);
void launch_col_minmax(const float*, long, int, float*, float*, int,
                       hipStream_t);
void launch_glm_value_grad(const float*, const float*, const float*, float,
                           long, int, int, double*, int, hipStream_t);
void launch_kmeans_assign(const float*, const float*, long, int, int, int*,
                          float*, float*, double*, int, hipStream_t);
void launch_closest_center(const float*, const float*, long, int, int, int*,
                           float*, int, hipStream_t);
void launch_affine_transform(const float*, const float*, const float*,
                             float*, long, int, int, hipStream_t);
void launch_kmeans_probe(const float*, const float*, long, int, int, int*,
                         float*, float*, double*, int, int, hipStream_t);
void launch_glm_probe(const float*, const float*, const float*, float, long,
                      int, int, double*, int, hipStream_t);
size_t kmeans_smem_bytes(int k, int d, int update);
void launch_kmeans_assign_split(const float*, const float*, long, int, int,
                                int*, float*, float*, double*, int,
                                hipStream_t);
size_t kmeans_split_smem_bytes(int k, int d);
}

namespace {

constexpr int kMaxBlocks = 1024;

void check_f32_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.dtype() == torch::kFloat32, name, " must be float32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

std::vector<torch::Tensor> col_moments_partial(torch::Tensor x) {
  check_f32_2d(x, "x");
  TORCH_CHECK(x.dim() == 2);
  const long m = x.size(0);
  const int f = (int)x.size(1);
  const int nb = (int)std::max(1L, std::min((long)kMaxBlocks,
                                            (m + 255) / 256));
  auto opts = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto ns = torch::zeros({nb}, opts);
  auto sums = torch::zeros({nb, f}, opts);
  auto m2s = torch::zeros({nb, f}, opts);
  launch_col_moments(x.data_ptr<float>(), m, f, ns.data_ptr<double>(),
                     sums.data_ptr<double>(), m2s.data_ptr<double>(), nb,
                     cur_stream());
  return {ns, sums, m2s};
}

std::vector<torch::Tensor> col_minmax_partial(torch::Tensor x) {
  check_f32_2d(x, "x");
  TORCH_CHECK(x.dim() == 2);
  const long m = x.size(0);
  const int f = (int)x.size(1);
  const int nb = (int)std::max(1L, std::min((long)kMaxBlocks,
                                            (m + 255) / 256));
  auto opts = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto mins = torch::empty({nb, f}, opts);
  auto maxs = torch::empty({nb, f}, opts);
  launch_col_minmax(x.data_ptr<float>(), m, f, mins.data_ptr<float>(),
                    maxs.data_ptr<float>(), nb, cur_stream());
  return {mins, maxs};
}

std::vector<torch::Tensor> glm_value_grad(torch::Tensor x, torch::Tensor y,
                                          torch::Tensor w, double b,
                                          long family) {
  check_f32_2d(x, "x");
  check_f32_2d(y, "y");
  check_f32_2d(w, "w");
  TORCH_CHECK(x.dim() == 2 && y.dim() == 1 && w.dim() == 1);
  const long m = x.size(0);
  const int d = (int)x.size(1);
  TORCH_CHECK(d <= 512, "glm kernel supports d <= 512 (got ", d, ")");
  TORCH_CHECK(y.size(0) == m && w.size(0) == d);
  const long tiles = (m + 63) / 64;
  const int nb = (int)std::max(1L, std::min((long)kMaxBlocks, tiles));
  auto opts = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  // one fused slab [nb, d+2] = (gw | loss | gb): ONE reduce launch
  auto slab = torch::zeros({nb, d + 2}, opts);
  launch_glm_value_grad(x.data_ptr<float>(), y.data_ptr<float>(),
                        w.data_ptr<float>(), (float)b, m, d, (int)family,
                        slab.data_ptr<double>(), nb, cur_stream());
  auto total = slab.sum(0);
  using torch::indexing::Slice;
  return {total.index({d}), total.index({Slice(0, d)}), total.index({d + 1})};
}

std::vector<torch::Tensor> kmeans_assign(torch::Tensor x,
                                         torch::Tensor centers) {
  check_f32_2d(x, "x");
  check_f32_2d(centers, "centers");
  TORCH_CHECK(x.dim() == 2 && centers.dim() == 2);
  const long m = x.size(0);
  const int d = (int)x.size(1);
  const int k = (int)centers.size(0);
  TORCH_CHECK(centers.size(1) == d);
  TORCH_CHECK(k <= 256, "fused kmeans kernel supports k <= 256 (got ", k,
              "); use the rocBLAS-composed path for more centers");
  TORCH_CHECK(kmeans_smem_bytes(k, d, 1) <= 160 * 1024 - 1024,
              "k*d too large for LDS (", kmeans_smem_bytes(k, d, 1),
              " bytes)");
  const int nb = (int)std::max(1L, std::min(256L, (m + 255) / 256));
  auto f32 = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto f64 = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto labels = torch::empty({m}, torch::TensorOptions()
                                      .dtype(torch::kInt32)
                                      .device(x.device()));
  auto sums_slab = torch::zeros({nb, k, d}, f32);
  auto cnt_slab = torch::zeros({nb, k}, f32);
  auto inertia_slab = torch::zeros({nb}, f64);
  launch_kmeans_assign(x.data_ptr<float>(), centers.data_ptr<float>(), m, d,
                       k, labels.data_ptr<int>(), sums_slab.data_ptr<float>(),
                       cnt_slab.data_ptr<float>(),
                       inertia_slab.data_ptr<double>(), nb, cur_stream());
  return {labels, sums_slab.sum(0).to(torch::kFloat64),
          cnt_slab.sum(0).to(torch::kFloat64), inertia_slab.sum()};
}


std::vector<torch::Tensor> kmeans_assign_split(torch::Tensor x,
                                               torch::Tensor centers) {
  check_f32_2d(x, "x");
  check_f32_2d(centers, "centers");
  TORCH_CHECK(x.dim() == 2 && centers.dim() == 2);
  const long m = x.size(0);
  const int d = (int)x.size(1);
  const int k = (int)centers.size(0);
  TORCH_CHECK(centers.size(1) == d);
  TORCH_CHECK(k <= 256, "fused kmeans kernel supports k <= 256 (got ", k,
              ")");
  TORCH_CHECK(kmeans_split_smem_bytes(k, d) <= 160 * 1024 - 1024,
              "k*d too large for the split kernel LDS (",
              kmeans_split_smem_bytes(k, d), " bytes)");
  const int nb = (int)std::max(1L, std::min(256L, (m + 255) / 256));
  auto f32 = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto f64 = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto labels = torch::empty({m}, torch::TensorOptions()
                                      .dtype(torch::kInt32)
                                      .device(x.device()));
  auto sums_slab = torch::zeros({nb, k, d}, f32);
  auto cnt_slab = torch::zeros({nb, k}, f32);
  auto inertia_slab = torch::zeros({nb}, f64);
  launch_kmeans_assign_split(x.data_ptr<float>(), centers.data_ptr<float>(),
                             m, d, k, labels.data_ptr<int>(),
                             sums_slab.data_ptr<float>(),
                             cnt_slab.data_ptr<float>(),
                             inertia_slab.data_ptr<double>(), nb,
                             cur_stream());
  return {labels, sums_slab.sum(0).to(torch::kFloat64),
          cnt_slab.sum(0).to(torch::kFloat64), inertia_slab.sum()};
}

bool kmeans_split_fits(long k, long d) {
  return k <= 256 &&
         kmeans_split_smem_bytes((int)k, (int)d) <= 160 * 1024 - 1024;
}

std::vector<torch::Tensor> closest_center(torch::Tensor x,
                                          torch::Tensor centers) {
  check_f32_2d(x, "x");
  check_f32_2d(centers, "centers");
  const long m = x.size(0);
  const int d = (int)x.size(1);
  const int k = (int)centers.size(0);
  TORCH_CHECK(centers.size(1) == d);
  TORCH_CHECK(k <= 256, "fused closest_center supports k <= 256");
  TORCH_CHECK(kmeans_smem_bytes(k, d, 0) <= 160 * 1024 - 1024,
              "k*d too large for LDS");
  const int nb = (int)std::max(1L, std::min(256L, (m + 255) / 256));
  auto labels = torch::empty({m}, torch::TensorOptions()
                                      .dtype(torch::kInt32)
                                      .device(x.device()));
  auto dists = torch::empty({m}, torch::TensorOptions()
                                     .dtype(torch::kFloat32)
                                     .device(x.device()));
  launch_closest_center(x.data_ptr<float>(), centers.data_ptr<float>(), m, d,
                        k, labels.data_ptr<int>(), dists.data_ptr<float>(),
                        nb, cur_stream());
  return {labels, dists};
}

torch::Tensor glm_phase_probe(torch::Tensor x, torch::Tensor y,
                              torch::Tensor w, long phases) {
  const long m = x.size(0);
  const int d = (int)x.size(1);
  const long tiles = (m + 63) / 64;
  const int nb = (int)std::max(1L, std::min((long)kMaxBlocks, tiles));
  auto opts = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto slab = torch::zeros({nb, d + 2}, opts);
  launch_glm_probe(x.data_ptr<float>(), y.data_ptr<float>(),
                   w.data_ptr<float>(), 0.1f, m, d, (int)phases,
                   slab.data_ptr<double>(), nb, cur_stream());
  return slab;
}

torch::Tensor kmeans_phase_probe(torch::Tensor x, torch::Tensor centers,
                                 long phases) {
  const long m = x.size(0);
  const int d = (int)x.size(1);
  const int k = (int)centers.size(0);
  const int nb = (int)std::max(1L, std::min(512L, (m + 255) / 256));
  auto f32 = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto f64 = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto labels = torch::zeros({m}, torch::TensorOptions()
                                      .dtype(torch::kInt32)
                                      .device(x.device()));
  auto sums_slab = torch::zeros({nb, k, d}, f32);
  auto cnt_slab = torch::zeros({nb, k}, f32);
  auto inertia_slab = torch::zeros({nb}, f64);
  launch_kmeans_probe(x.data_ptr<float>(), centers.data_ptr<float>(), m, d,
                      k, labels.data_ptr<int>(), sums_slab.data_ptr<float>(),
                      cnt_slab.data_ptr<float>(),
                      inertia_slab.data_ptr<double>(), nb, (int)phases,
                      cur_stream());
  return labels;
}

torch::Tensor affine_transform(torch::Tensor x, torch::Tensor a,
                               torch::Tensor b, torch::Tensor out) {
  check_f32_2d(x, "x");
  check_f32_2d(a, "a");
  check_f32_2d(b, "b");
  check_f32_2d(out, "out");
  const long n = x.numel();
  const int d = (int)x.size(-1);
  const int nb = (int)std::max(1L, std::min(2048L, (n / 4 + 255) / 256));
  launch_affine_transform(x.data_ptr<float>(), a.data_ptr<float>(),
                          b.data_ptr<float>(), out.data_ptr<float>(), n, d,
                          nb, cur_stream());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("col_moments_partial", &col_moments_partial,
          "per-WG column (n, sum, M2) partials");
  mod.def("col_minmax_partial", &col_minmax_partial,
          "per-WG column (min, max) partials");
  mod.def("glm_value_grad", &glm_value_grad,
          "fused GLM (loss, grad_w, grad_b) over one shard");
  mod.def("kmeans_assign", &kmeans_assign,
          "fused KMeans labels + centroid partial sums/counts/inertia");
  mod.def("kmeans_assign_split", &kmeans_assign_split,
          "wave-specialized assign+update (2 update waves, software "
          "pipelined over tiles)");
  mod.def("kmeans_split_fits", &kmeans_split_fits,
          "does (k, d) fit the split kernel LDS budget");
  mod.def("closest_center", &closest_center,
          "fused argmin-distance labels + squared distances");
  mod.def("affine_transform", &affine_transform,
          "fused per-column affine transform");
  mod.def("glm_phase_probe", &glm_phase_probe,
          "glm ablation (bit0 stage, bit1 z, bit2 grad)");
  mod.def("kmeans_phase_probe", &kmeans_phase_probe,
          "phase-mask ablation probe (bit0 stage, bit1 compute, bit2 update)");
}
