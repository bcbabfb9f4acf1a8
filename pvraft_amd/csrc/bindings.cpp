// Python bindings for the pvraft_amd CDNA4 kernels (torch extension).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

void launch_knn_graph(const float*, int*, int, int, int, hipStream_t);
void launch_gather_edge_fwd(const float*, const int*, const float*, float*,
                            int, int, int, int, hipStream_t);
void launch_gather_edge_bwd(const float*, const int*, float*, int, int, int,
                            int, hipStream_t);
void launch_voxel_corr_fwd(const float*, const float*, const float*, float*,
                           int, int, int, int, float, hipStream_t);
void launch_voxel_corr_bwd(const float*, const float*, const float*, float*,
                           int, int, int, int, float, hipStream_t);
void launch_knn_corr_fwd(const float*, const float*, const float*, float*,
                         int*, int, int, int, int, hipStream_t);
void launch_knn_corr_bwd(const float*, const int*, float*, int, int, int, int,
                         hipStream_t);

namespace {

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be float32");
}

hipStream_t stream() { return at::hip::getCurrentHIPStream().stream(); }

}  // namespace

torch::Tensor knn_graph(torch::Tensor xyz, int64_t k) {
  check_f32(xyz, "xyz");
  TORCH_CHECK(xyz.dim() == 3 && xyz.size(2) == 3, "xyz must be (B,N,3)");
  const int B = xyz.size(0), N = xyz.size(1);
  TORCH_CHECK(k >= 1 && k <= 48 && k <= N, "knn_graph requires 1 <= k <= 48, k <= N");
  auto out = torch::empty({B, N, k}, xyz.options().dtype(torch::kInt32));
  launch_knn_graph(xyz.data_ptr<float>(), out.data_ptr<int>(), B, N, (int)k, stream());
  return out;
}

torch::Tensor gather_edge_concat_fwd(torch::Tensor feats, torch::Tensor idx,
                                     torch::Tensor xyz) {
  check_f32(feats, "feats");
  check_f32(xyz, "xyz");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous(), "idx must be contiguous int32");
  const int B = feats.size(0), N = feats.size(1), C = feats.size(2), K = idx.size(2);
  auto out = torch::empty({B, C + 3, K, N}, feats.options());
  launch_gather_edge_fwd(feats.data_ptr<float>(), idx.data_ptr<int>(),
                         xyz.data_ptr<float>(), out.data_ptr<float>(), B, N, K,
                         C, stream());
  return out;
}

torch::Tensor gather_edge_concat_bwd(torch::Tensor gout, torch::Tensor idx,
                                     int64_t C) {
  check_f32(gout, "gout");
  const int B = gout.size(0), K = gout.size(2), N = gout.size(3);
  TORCH_CHECK(gout.size(1) == C + 3, "gout channel mismatch");
  auto gfeats = torch::empty({B, N, C}, gout.options());
  launch_gather_edge_bwd(gout.data_ptr<float>(), idx.data_ptr<int>(),
                         gfeats.data_ptr<float>(), B, N, K, (int)C, stream());
  return gfeats;
}

torch::Tensor voxel_corr_fwd(torch::Tensor corr, torch::Tensor xyz,
                             torch::Tensor coords, double base_scale,
                             int64_t num_levels, int64_t resolution) {
  check_f32(corr, "corr");
  check_f32(xyz, "xyz");
  check_f32(coords, "coords");
  TORCH_CHECK(resolution == 3, "HIP voxel_corr supports resolution=3");
  TORCH_CHECK(num_levels >= 1 && num_levels <= 4, "1 <= num_levels <= 4");
  const int B = corr.size(0), N = corr.size(1), K = corr.size(2);
  auto out = torch::empty({B, num_levels * 27, N}, corr.options());
  launch_voxel_corr_fwd(corr.data_ptr<float>(), xyz.data_ptr<float>(),
                        coords.data_ptr<float>(), out.data_ptr<float>(), B, N,
                        K, (int)num_levels, (float)base_scale, stream());
  return out;
}

torch::Tensor voxel_corr_bwd(torch::Tensor gout, torch::Tensor xyz,
                             torch::Tensor coords, double base_scale,
                             int64_t num_levels, int64_t resolution) {
  check_f32(gout, "gout");
  TORCH_CHECK(resolution == 3, "HIP voxel_corr supports resolution=3");
  const int B = xyz.size(0), N = xyz.size(1), K = xyz.size(2);
  auto gcorr = torch::empty({B, N, K}, gout.options());
  launch_voxel_corr_bwd(gout.data_ptr<float>(), xyz.data_ptr<float>(),
                        coords.data_ptr<float>(), gcorr.data_ptr<float>(), B,
                        N, K, (int)num_levels, (float)base_scale, stream());
  return gcorr;
}

std::vector<torch::Tensor> knn_corr_fwd(torch::Tensor corr, torch::Tensor xyz,
                                        torch::Tensor coords, int64_t k) {
  check_f32(corr, "corr");
  check_f32(xyz, "xyz");
  check_f32(coords, "coords");
  const int B = corr.size(0), N = corr.size(1), K = corr.size(2);
  TORCH_CHECK(K <= 512, "HIP knn_corr supports K <= 512");
  TORCH_CHECK(k >= 1 && k <= K, "need 1 <= k <= K");
  auto out = torch::empty({B, 4, N, k}, corr.options());
  auto idx = torch::empty({B, N, k}, corr.options().dtype(torch::kInt32));
  launch_knn_corr_fwd(corr.data_ptr<float>(), xyz.data_ptr<float>(),
                      coords.data_ptr<float>(), out.data_ptr<float>(),
                      idx.data_ptr<int>(), B, N, K, (int)k, stream());
  return {out, idx};
}

torch::Tensor knn_corr_bwd(torch::Tensor gout, torch::Tensor idx, int64_t K) {
  check_f32(gout, "gout");
  const int B = gout.size(0), N = gout.size(2), k = gout.size(3);
  auto gcorr = torch::zeros({B, N, K}, gout.options());
  launch_knn_corr_bwd(gout.data_ptr<float>(), idx.data_ptr<int>(),
                      gcorr.data_ptr<float>(), B, N, (int)K, k, stream());
  return gcorr;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("knn_graph", &knn_graph, "fused kNN graph (CDNA4)");
  m.def("gather_edge_concat_fwd", &gather_edge_concat_fwd);
  m.def("gather_edge_concat_bwd", &gather_edge_concat_bwd);
  m.def("voxel_corr_fwd", &voxel_corr_fwd);
  m.def("voxel_corr_bwd", &voxel_corr_bwd);
  m.def("knn_corr_fwd", &knn_corr_fwd);
  m.def("knn_corr_bwd", &knn_corr_bwd);
}
