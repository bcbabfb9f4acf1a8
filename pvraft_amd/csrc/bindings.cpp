// Python bindings for the pvraft_amd CDNA4 kernels (torch extension).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

void launch_knn_graph(const float*, int*, int, int, int, hipStream_t);
void launch_morton_keys(const float*, const float*, const float*, long*,
                        int, long, hipStream_t);
void launch_gather_edge_fwd(const void*, const int*, const float*, void*,
                            int, int, int, int, bool, hipStream_t);
void launch_gather_edge_bwd(const void*, const int*, float*, int, int, int,
                            int, bool, hipStream_t);
void launch_gather_edge_bwd_csr(const void*, const int*, const int*, void*,
                                int, int, int, int, bool, hipStream_t);
void launch_voxel_corr_fwd(const float*, const float*, const float*, float*,
                           int, int, int, int, float, hipStream_t);
void launch_voxel_corr_bwd(const float*, const float*, const float*, float*,
                           int, int, int, int, float, hipStream_t);
void launch_knn_corr_fwd(const float*, const float*, const float*, float*,
                         int*, int, int, int, int, hipStream_t);
void launch_knn_corr_bwd(const float*, const int*, float*, int, int, int, int,
                         hipStream_t);

void launch_gn_fwd(const void*, void*, float*, float*, float*, const float*,
                   const float*, int, long, long, int, int, float, int, float,
                   const float*, bool, hipStream_t);
void launch_gn_bwd(const void*, const void*, const float*, const float*,
                   const float*, const float*, float*, float*, float*,
                   float*, int, void*, int, long, long, int, int, int, float,
                   const float*, bool, hipStream_t);
int gn_reduce_chunks(long, int, int, int);
int gnmp_reduce_chunks(long, int, int, int, int);
int gnmp_bwd_reduce_chunks(long, int, int, int);
void launch_gnmp_fwd(const void*, void*, unsigned char*, float*, float*,
                     float*, const float*, const float*, int, long, long, int,
                     int, int, float, int, float, const float*, bool,
                     hipStream_t);
void launch_pv_corr_fused_fwd(const float*, const float*, const float*,
                              float*, float*, int*, int, int, int, int, int,
                              float, hipStream_t);
void launch_pv_corr_fused_bwd(const float*, const float*, const float*,
                              const float*, const int*, float*, int, int, int,
                              int, int, float, hipStream_t);
void launch_topk_rows(const float*, float*, int*, long, int, int,
                      hipStream_t);
void launch_seq_loss_fwd(const float* const*, const float*, const float*,
                         float*, float*, float*, long, int, int, float,
                         hipStream_t);
void launch_seq_loss_bwd(const float* const*, float* const*, const float*,
                         const float*, const float*, const float*, long, int,
                         int, float, hipStream_t);
void launch_gru_zr_fwd(const void*, const void*, void*, void*, void*, long,
                       long, bool, hipStream_t);
void launch_gru_zr_bwd(const void*, const void*, const void*, const void*,
                       const void*, void*, void*, long, long, bool,
                       hipStream_t);
void launch_gru_q_fwd(const void*, const void*, const void*, void*, void*,
                      long, long, bool, hipStream_t);
void launch_gru_q_bwd(const void*, const void*, const void*, const void*,
                      void*, void*, void*, long, long, bool, hipStream_t);
void launch_transpose(const void*, void*, long, int, long, long, bool,
                      hipStream_t);
void launch_pw_wgrad(const void*, const void*, float*, float*, int, int, int,
                     long, int, hipStream_t);
void launch_gn_bwd_extract(float*, float*, float*, float*, int, int, int,
                           hipStream_t);
void launch_egnmp_fwd(const void*, const int*, float*, float*, float*,
                      float*, const float*, const float*, float*, float*,
                      unsigned char*, unsigned char*, float*, void*,
                      unsigned char*, float*,
                      int, long, int, int, int, float, int, float,
                      const float*, bool, int, hipStream_t);
void launch_egnmp_bwd(const void*, const void*, const int*,
                      const unsigned char*, const float*, const float*,
                      const int*, const int*,
                      const unsigned char*, const float*, const float*,
                      const float*, const float*, float*, float*, void*, int,
                      long, int, int, int, int, float, const float*, bool,
                      int, hipStream_t);
int egnmp_reduce_chunks(long, int, int);
struct CastDesc { const float* src; void* dst; int n; };
struct CastChunk { CastDesc d[128]; int count; };
void launch_multi_cast(const CastChunk*, int, hipStream_t);
void launch_kg_fwd(const float*, const float*, const float*, float*, float*,
                   float*, float*, const float*, const float*, float*,
                   float*, unsigned char*, unsigned char*, void*,
                   unsigned char*, float*, int, long, int, int, int, float,
                   const float*, bool, int, hipStream_t);
void launch_kg_bwd(const void*, const float*, const float*, const float*,
                   const unsigned char*, const float*, const float*,
                   const float*, const float*, const float*, float*, float*,
                   float*, float*, float*, float*, int, long, int, int, int,
                   const float*, bool, int, hipStream_t);
void launch_gnmp_bwd(const void*, const void*, const unsigned char*,
                     const float*, const float*, const float*, const float*,
                     float*, float*, float*, float*, int, void*, int, long,
                     long, int, int, int, int, float, const float*, bool,
                     hipStream_t);

namespace {

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be float32");
}

hipStream_t stream() { return at::hip::getCurrentHIPStream().stream(); }

// zero-filled workspace via hipMemsetAsync (cheaper than an ATen fill
// kernel; these workspaces are allocated on every call and the fill
// launches were ~2 ms/step in aggregate).
//
// NEVER memset while the stream is capturing into a hipGraph: on ROCm 7.2
// a memset enqueued during capture is mis-ordered at replay against the
// kernels consuming the workspace -- replays then intermittently read
// partially-zeroed GroupNorm/wgrad workspaces and produce garbage
// gradients (reproduced 6/6 at tiny config, OK 6/6 with the memset
// disabled; see profiles/README.md).  Inside capture the plain fill
// kernel is captured instead, which replays correctly.
torch::Tensor zeros_fast(at::IntArrayRef sizes,
                         const torch::TensorOptions& opt) {
  static const bool disabled = [] {
    const char* e = getenv("PVRAFT_NO_MEMSET");
    return e && e[0] == '1';
  }();
  hipStreamCaptureStatus cap = hipStreamCaptureStatusNone;
  hipStreamIsCapturing(stream(), &cap);
  if (disabled || cap != hipStreamCaptureStatusNone)
    return torch::zeros(sizes, opt);
  auto t = torch::empty(sizes, opt);
  if (t.numel() > 0)
    hipMemsetAsync(t.data_ptr(), 0, t.numel() * t.element_size(), stream());
  return t;
}

// persistent zero workspace for the GN forward reduction: the finalize
// kernel re-zeroes it after consuming, so one buffer per size serves every
// call and every hipGraph replay (stream-ordered) with no per-call
// allocation or fill.  Held forever (a few KB total).
torch::Tensor& persistent_ws(long len, const torch::TensorOptions& opt) {
  // leaked on purpose: a static map of CUDA tensors must not run its
  // destructor during process teardown (races CUDA context destruction)
  static auto* cache = new std::unordered_map<long, torch::Tensor>();
  auto it = cache->find(len);
  if (it == cache->end())
    it = cache->emplace(len, torch::zeros({len}, opt)).first;
  return it->second;
}

}  // namespace

torch::Tensor morton_keys(torch::Tensor xyz, torch::Tensor mn,
                          torch::Tensor inv_ext) {
  check_f32(xyz, "xyz");
  check_f32(mn, "mn");
  check_f32(inv_ext, "inv_ext");
  TORCH_CHECK(xyz.dim() == 3 && xyz.size(2) == 3, "xyz must be (B,N,3)");
  const int B = xyz.size(0);
  const long N = xyz.size(1);
  auto keys = torch::empty({B, N}, xyz.options().dtype(torch::kInt64));
  launch_morton_keys(xyz.data_ptr<float>(), mn.data_ptr<float>(),
                     inv_ext.data_ptr<float>(), keys.data_ptr<long>(), B, N,
                     stream());
  return keys;
}

torch::Tensor knn_graph(torch::Tensor xyz, int64_t k) {
  check_f32(xyz, "xyz");
  TORCH_CHECK(xyz.dim() == 3 && xyz.size(2) == 3, "xyz must be (B,N,3)");
  const int B = xyz.size(0), N = xyz.size(1);
  TORCH_CHECK(k >= 1 && k <= 48 && k <= N, "knn_graph requires 1 <= k <= 48, k <= N");
  auto out = torch::empty({B, N, k}, xyz.options().dtype(torch::kInt32));
  launch_knn_graph(xyz.data_ptr<float>(), out.data_ptr<int>(), B, N, (int)k,
                   stream());
  return out;
}

torch::Tensor gather_edge_concat_fwd(torch::Tensor feats, torch::Tensor idx,
                                     torch::Tensor xyz) {
  TORCH_CHECK(feats.is_cuda() && feats.is_contiguous(), "feats must be contiguous GPU");
  check_f32(xyz, "xyz");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous(), "idx must be contiguous int32");
  const bool bf16 = feats.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || feats.scalar_type() == torch::kFloat32, "fp32/bf16 only");
  const int B = feats.size(0), N = feats.size(1), C = feats.size(2), K = idx.size(2);
  auto out = torch::empty({B, C + 3, K, N}, feats.options());
  launch_gather_edge_fwd(feats.data_ptr(), idx.data_ptr<int>(),
                         xyz.data_ptr<float>(), out.data_ptr(), B, N, K, C,
                         bf16, stream());
  return out;
}

torch::Tensor gather_edge_concat_bwd(torch::Tensor gout, torch::Tensor idx,
                                     int64_t C) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous(), "gout must be contiguous GPU");
  const bool bf16 = gout.scalar_type() == torch::kBFloat16;
  const int B = gout.size(0), K = gout.size(2), N = gout.size(3);
  TORCH_CHECK(gout.size(1) == C + 3, "gout channel mismatch");
  auto gfeats = torch::empty({B, N, C}, gout.options().dtype(torch::kFloat32));
  launch_gather_edge_bwd(gout.data_ptr(), idx.data_ptr<int>(),
                         gfeats.data_ptr<float>(), B, N, K, (int)C, bf16,
                         stream());
  return bf16 ? gfeats.to(torch::kBFloat16) : gfeats;
}

torch::Tensor gather_edge_bwd_csr(torch::Tensor gT, torch::Tensor order,
                                  torch::Tensor offsets, int64_t K) {
  TORCH_CHECK(gT.is_cuda() && gT.is_contiguous(), "gT must be contiguous GPU");
  const bool bf16 = gT.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || gT.scalar_type() == torch::kFloat32, "fp32/bf16 only");
  TORCH_CHECK(order.scalar_type() == torch::kInt32 && order.is_contiguous());
  TORCH_CHECK(offsets.scalar_type() == torch::kInt32 && offsets.is_contiguous());
  const int B = gT.size(0), NK = gT.size(1), C = gT.size(2);
  const int N = NK / (int)K;
  auto grad = torch::empty({B, N, C}, gT.options());
  launch_gather_edge_bwd_csr(gT.data_ptr(), order.data_ptr<int>(),
                             offsets.data_ptr<int>(), grad.data_ptr(), B, N,
                             (int)K, C, bf16, stream());
  return grad;
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
  auto out = torch::empty({B, 4, k, N}, corr.options());
  auto idx = torch::empty({B, N, k}, corr.options().dtype(torch::kInt32));
  launch_knn_corr_fwd(corr.data_ptr<float>(), xyz.data_ptr<float>(),
                      coords.data_ptr<float>(), out.data_ptr<float>(),
                      idx.data_ptr<int>(), B, N, K, (int)k, stream());
  return {out, idx};
}

torch::Tensor knn_corr_bwd(torch::Tensor gout, torch::Tensor idx, int64_t K) {
  check_f32(gout, "gout");
  const int B = gout.size(0), k = gout.size(2), N = gout.size(3);
  auto gcorr = zeros_fast({B, N, K}, gout.options());
  launch_knn_corr_bwd(gout.data_ptr<float>(), idx.data_ptr<int>(),
                      gcorr.data_ptr<float>(), B, N, (int)K, k, stream());
  return gcorr;
}

// x (B, C, S) contiguous (S = flattened spatial); weight/bias (C) fp32.
// act: 0 none, 1 LeakyReLU(slope).  Returns {y, mean (B*G), rstd (B*G)}.
// slope_t: scalar fp32 tensor for act==2 (learnable PReLU slope)
std::vector<torch::Tensor> group_norm_act_fwd(torch::Tensor x, int64_t G,
                                              torch::Tensor weight,
                                              torch::Tensor bias, double eps,
                                              int64_t act, double slope,
                                              c10::optional<torch::Tensor> slope_t) {
  const float* slope_ptr = nullptr;
  if (act == 2) {
    TORCH_CHECK(slope_t.has_value() && slope_t->numel() == 1, "act=2 needs a scalar slope tensor");
    slope_ptr = slope_t->data_ptr<float>();
  }
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous GPU");
  TORCH_CHECK(x.dim() == 3, "x must be (B, C, S)");
  check_f32(weight, "weight");
  check_f32(bias, "bias");
  const int B = x.size(0), C = x.size(1);
  const long S = x.size(2);
  TORCH_CHECK(C % G == 0, "C % G != 0");
  const int rows = B * G;
  const long row_len = (C / G) * S;
  auto fopt = x.options().dtype(torch::kFloat32);
  // per-call deterministic partial-sum scratch (one slot pair per reduce
  // block) -- no persistent zeroed workspace, no finalize launch
  const int rchunks = gn_reduce_chunks(S, B, C, (int)G);
  auto scratch = torch::empty({(long)B * C * 2 * rchunks}, fopt);
  auto mean = torch::empty({rows}, fopt);
  auto rstd = torch::empty({rows}, fopt);
  auto y = torch::empty_like(x);
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat32,
              "group_norm_act: dtype must be float32 or bfloat16");
  launch_gn_fwd(x.data_ptr(), y.data_ptr(), scratch.data_ptr<float>(),
                mean.data_ptr<float>(), rstd.data_ptr<float>(),
                weight.data_ptr<float>(), bias.data_ptr<float>(), rows,
                row_len, S, C, (int)G, (float)eps, (int)act, (float)slope,
                slope_ptr, bf16, stream());
  return {y, mean, rstd};
}


// Finish a GN-family backward: either materialise fresh dweight/dbias/
// dslope tensors (plain autograd) or ACCUMULATE into the parameters' own
// fp32 grad buffers (deferred mode, returns an empty vector tail).
static std::vector<torch::Tensor> gn_grads_finish(
    torch::Tensor& ws, torch::Tensor dx, int rows, int C,
    const torch::TensorOptions& fopt,
    c10::optional<torch::Tensor> wt, c10::optional<torch::Tensor> bt,
    c10::optional<torch::Tensor> st) {
  if (wt.has_value()) {
    TORCH_CHECK(wt->is_cuda() && wt->is_contiguous() &&
                wt->scalar_type() == torch::kFloat32 && wt->numel() == C &&
                bt.has_value() && bt->numel() == C,
                "gn accumulate targets must be fp32 (C)");
    float* sp = nullptr;
    if (st.has_value() && st->defined() && st->numel() == 1)
      sp = st->data_ptr<float>();
    launch_gn_bwd_extract(ws.data_ptr<float>(), wt->data_ptr<float>(),
                          bt->data_ptr<float>(), sp, rows, C, 1, stream());
    return {dx};
  }
  auto dweight = torch::empty({C}, fopt);
  auto dbias = torch::empty({C}, fopt);
  auto dslope = torch::empty({1}, fopt);
  launch_gn_bwd_extract(ws.data_ptr<float>(), dweight.data_ptr<float>(),
                        dbias.data_ptr<float>(), dslope.data_ptr<float>(),
                        rows, C, 0, stream());
  return {dx, dweight, dbias, dslope};
}

std::vector<torch::Tensor> group_norm_act_bwd(torch::Tensor dy, torch::Tensor x,
                                              torch::Tensor mean,
                                              torch::Tensor rstd, int64_t G,
                                              torch::Tensor weight,
                                              torch::Tensor bias, int64_t act,
                                              double slope,
                                              c10::optional<torch::Tensor> slope_t,
                                              c10::optional<torch::Tensor> wtarget = c10::nullopt,
                                              c10::optional<torch::Tensor> btarget = c10::nullopt,
                                              c10::optional<torch::Tensor> starget = c10::nullopt) {
  const float* slope_ptr = nullptr;
  if (act == 2) {
    TORCH_CHECK(slope_t.has_value() && slope_t->numel() == 1, "act=2 needs a scalar slope tensor");
    slope_ptr = slope_t->data_ptr<float>();
  }
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous(), "dy/x must be contiguous");
  const int B = x.size(0), C = x.size(1);
  const long S = x.size(2);
  const int rows = B * G;
  const long row_len = (C / G) * S;
  auto fopt = x.options().dtype(torch::kFloat32);
  const int rchunks = gn_reduce_chunks(S, B, C, (int)G);
  auto scratch = torch::empty({(long)B * C * 5 * rchunks}, fopt);
  auto dx = torch::empty_like(x);
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat32,
              "group_norm_act: dtype must be float32 or bfloat16");
  TORCH_CHECK(dy.scalar_type() == x.scalar_type(), "dy/x dtype mismatch");
  // channel grads drain inside the apply pass: into fresh tensors, or
  // ACCUMULATED into the parameters' grad buffers (deferred mode)
  torch::Tensor dweight, dbias, dslope;
  float *dwp, *dbp, *dsp;
  int accumulate = 0;
  if (wtarget.has_value()) {
    accumulate = 1;
    dwp = wtarget->data_ptr<float>();
    dbp = btarget->data_ptr<float>();
    dsp = (starget.has_value() && starget->defined() && starget->numel() == 1)
              ? starget->data_ptr<float>() : nullptr;
  } else {
    dweight = torch::empty({C}, fopt);
    dbias = torch::empty({C}, fopt);
    dslope = torch::empty({1}, fopt);
    dwp = dweight.data_ptr<float>();
    dbp = dbias.data_ptr<float>();
    dsp = dslope.data_ptr<float>();
  }
  launch_gn_bwd(dy.data_ptr(), x.data_ptr(), mean.data_ptr<float>(),
                rstd.data_ptr<float>(), weight.data_ptr<float>(),
                bias.data_ptr<float>(), scratch.data_ptr<float>(), dwp, dbp,
                dsp, accumulate, dx.data_ptr(), rows, row_len, S, C, (int)G,
                (int)act, (float)slope, slope_ptr, bf16, stream());
  if (accumulate) return {dx};
  return {dx, dweight, dbias, dslope};
}

// x (B, C, K, N); GN stats over full (K, N); returns pooled
// {y (B,C,N), argmax (B,C,N) u8, mean, rstd}
std::vector<torch::Tensor> group_norm_act_maxpool_fwd(
    torch::Tensor x, int64_t G, torch::Tensor weight, torch::Tensor bias,
    double eps, int64_t act, double slope,
    c10::optional<torch::Tensor> slope_t) {
  const float* slope_ptr = nullptr;
  if (act == 2) {
    TORCH_CHECK(slope_t.has_value() && slope_t->numel() == 1, "act=2 needs a scalar slope tensor");
    slope_ptr = slope_t->data_ptr<float>();
  }
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4, "x must be contiguous (B,C,K,N)");
  check_f32(weight, "weight");
  check_f32(bias, "bias");
  const int B = x.size(0), C = x.size(1), K = x.size(2);
  const long N = x.size(3);
  TORCH_CHECK(K <= 255, "maxpool K must fit u8 argmax");
  const int rows = B * G;
  const long row_len = (C / G) * K * N;
  auto fopt = x.options().dtype(torch::kFloat32);
  const int rchunks = gnmp_reduce_chunks(N, K, B, C, (int)G);
  auto scratch = torch::empty({(long)B * C * 2 * rchunks}, fopt);
  auto mean = torch::empty({rows}, fopt);
  auto rstd = torch::empty({rows}, fopt);
  auto y = torch::empty({B, C, N}, x.options());
  auto am = torch::empty({B, C, N}, x.options().dtype(torch::kUInt8));
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat32, "fp32/bf16 only");
  launch_gnmp_fwd(x.data_ptr(), y.data_ptr(), am.data_ptr<unsigned char>(),
                  scratch.data_ptr<float>(), mean.data_ptr<float>(),
                  rstd.data_ptr<float>(), weight.data_ptr<float>(),
                  bias.data_ptr<float>(), rows, row_len, N, K, C, (int)G,
                  (float)eps, (int)act, (float)slope, slope_ptr, bf16, stream());
  return {y, am, mean, rstd};
}

std::vector<torch::Tensor> group_norm_act_maxpool_bwd(
    torch::Tensor dy, torch::Tensor x, torch::Tensor am, torch::Tensor mean,
    torch::Tensor rstd, int64_t G, torch::Tensor weight, torch::Tensor bias,
    int64_t act, double slope, c10::optional<torch::Tensor> slope_t,
    c10::optional<torch::Tensor> wtarget = c10::nullopt,
    c10::optional<torch::Tensor> btarget = c10::nullopt,
    c10::optional<torch::Tensor> starget = c10::nullopt) {
  const float* slope_ptr = nullptr;
  if (act == 2) {
    TORCH_CHECK(slope_t.has_value() && slope_t->numel() == 1, "act=2 needs a scalar slope tensor");
    slope_ptr = slope_t->data_ptr<float>();
  }
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous(), "dy/x must be contiguous");
  const int B = x.size(0), C = x.size(1), K = x.size(2);
  const long N = x.size(3);
  const int rows = B * G;
  const long row_len = (C / G) * K * N;
  auto fopt = x.options().dtype(torch::kFloat32);
  const int rchunks = gnmp_bwd_reduce_chunks(N, B, C, (int)G);
  auto scratch = torch::empty({(long)B * C * 5 * rchunks}, fopt);
  auto dx = torch::empty_like(x);
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  torch::Tensor dweight, dbias, dslope;
  float *dwp, *dbp, *dsp;
  int accumulate = 0;
  if (wtarget.has_value()) {
    accumulate = 1;
    dwp = wtarget->data_ptr<float>();
    dbp = btarget->data_ptr<float>();
    dsp = (starget.has_value() && starget->defined() && starget->numel() == 1)
              ? starget->data_ptr<float>() : nullptr;
  } else {
    dweight = torch::empty({C}, fopt);
    dbias = torch::empty({C}, fopt);
    dslope = torch::empty({1}, fopt);
    dwp = dweight.data_ptr<float>();
    dbp = dbias.data_ptr<float>();
    dsp = dslope.data_ptr<float>();
  }
  launch_gnmp_bwd(dy.data_ptr(), x.data_ptr(), am.data_ptr<unsigned char>(),
                  mean.data_ptr<float>(), rstd.data_ptr<float>(),
                  weight.data_ptr<float>(), bias.data_ptr<float>(),
                  scratch.data_ptr<float>(), dwp, dbp, dsp, accumulate,
                  dx.data_ptr(), rows, row_len, N, K, C, (int)G, (int)act,
                  (float)slope, slope_ptr, bf16, stream());
  if (accumulate) return {dx};
  return {dx, dweight, dbias, dslope};
}

// SetConv stage 1 on the linearly-restructured operands: wg (B, N, M)
// point-major = (fc1 W @ [feats; xyz]) per point; pooled output y and u8
// argmax are point-major too.  GN stats over the full (M/G, K, N) edge
// extent, identical to group_norm_act_maxpool.
std::vector<torch::Tensor> edge_gnmp_fwd(torch::Tensor wg, torch::Tensor idx,
                                         int64_t G, torch::Tensor weight,
                                         torch::Tensor bias, double eps,
                                         int64_t act, double slope,
                                         c10::optional<torch::Tensor> slope_t) {
  const float* slope_ptr = nullptr;
  if (act == 2) {
    TORCH_CHECK(slope_t.has_value() && slope_t->numel() == 1, "act=2 needs a scalar slope tensor");
    slope_ptr = slope_t->data_ptr<float>();
  }
  TORCH_CHECK(wg.is_cuda() && wg.is_contiguous() && wg.dim() == 3, "wg must be contiguous (B,N,M)");
  TORCH_CHECK(idx.scalar_type() == torch::kInt32 && idx.is_contiguous(), "idx must be contiguous int32");
  check_f32(weight, "weight");
  check_f32(bias, "bias");
  const int B = wg.size(0), M = wg.size(2), K = idx.size(2);
  const long N = wg.size(1);
  TORCH_CHECK(idx.size(0) == B && idx.size(1) == N, "idx/wg shape mismatch");
  TORCH_CHECK(M % G == 0 && G <= 8 && M <= 256 && M % 4 == 0,
              "edge_gnmp: need M % G == 0, M % 4 == 0, G <= 8, M <= 256");
  TORCH_CHECK(K <= 255, "edge_gnmp: K must fit u8 argmax");
  const bool bf16 = wg.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || wg.scalar_type() == torch::kFloat32, "fp32/bf16 only");
  const int rows = B * (int)G;
  auto fopt = wg.options().dtype(torch::kFloat32);
  auto& ws = persistent_ws((long)rows * 2, fopt);
  const int rchunks = egnmp_reduce_chunks(N, M, B);
  auto scratch = torch::empty({(long)rows * 2, (long)rchunks * B}, fopt);
  auto mean = torch::empty({rows}, fopt);
  auto rstd = torch::empty({rows}, fopt);
  auto y = torch::empty_like(wg);
  auto am = torch::empty({B, N, M}, wg.options().dtype(torch::kUInt8));
  // per-(n, c) gather extremes tracked by the reduce pass (the pick pass
  // is elementwise; the second full gather sweep is gone).  fp32: the
  // backward derives activation-branch signs from them.
  auto vmax = torch::empty({B, N, M}, fopt);
  auto vmin = torch::empty({B, N, M}, fopt);
  auto amax = torch::empty({B, N, M}, wg.options().dtype(torch::kUInt8));
  auto amin = torch::empty({B, N, M}, wg.options().dtype(torch::kUInt8));
  // saved for the closed-form backward centre term: per-point gather sum
  // (fp32) and the selected pre-GN extreme value
  auto vsum = torch::empty({B, N, M}, fopt);
  auto vsel = torch::empty({B, N, M}, fopt);
  launch_egnmp_fwd(wg.data_ptr(), idx.data_ptr<int>(),
                   scratch.data_ptr<float>(), ws.data_ptr<float>(),
                   mean.data_ptr<float>(), rstd.data_ptr<float>(),
                   weight.data_ptr<float>(), bias.data_ptr<float>(),
                   vmax.data_ptr<float>(), vmin.data_ptr<float>(),
                   amax.data_ptr<unsigned char>(),
                   amin.data_ptr<unsigned char>(), vsum.data_ptr<float>(),
                   y.data_ptr(),
                   am.data_ptr<unsigned char>(), vsel.data_ptr<float>(), B, N,
                   K, M, (int)G, (float)eps, (int)act, (float)slope,
                   slope_ptr, bf16, rchunks, stream());
  return {y, am, vsel, vsum, mean, rstd};
}

std::vector<torch::Tensor> edge_gnmp_bwd(
    torch::Tensor dy, torch::Tensor wg, torch::Tensor idx, torch::Tensor am,
    torch::Tensor vsel, torch::Tensor vsum,
    torch::Tensor offsets, torch::Tensor order_n, torch::Tensor order_j,
    torch::Tensor mean, torch::Tensor rstd, int64_t G,
    torch::Tensor weight, torch::Tensor bias,
    int64_t act, double slope, c10::optional<torch::Tensor> slope_t,
    c10::optional<torch::Tensor> wtarget = c10::nullopt,
    c10::optional<torch::Tensor> btarget = c10::nullopt,
    c10::optional<torch::Tensor> starget = c10::nullopt) {
  const float* slope_ptr = nullptr;
  if (act == 2) {
    TORCH_CHECK(slope_t.has_value() && slope_t->numel() == 1, "act=2 needs a scalar slope tensor");
    slope_ptr = slope_t->data_ptr<float>();
  }
  TORCH_CHECK(dy.is_contiguous() && wg.is_contiguous(), "dy/wg must be contiguous");
  TORCH_CHECK(dy.scalar_type() == wg.scalar_type(), "dy/wg dtype mismatch");
  TORCH_CHECK(offsets.scalar_type() == torch::kInt32 && offsets.is_contiguous());
  TORCH_CHECK(order_n.scalar_type() == torch::kInt32 && order_n.is_contiguous());
  TORCH_CHECK(order_j.scalar_type() == torch::kUInt8 && order_j.is_contiguous());
  TORCH_CHECK(offsets.scalar_type() == torch::kInt32 && offsets.is_contiguous());
  TORCH_CHECK(order_n.scalar_type() == torch::kInt32 && order_n.is_contiguous());
  TORCH_CHECK(order_j.scalar_type() == torch::kUInt8 && order_j.is_contiguous());
  const int B = wg.size(0), M = wg.size(2), K = idx.size(2);
  const long N = wg.size(1);
  const bool bf16 = wg.scalar_type() == torch::kBFloat16;
  const int rows = B * (int)G;
  auto fopt = wg.options().dtype(torch::kFloat32);
  const long ws_len = (long)rows * 2 + M * 2 + 1;
  auto& ws = persistent_ws(ws_len, fopt);
  const int rchunks = egnmp_reduce_chunks(N, M, B);
  auto scratch = torch::empty({ws_len, (long)rchunks * B}, fopt);
  auto dwg = torch::empty_like(wg);
  launch_egnmp_bwd(dy.data_ptr(), wg.data_ptr(), idx.data_ptr<int>(),
                   am.data_ptr<unsigned char>(), vsel.data_ptr<float>(),
                   vsum.data_ptr<float>(),
                   offsets.data_ptr<int>(), order_n.data_ptr<int>(),
                   order_j.data_ptr<unsigned char>(),
                   mean.data_ptr<float>(),
                   rstd.data_ptr<float>(), weight.data_ptr<float>(),
                   bias.data_ptr<float>(), scratch.data_ptr<float>(),
                   ws.data_ptr<float>(), dwg.data_ptr(), B, N, K, M, (int)G,
                   (int)act, (float)slope, slope_ptr, bf16, rchunks,
                   stream());
  return gn_grads_finish(ws, dwg, rows, M, fopt, wtarget, btarget, starget);
}

// Fused kNN-correlation branch head (csrc/knn_gnmp.hip): conv(4->C) ->
// GroupNorm(G) -> PReLU -> max over K, without the (B, C, K, N)
// intermediate.  raw (B, 4, K, N) fp32; y is (B, N, C) point-major (the
// caller transposes with the narrow-R fast path).
std::vector<torch::Tensor> knn_gnmp_fwd(torch::Tensor raw, torch::Tensor W,
                                        torch::Tensor cb, int64_t G,
                                        torch::Tensor gamma,
                                        torch::Tensor beta, double eps,
                                        torch::Tensor slope_t,
                                        bool out_bf16) {
  check_f32(raw, "raw");
  check_f32(W, "W");
  check_f32(cb, "cb");
  check_f32(gamma, "gamma");
  check_f32(beta, "beta");
  TORCH_CHECK(raw.dim() == 4 && raw.size(1) == 4, "raw must be (B,4,K,N)");
  TORCH_CHECK(slope_t.numel() == 1 && slope_t.scalar_type() == torch::kFloat32);
  const int B = raw.size(0), K = raw.size(2);
  const long N = raw.size(3);
  const int C = W.size(0);
  TORCH_CHECK(W.dim() == 2 && W.size(1) == 4, "W must be (C,4)");
  TORCH_CHECK(C % 16 == 0 && C % G == 0 && C / G >= 4 && K <= 255,
              "knn_gnmp: need C % 16 == 0, C/G >= 4, K <= 255");
  auto fopt = raw.options();
  const int rows = B * (int)G;
  auto& ws = persistent_ws((long)rows * 2, fopt);
  int nblk = (int)((N + 255) / 256);
  if (nblk < 1) nblk = 1;
  const int chunks_f = C / 4;
  // transient (write-then-read within this enqueued call): persistent
  // buffers, NOT per-call pool allocations -- inside a captured training
  // step the per-call workspaces (~85 MB x 12 calls) push the graph pool
  // into the ROCm page-mapping-bug regime (silent garbage at replay;
  // the op alone replays clean, scripts/graph_kg_repro.py)
  auto scratch = persistent_ws((long)rows * 2 * nblk * chunks_f * B, fopt)
                     .view({(long)rows * 2, (long)nblk * chunks_f * B});
  auto mean = torch::empty({rows}, fopt);
  auto rstd = torch::empty({rows}, fopt);
  auto vmax = torch::empty({B, N, C}, fopt);
  auto vmin = torch::empty({B, N, C}, fopt);
  auto vsel = torch::empty({B, N, C}, fopt);
  auto amax = torch::empty({B, N, C}, raw.options().dtype(torch::kUInt8));
  auto amin = torch::empty({B, N, C}, raw.options().dtype(torch::kUInt8));
  auto am = torch::empty({B, N, C}, raw.options().dtype(torch::kUInt8));
  auto y = torch::empty({B, N, C}, raw.options().dtype(
                                       out_bf16 ? torch::kBFloat16
                                                : torch::kFloat32));
  launch_kg_fwd(raw.data_ptr<float>(), W.data_ptr<float>(),
                cb.data_ptr<float>(), scratch.data_ptr<float>(),
                ws.data_ptr<float>(), mean.data_ptr<float>(),
                rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                beta.data_ptr<float>(), vmax.data_ptr<float>(),
                vmin.data_ptr<float>(), amax.data_ptr<unsigned char>(),
                amin.data_ptr<unsigned char>(), y.data_ptr(),
                am.data_ptr<unsigned char>(), vsel.data_ptr<float>(), B, N,
                K, C, (int)G, (float)eps, slope_t.data_ptr<float>(),
                out_bf16, nblk, stream());
  return {y, am, vsel, mean, rstd};
}

std::vector<torch::Tensor> knn_gnmp_bwd(
    torch::Tensor dyT, torch::Tensor raw, torch::Tensor W, torch::Tensor cb,
    torch::Tensor am, torch::Tensor vsel, torch::Tensor mean,
    torch::Tensor rstd, int64_t G, torch::Tensor gamma, torch::Tensor beta,
    torch::Tensor slope_t) {
  TORCH_CHECK(dyT.is_cuda() && dyT.is_contiguous() && dyT.dim() == 3,
              "dyT must be contiguous (B,N,C)");
  check_f32(raw, "raw");
  const int B = raw.size(0), K = raw.size(2);
  const long N = raw.size(3);
  const int C = W.size(0);
  const bool bf16 = dyT.scalar_type() == torch::kBFloat16;
  auto fopt = raw.options();
  const int rows = B * (int)G;
  int nblk = (int)((N + 255) / 256);
  if (nblk < 1) nblk = 1;
  const long n_out = (long)rows * 2 + C * 2 + 1;
  // transient call-internal workspaces: persistent (see knn_gnmp_fwd).
  // draw (the returned gradient) and ws2 (dW/dcb grad views) stay
  // per-call: autograd's execution order gives no cross-node liveness
  // guarantee for returned gradients.
  auto scratch = persistent_ws(n_out * nblk * B, fopt)
                     .view({n_out, (long)nblk * B});
  auto ws = persistent_ws(n_out, fopt);
  const int chunks_b = C / 8;
  auto wscratch =
      persistent_ws((long)C * 5 * nblk * chunks_b * 4 * B, fopt)
          .view({(long)C * 5, (long)nblk * chunks_b * 4 * B});
  auto ws2 = torch::empty({(long)C * 5}, fopt);
  auto draw_part =
      persistent_ws((long)chunks_b * B * 4 * K * N, fopt)
          .view({(long)chunks_b, B, 4, K, N});
  auto draw = torch::empty({B, 4, K, N}, fopt);
  launch_kg_bwd(dyT.data_ptr(), raw.data_ptr<float>(), W.data_ptr<float>(),
                cb.data_ptr<float>(), am.data_ptr<unsigned char>(),
                vsel.data_ptr<float>(), mean.data_ptr<float>(),
                rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                beta.data_ptr<float>(), scratch.data_ptr<float>(),
                ws.data_ptr<float>(), wscratch.data_ptr<float>(),
                ws2.data_ptr<float>(), draw_part.data_ptr<float>(),
                draw.data_ptr<float>(), B, N, K, C, (int)G,
                slope_t.data_ptr<float>(), bf16, nblk, stream());
  auto tail = gn_grads_finish(ws, draw, rows, C, fopt, c10::nullopt,
                              c10::nullopt, c10::nullopt);
  auto dW = ws2.narrow(0, 0, (long)C * 4).view({C, 4});
  auto dcb = ws2.narrow(0, (long)C * 4, C);
  // tail = {draw, dgamma, dbeta, dslope}
  return {draw, dW, dcb, tail[1], tail[2], tail[3]};
}

// Capture-safe batched fp32->bf16 mirror refresh (csrc/cast_pack.hip):
// pointer tables ride kernel arguments, so a hipGraph replay re-reads the
// live fp32 weights (ATen _foreach_copy_'s staging upload does not).
void multi_cast_bf16(std::vector<torch::Tensor> srcs,
                     std::vector<torch::Tensor> dsts) {
  const int n = (int)srcs.size();
  TORCH_CHECK((int)dsts.size() == n, "src/dst count mismatch");
  std::vector<CastChunk> chunks;
  chunks.reserve((n + 127) / 128);
  CastChunk cur;
  cur.count = 0;
  for (int i = 0; i < n; ++i) {
    auto& s = srcs[i];
    auto& d = dsts[i];
    TORCH_CHECK(s.is_cuda() && s.is_contiguous() &&
                s.scalar_type() == torch::kFloat32, "src must be fp32 contig");
    TORCH_CHECK(d.is_cuda() && d.is_contiguous() &&
                d.scalar_type() == torch::kBFloat16 && d.numel() == s.numel(),
                "dst must be matching bf16");
    cur.d[cur.count++] = CastDesc{s.data_ptr<float>(), d.data_ptr(),
                                  (int)s.numel()};
    if (cur.count == 128) {
      chunks.push_back(cur);
      cur.count = 0;
    }
  }
  if (cur.count) chunks.push_back(cur);
  if (!chunks.empty())
    launch_multi_cast(chunks.data(), (int)chunks.size(), stream());
}

// dy (B, Co, S) bf16, x (B, Ci, S) bf16 -> dW (Co, Ci) fp32 (split-K MFMA)
std::vector<torch::Tensor> pw_wgrad(torch::Tensor dy, torch::Tensor x,
                                    int64_t schunks = 0, bool with_bias = false) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 3);
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 && x.scalar_type() == torch::kBFloat16,
              "pw_wgrad is bf16-only");
  TORCH_CHECK(dy.size(0) == x.size(0) && dy.size(2) == x.size(2), "shape mismatch");
  const int B = dy.size(0), Co = dy.size(1), Ci = x.size(1);
  const long S = dy.size(2);
  // dw and dbias share one zeroed allocation (both are atomic-accumulated;
  // a two-stage partial-store + reduce variant measured 2x SLOWER -- the
  // partial-buffer write+read traffic dwarfs the L2 atomics on this tiny
  // output, scripts/kernel_bench.py)
  auto flat = zeros_fast({(long)Co * Ci + (with_bias ? Co : 0)},
                         dy.options().dtype(torch::kFloat32));
  auto dw = flat.narrow(0, 0, (long)Co * Ci).view({Co, Ci});
  torch::Tensor dbias;
  float* dbias_ptr = nullptr;
  if (with_bias) {
    dbias = flat.narrow(0, (long)Co * Ci, Co);
    dbias_ptr = dbias.data_ptr<float>();
  }
  launch_pw_wgrad(dy.data_ptr(), x.data_ptr(), dw.data_ptr<float>(), dbias_ptr,
                  B, Co, Ci, S, (int)schunks, stream());
  return {dw, dbias};
}

// Fused MFMA 1x1-conv forward: y = act(sum_i W_i @ x_i + bias + addend)
// in one launch (csrc/pw_fwd.hip)
void launch_pw_fwd(const void**, const void**, const int*, int, const float*,
                   const void*, long, int, void*, int, int, long, int,
                   hipStream_t);

torch::Tensor pw_fwd(std::vector<torch::Tensor> ws,
                     std::vector<torch::Tensor> xs,
                     c10::optional<torch::Tensor> bias,
                     c10::optional<torch::Tensor> addend, int64_t act) {
  const int n = (int)ws.size();
  TORCH_CHECK(n >= 1 && n <= 4 && (int)xs.size() == n, "1..4 parts");
  const int Co = ws[0].size(0);
  const int B = xs[0].size(0);
  const long S = xs[0].size(2);
  const void* wp[4];
  const void* xp[4];
  int cis[4];
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(ws[i].is_cuda() && ws[i].is_contiguous() && ws[i].dim() == 2 &&
                ws[i].scalar_type() == torch::kBFloat16, "w must be (Co,Ci) bf16");
    TORCH_CHECK(xs[i].is_cuda() && xs[i].is_contiguous() && xs[i].dim() == 3 &&
                xs[i].scalar_type() == torch::kBFloat16, "x must be (B,Ci,S) bf16");
    TORCH_CHECK(ws[i].size(0) == Co && xs[i].size(0) == B && xs[i].size(2) == S);
    TORCH_CHECK(ws[i].size(1) == xs[i].size(1), "w/x channel mismatch");
    TORCH_CHECK(ws[i].size(1) <= 224, "pw_fwd: Ci <= 224");
    wp[i] = ws[i].data_ptr();
    xp[i] = xs[i].data_ptr();
    cis[i] = (int)ws[i].size(1);
  }
  TORCH_CHECK(Co <= 256, "pw_fwd: Co <= 256");
  const float* bptr = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32 && bias->is_contiguous() &&
                bias->numel() == Co, "bias must be fp32 (Co)");
    bptr = bias->data_ptr<float>();
  }
  const void* aptr = nullptr;
  long abstride = 0;
  int afp32 = 0;
  if (addend.has_value() && addend->defined()) {
    auto& a = *addend;
    TORCH_CHECK(a.is_cuda() && a.dim() == 3 && a.size(0) == B &&
                a.size(1) == Co && a.size(2) == S, "addend must be (B,Co,S) view");
    TORCH_CHECK(a.stride(2) == 1 && a.stride(1) == S,
                "addend rows must be contiguous");
    afp32 = a.scalar_type() == torch::kFloat32;
    TORCH_CHECK(afp32 || a.scalar_type() == torch::kBFloat16, "addend fp32/bf16");
    aptr = a.data_ptr();
    abstride = a.stride(0);
  }
  auto y = torch::empty({B, (long)Co, S}, xs[0].options());
  launch_pw_fwd(wp, xp, cis, n, bptr, aptr, abstride, afp32, y.data_ptr(), B,
                Co, S, (int)act, stream());
  return y;
}

// Batched deferred weight gradients: many (dy, x) -> dW problems in ONE
// kernel launch, each ACCUMULATED into its weight's existing fp32 grad
// buffer (no per-call zero-fill, no autograd accumulate-add).  ~180 such
// problems per train step (the GRU loop's conv stacks) were
// launch/atomic-floor-bound when launched one-by-one.
// dbs entries with numel 0 mean "no bias gradient for this job".
// Returns {host_desc, dev_desc} -- the caller must keep both alive while a
// hipGraph captured over this call can still replay (the recorded H2D copy
// re-reads the pinned host buffer).
#include "pw_wgrad_job.h"
void launch_pw_wgrad_batched(const void*, const unsigned int*, int,
                             hipStream_t);
void launch_desc_fill(void*, const void*, long, hipStream_t);

std::vector<torch::Tensor> pw_wgrad_batched(std::vector<torch::Tensor> dys,
                                            std::vector<torch::Tensor> xs,
                                            std::vector<torch::Tensor> dws,
                                            std::vector<torch::Tensor> dbs) {
  const int n = (int)dys.size();
  TORCH_CHECK(n >= 1 && n <= 4095, "1..4095 jobs");
  TORCH_CHECK((int)xs.size() == n && (int)dws.size() == n && (int)dbs.size() == n);
  std::vector<PwWgradJob> jobs(n);
  long base_blocks = 0;
  const int TILE = 64;
  for (int j = 0; j < n; ++j) {
    auto& dy = dys[j];
    auto& x = xs[j];
    auto& dw = dws[j];
    TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 3 &&
                dy.scalar_type() == torch::kBFloat16, "dy must be (B,Co,S) bf16");
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3 &&
                x.scalar_type() == torch::kBFloat16, "x must be (B,Ci,S) bf16");
    TORCH_CHECK(dy.size(0) == x.size(0) && dy.size(2) == x.size(2), "dy/x mismatch");
    TORCH_CHECK(dw.is_cuda() && dw.is_contiguous() &&
                dw.scalar_type() == torch::kFloat32, "dw must be contiguous fp32");
    const int Co = dy.size(1), Ci = x.size(1);
    TORCH_CHECK(dw.numel() == (long)Co * Ci, "dw numel mismatch");
    TORCH_CHECK(Co <= 16 * TILE && Ci <= 16 * TILE, "Co/Ci too large for 4-bit tile index");
    float* dbias = nullptr;
    if (dbs[j].defined() && dbs[j].numel() > 0) {
      TORCH_CHECK(dbs[j].is_cuda() && dbs[j].is_contiguous() &&
                  dbs[j].scalar_type() == torch::kFloat32 && dbs[j].numel() == Co);
      dbias = dbs[j].data_ptr<float>();
    }
    jobs[j] = PwWgradJob{dy.data_ptr(), x.data_ptr(), dw.data_ptr<float>(),
                         dbias, Co, Ci, (int)dy.size(0), 1, dy.size(2)};
    base_blocks += (long)((Co + TILE - 1) / TILE) * ((Ci + TILE - 1) / TILE) *
                   dy.size(0);
  }
  // split each job's reduction so every block gets ~96 K-loop iterations
  // (~3072 reduction elements): a single global split starved the few
  // huge-S jobs (S ~ 262k, the kNN-branch convs), whose straggler blocks
  // then dominated the whole launch
  (void)base_blocks;
  for (int j = 0; j < n; ++j) {
    long sc = (jobs[j].S + 3071) / 3072;
    if (sc < 1) sc = 1;
    if ((long)jobs[j].B * sc > 4095) sc = 4095 / jobs[j].B;
    jobs[j].schunks = (int)sc;
  }
  std::vector<unsigned int> map;
  map.reserve(4096);
  for (int j = 0; j < n; ++j) {
    const int to = (jobs[j].Co + TILE - 1) / TILE;
    const int ti = (jobs[j].Ci + TILE - 1) / TILE;
    const int bzs = jobs[j].B * jobs[j].schunks;
    for (int bz = 0; bz < bzs; ++bz)
      for (int a = 0; a < to; ++a)
        for (int b = 0; b < ti; ++b)
          map.push_back(((unsigned)j << 20) | ((unsigned)bz << 8) |
                        ((unsigned)a << 4) | (unsigned)b);
  }
  const long jbytes = (long)n * sizeof(PwWgradJob);
  const long mbytes = (long)map.size() * sizeof(unsigned int);
  std::vector<unsigned char> blob(jbytes + mbytes);
  memcpy(blob.data(), jobs.data(), jbytes);
  memcpy(blob.data() + jbytes, map.data(), mbytes);
  auto dev = torch::empty({jbytes + mbytes}, dys[0].options().dtype(torch::kUInt8));
  // descriptor bytes travel as kernel ARGUMENTS (desc_fill_kernel): plain
  // launches that hipGraph capture records like any other kernel -- no
  // H2D memcpy / events / pinned memory, which all invalidate capture on
  // this stack.  Outside capture the same path is a handful of ~2 us
  // launches.
  launch_desc_fill(dev.data_ptr(), blob.data(), jbytes + mbytes, stream());
  launch_pw_wgrad_batched(dev.data_ptr(),
                          (const unsigned int*)((char*)dev.data_ptr() + jbytes),
                          (int)map.size(), stream());
  // caller keeps dev alive for the lifetime of any capturing graph
  return {dev};
}

// (B, R, C) view (row-contiguous, arbitrary batch stride) -> (B, C, R)
torch::Tensor batched_transpose(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3, "x must be 3-D GPU");
  TORCH_CHECK(x.stride(2) == 1 && x.stride(1) == x.size(2),
              "rows must be contiguous");
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat32, "fp32/bf16 only");
  const int B = x.size(0);
  const long R = x.size(1), C = x.size(2);
  auto out = torch::empty({B, C, R}, x.options());
  launch_transpose(x.data_ptr(), out.data_ptr(), x.stride(0), B, R, C, bf16,
                   stream());
  return out;
}

std::vector<torch::Tensor> pv_corr_fused_fwd(torch::Tensor corr,
                                             torch::Tensor xyz,
                                             torch::Tensor coords,
                                             double base_scale,
                                             int64_t num_levels, int64_t k) {
  check_f32(corr, "corr");
  check_f32(xyz, "xyz");
  check_f32(coords, "coords");
  const int B = corr.size(0), N = corr.size(1), K = corr.size(2);
  TORCH_CHECK(K <= 512 && num_levels >= 1 && num_levels <= 4 && k >= 1 && k <= 64 && k <= K);
  auto vox = torch::empty({B, num_levels * 27, N}, corr.options());
  auto knn = torch::empty({B, 4, k, N}, corr.options());
  auto idx = torch::empty({B, N, k}, corr.options().dtype(torch::kInt32));
  launch_pv_corr_fused_fwd(corr.data_ptr<float>(), xyz.data_ptr<float>(),
                           coords.data_ptr<float>(), vox.data_ptr<float>(),
                           knn.data_ptr<float>(), idx.data_ptr<int>(), B, N, K,
                           (int)num_levels, (int)k, (float)base_scale, stream());
  return {vox, knn, idx};
}

torch::Tensor pv_corr_fused_bwd(torch::Tensor g_vox, torch::Tensor g_knn,
                                torch::Tensor xyz, torch::Tensor coords,
                                torch::Tensor knn_idx, int64_t num_levels,
                                int64_t k, double base_scale) {
  check_f32(g_vox, "g_vox");
  check_f32(g_knn, "g_knn");
  const int B = xyz.size(0), N = xyz.size(1), K = xyz.size(2);
  auto gcorr = torch::empty({B, N, K}, g_vox.options());
  launch_pv_corr_fused_bwd(g_vox.data_ptr<float>(), g_knn.data_ptr<float>(),
                           xyz.data_ptr<float>(), coords.data_ptr<float>(),
                           knn_idx.data_ptr<int>(), gcorr.data_ptr<float>(), B,
                           N, K, (int)num_levels, (int)k, (float)base_scale,
                           stream());
  return gcorr;
}

// Fused bf16 MFMA correlation GEMM + streaming top-K truncation
// (reference corr.py:95-99 + corr.py:37): f1t (B,N,C), f2t (B,M,C) bf16
// point-major -> top-K values (B,N,K) fp32 + indices (B,N,K) i32; the
// (N, M) correlation matrix is never materialised.  Top-K SET is
// unordered (all consumers are order-invariant).
void launch_corr_topk(const void*, const void*, float*, float*, int*, float*,
                      int*, int*, int, int, int, int, int, float,
                      hipStream_t);

std::vector<torch::Tensor> corr_topk(torch::Tensor f1t, torch::Tensor f2t,
                                     int64_t K) {
  TORCH_CHECK(f1t.is_cuda() && f1t.is_contiguous() && f1t.dim() == 3 &&
              f1t.scalar_type() == torch::kBFloat16, "f1t must be (B,N,C) bf16");
  TORCH_CHECK(f2t.is_cuda() && f2t.is_contiguous() && f2t.dim() == 3 &&
              f2t.scalar_type() == torch::kBFloat16, "f2t must be (B,M,C) bf16");
  const int B = f1t.size(0), N = f1t.size(1), C = f1t.size(2);
  const int M = f2t.size(1);
  TORCH_CHECK(f2t.size(0) == B && f2t.size(2) == C, "f1t/f2t mismatch");
  TORCH_CHECK(C % 32 == 0 && C <= 256, "need C % 32 == 0, C <= 256");
  TORCH_CHECK(K >= 1 && K <= M && K <= 1024, "need 1 <= K <= min(M, 1024)");
  auto fopt = f1t.options().dtype(torch::kFloat32);
  auto iopt = f1t.options().dtype(torch::kInt32);
  const long R = (long)B * N;
  auto thr = torch::empty({R, 2}, fopt);
  auto out_v = torch::empty({B, (long)N, K}, fopt);
  auto out_i = torch::empty({B, (long)N, K}, iopt);
  auto band_v = torch::empty({R, 704}, fopt);
  auto band_i = torch::empty({R, 704}, iopt);
  auto cnt = zeros_fast({R, 2}, iopt);
  launch_corr_topk(f1t.data_ptr(), f2t.data_ptr(), thr.data_ptr<float>(),
                   out_v.data_ptr<float>(), out_i.data_ptr<int>(),
                   band_v.data_ptr<float>(), band_i.data_ptr<int>(),
                   cnt.data_ptr<int>(), B, N, M, C, (int)K,
                   1.0f / sqrtf((float)C), stream());
  return {out_v, out_i};
}

// vals (R, M) fp32 -> top-K largest per row: values (R, K), idx (R, K)
std::vector<torch::Tensor> topk_rows(torch::Tensor vals, int64_t K) {
  check_f32(vals, "vals");
  TORCH_CHECK(vals.dim() == 2, "vals must be (R, M)");
  const long R = vals.size(0);
  const int M = vals.size(1);
  TORCH_CHECK(M <= 8192, "topk_rows supports M <= 8192");
  TORCH_CHECK(K >= 1 && K <= M, "need 1 <= K <= M");
  auto out_v = torch::empty({R, K}, vals.options());
  auto out_i = torch::empty({R, K}, vals.options().dtype(torch::kInt32));
  launch_topk_rows(vals.data_ptr<float>(), out_v.data_ptr<float>(),
                   out_i.data_ptr<int>(), R, M, (int)K, stream());
  return {out_v, out_i};
}

namespace {

void check_gru(const torch::Tensor& t, const torch::Tensor& like,
               const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous(), name,
              " must be a contiguous GPU tensor");
  TORCH_CHECK(t.scalar_type() == like.scalar_type(), name,
              " dtype mismatch in fused GRU gates");
}

}  // namespace

// pre (B,2H,N) [z|r preactivations], h (B,H,N) -> z, r, rh (each (B,H,N))
std::vector<torch::Tensor> gru_zr_fwd(torch::Tensor pre, torch::Tensor h) {
  TORCH_CHECK(pre.is_cuda() && pre.is_contiguous() && h.is_contiguous());
  TORCH_CHECK(pre.dim() == 3 && h.dim() == 3 && pre.size(0) == h.size(0) &&
              pre.size(1) == 2 * h.size(1) && pre.size(2) == h.size(2));
  const bool bf16 = pre.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || pre.scalar_type() == torch::kFloat32);
  check_gru(h, pre, "h");
  auto z = torch::empty_like(h), r = torch::empty_like(h),
       rh = torch::empty_like(h);
  launch_gru_zr_fwd(pre.data_ptr(), h.data_ptr(), z.data_ptr(), r.data_ptr(),
                    rh.data_ptr(), h.size(0), (long)h.size(1) * h.size(2),
                    bf16, stream());
  return {z, r, rh};
}

std::vector<torch::Tensor> gru_zr_bwd(torch::Tensor dz, torch::Tensor drh,
                                      torch::Tensor z, torch::Tensor r,
                                      torch::Tensor h) {
  const bool bf16 = z.scalar_type() == torch::kBFloat16;
  for (auto* t : {&dz, &drh, &r, &h}) check_gru(*t, z, "gru_zr_bwd input");
  auto dpre = torch::empty({h.size(0), 2 * h.size(1), h.size(2)}, h.options());
  auto dh = torch::empty_like(h);
  launch_gru_zr_bwd(dz.data_ptr(), drh.data_ptr(), z.data_ptr(), r.data_ptr(),
                    h.data_ptr(), dpre.data_ptr(), dh.data_ptr(), h.size(0),
                    (long)h.size(1) * h.size(2), bf16, stream());
  return {dpre, dh};
}

// pre_q, z, h (B,H,N) -> q, hnew
std::vector<torch::Tensor> gru_q_fwd(torch::Tensor pre, torch::Tensor z,
                                     torch::Tensor h) {
  const bool bf16 = pre.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || pre.scalar_type() == torch::kFloat32);
  for (auto* t : {&pre, &z, &h}) check_gru(*t, pre, "gru_q_fwd input");
  auto q = torch::empty_like(h), hnew = torch::empty_like(h);
  launch_gru_q_fwd(pre.data_ptr(), z.data_ptr(), h.data_ptr(), q.data_ptr(),
                   hnew.data_ptr(), h.size(0), (long)h.size(1) * h.size(2),
                   bf16, stream());
  return {q, hnew};
}

std::vector<torch::Tensor> gru_q_bwd(torch::Tensor dhnew, torch::Tensor q,
                                     torch::Tensor z, torch::Tensor h) {
  const bool bf16 = q.scalar_type() == torch::kBFloat16;
  for (auto* t : {&dhnew, &z, &h}) check_gru(*t, q, "gru_q_bwd input");
  auto dpre = torch::empty_like(h), dz = torch::empty_like(h),
       dh = torch::empty_like(h);
  launch_gru_q_bwd(dhnew.data_ptr(), q.data_ptr(), z.data_ptr(), h.data_ptr(),
                   dpre.data_ptr(), dz.data_ptr(), dh.data_ptr(), h.size(0),
                   (long)h.size(1) * h.size(2), bf16, stream());
  return {dpre, dz, dh};
}

// fused gamma-weighted sequence loss over T flows -> {loss, denom}
std::vector<torch::Tensor> seq_loss_fwd(std::vector<torch::Tensor> flows,
                                        torch::Tensor gt, torch::Tensor mask,
                                        double gamma) {
  const int T = (int)flows.size();
  TORCH_CHECK(T >= 1 && T <= 32, "seq_loss supports 1..32 flows");
  check_f32(gt, "gt");
  check_f32(mask, "mask");
  const long BN = gt.numel() / 3;
  TORCH_CHECK(mask.numel() == BN, "mask/gt shape mismatch");
  const float* ptrs[32];
  for (int t = 0; t < T; ++t) {
    check_f32(flows[t], "flow");
    TORCH_CHECK(flows[t].numel() == BN * 3, "flow shape mismatch");
    ptrs[t] = flows[t].data_ptr<float>();
  }
  auto fopt = gt.options();
  // persistent self-cleaning workspace (finalize re-zeroes); leaked on
  // purpose so its destructor never races CUDA teardown at exit
  static torch::Tensor* ws_p = new torch::Tensor();
  torch::Tensor& ws = *ws_p;
  if (!ws.defined()) ws = torch::zeros({33}, fopt);
  auto loss = torch::empty({}, fopt);
  auto denom = torch::empty({}, fopt);
  launch_seq_loss_fwd(ptrs, gt.data_ptr<float>(), mask.data_ptr<float>(),
                      ws.data_ptr<float>(), loss.data_ptr<float>(),
                      denom.data_ptr<float>(), BN, T, 1, (float)gamma,
                      stream());
  return {loss, denom};
}

std::vector<torch::Tensor> seq_loss_bwd(std::vector<torch::Tensor> flows,
                                        torch::Tensor gt, torch::Tensor mask,
                                        torch::Tensor dloss,
                                        torch::Tensor denom, double gamma) {
  const int T = (int)flows.size();
  TORCH_CHECK(T >= 1 && T <= 32);
  const long BN = gt.numel() / 3;
  const float* ptrs[32];
  float* gptrs[32];
  std::vector<torch::Tensor> grads;
  grads.reserve(T);
  for (int t = 0; t < T; ++t) {
    ptrs[t] = flows[t].data_ptr<float>();
    grads.push_back(torch::empty_like(flows[t]));
    gptrs[t] = grads[t].data_ptr<float>();
  }
  launch_seq_loss_bwd(ptrs, gptrs, gt.data_ptr<float>(),
                      mask.data_ptr<float>(), dloss.data_ptr<float>(),
                      denom.data_ptr<float>(), BN, T, 1, (float)gamma,
                      stream());
  return grads;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("topk_rows", &topk_rows);
  m.def("corr_topk", &corr_topk);
  m.def("seq_loss_fwd", &seq_loss_fwd);
  m.def("seq_loss_bwd", &seq_loss_bwd);
  m.def("gru_zr_fwd", &gru_zr_fwd);
  m.def("gru_zr_bwd", &gru_zr_bwd);
  m.def("gru_q_fwd", &gru_q_fwd);
  m.def("gru_q_bwd", &gru_q_bwd);
  m.def("pv_corr_fused_fwd", &pv_corr_fused_fwd);
  m.def("pv_corr_fused_bwd", &pv_corr_fused_bwd);
  m.def("pw_wgrad", &pw_wgrad, pybind11::arg("dy"), pybind11::arg("x"), pybind11::arg("schunks") = 0, pybind11::arg("with_bias") = false);
  m.def("pw_wgrad_batched", &pw_wgrad_batched);
  m.def("pw_fwd", &pw_fwd);
  m.def("batched_transpose", &batched_transpose);
  m.def("group_norm_act_fwd", &group_norm_act_fwd);
  m.def("group_norm_act_bwd", &group_norm_act_bwd);
  m.def("group_norm_act_maxpool_fwd", &group_norm_act_maxpool_fwd);
  m.def("group_norm_act_maxpool_bwd", &group_norm_act_maxpool_bwd);
  m.def("edge_gnmp_fwd", &edge_gnmp_fwd);
  m.def("edge_gnmp_bwd", &edge_gnmp_bwd);
  m.def("morton_keys", &morton_keys);
  m.def("multi_cast_bf16", &multi_cast_bf16);
  m.def("knn_gnmp_fwd", &knn_gnmp_fwd);
  m.def("knn_gnmp_bwd", &knn_gnmp_bwd);
  m.def("knn_graph", &knn_graph, "fused kNN graph (CDNA4)");
  m.def("gather_edge_concat_fwd", &gather_edge_concat_fwd);
  m.def("gather_edge_concat_bwd", &gather_edge_concat_bwd);
  m.def("gather_edge_bwd_csr", &gather_edge_bwd_csr);
  m.def("voxel_corr_fwd", &voxel_corr_fwd);
  m.def("voxel_corr_bwd", &voxel_corr_bwd);
  m.def("knn_corr_fwd", &knn_corr_fwd);
  m.def("knn_corr_bwd", &knn_corr_bwd);
}
