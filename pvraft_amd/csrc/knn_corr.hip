// K5: kNN correlation lookup, forward + backward (capability of reference
// model/corr.py:75-93: squared distance to the K truncated candidates,
// topk-32, gather corr + relative xyz).
//
// Geometry: one wave per point (block = 256 = 4 points).  The candidate
// distances live in up to 8 statically-indexed VGPRs per lane (K <= 512);
// each of the k selection rounds does an unrolled per-lane min, a butterfly
// wave argmin (smallest distance, ties to the smallest index -- matching
// deterministic selection), and invalidates the winner with an unrolled
// compare-select.  No LDS, no scratch, no (B,N,K) distance tensor.
//
// Output (B, 4, k, N): channel 0 = corr at the selected candidate,
// channels 1..3 = candidate xyz - coords (k-major layout so the following
// conv2d/GN/pool pipeline pools over dim 2, same as the SetConv stage and
// reusable by the fused GN+act+maxpool kernel).  idx (B, N, k) is saved for
// backward; selections within a row are unique so backward is a plain
// scatter (no atomics), gradient flows to corr only (coords is detached by
// the caller every GRU iteration, reference RAFTSceneFlow.py:41).
#include <hip/hip_runtime.h>
#include "common.h"

#define MAXC 8  // max candidates per lane: K <= 8 * 64 = 512

__global__ __launch_bounds__(256) void knn_corr_fwd_kernel(
    const float *__restrict__ corr,    // (B, N, K)
    const float *__restrict__ xyz,     // (B, N, K, 3)
    const float *__restrict__ coords,  // (B, N, 3)
    float *__restrict__ out,           // (B, 4, k, N)
    int *__restrict__ out_idx,         // (B, N, k)
    int B, int N, int K, int k) {
  const long p = (long)blockIdx.x * 4 + wave_id();
  if (p >= (long)B * N) return;
  const int b = (int)(p / N);
  const int n = (int)(p % N);
  const int lane = lane_id();

  const float cx = coords[p * 3 + 0];
  const float cy = coords[p * 3 + 1];
  const float cz = coords[p * 3 + 2];
  const float *cand_xyz = xyz + p * K * 3;
  const float *cand_corr = corr + p * K;

  float d[MAXC];
#pragma unroll
  for (int t = 0; t < MAXC; ++t) {
    const int j = lane + t * WAVE;
    if (j < K) {
      const float dx = cand_xyz[j * 3 + 0] - cx;
      const float dy = cand_xyz[j * 3 + 1] - cy;
      const float dz = cand_xyz[j * 3 + 2] - cz;
      d[t] = dx * dx + dy * dy + dz * dz;
    } else {
      d[t] = INFINITY;
    }
  }

  float *dst = out + ((long)b * 4) * k * N + n;
  const long ch_stride = (long)k * N;
  int *idst = out_idx + p * k;

  for (int t = 0; t < k; ++t) {
    // per-lane min over its candidates
    float best = INFINITY;
    int bslot = 0;
#pragma unroll
    for (int s = 0; s < MAXC; ++s)
      if (d[s] < best) {
        best = d[s];
        bslot = s;
      }
    int bidx = lane + bslot * WAVE;
    if (best == INFINITY) bidx = 0x7fffffff;
    wave_argmin(best, bidx);
    // winner lane emits and invalidates its slot
    if (bidx != 0x7fffffff && (bidx % WAVE) == lane) {
      const int j = bidx;
      dst[(long)t * N] = cand_corr[j];
      dst[(long)t * N + ch_stride] = cand_xyz[j * 3 + 0] - cx;
      dst[(long)t * N + 2 * ch_stride] = cand_xyz[j * 3 + 1] - cy;
      dst[(long)t * N + 3 * ch_stride] = cand_xyz[j * 3 + 2] - cz;
      idst[t] = j;
      const int slot = bidx / WAVE;
#pragma unroll
      for (int s = 0; s < MAXC; ++s)
        if (s == slot) d[s] = INFINITY;
    }
  }
}

__global__ void knn_corr_bwd_kernel(
    const float *__restrict__ gout,  // (B, 4, k, N)
    const int *__restrict__ idx,     // (B, N, k)
    float *__restrict__ gcorr,       // (B, N, K) pre-zeroed
    int B, int N, int K, int k) {
  const long gid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * k * N;
  if (gid >= total) return;
  const int n = (int)(gid % N);
  const int t = (int)((gid / N) % k);
  const int b = (int)(gid / ((long)N * k));
  const long p = (long)b * N + n;
  const int j = idx[p * k + t];
  // channel 0 of gout feeds corr; rel-xyz channels carry no gradient
  gcorr[p * K + j] = gout[((long)b * 4 * k + t) * N + n];
}

void launch_knn_corr_fwd(const float *corr, const float *xyz,
                         const float *coords, float *out, int *out_idx, int B,
                         int N, int K, int k, hipStream_t stream) {
  const long pts = (long)B * N;
  hipLaunchKernelGGL(knn_corr_fwd_kernel, dim3((pts + 3) / 4), dim3(256), 0,
                     stream, corr, xyz, coords, out, out_idx, B, N, K, k);
}

void launch_knn_corr_bwd(const float *gout, const int *idx, float *gcorr,
                         int B, int N, int K, int k, hipStream_t stream) {
  const long total = (long)B * N * k;
  const int threads = 256;
  hipLaunchKernelGGL(knn_corr_bwd_kernel, dim3((total + threads - 1) / threads),
                     dim3(threads), 0, stream, gout, idx, gcorr, B, N, K, k);
}
