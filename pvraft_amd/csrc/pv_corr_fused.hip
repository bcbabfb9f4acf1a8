// K4+K5 fused: the per-GRU-iteration point-voxel correlation lookup
// (reference model/corr.py:44-93) as ONE kernel per direction.  Both the
// voxel pyramid and the kNN branch read the same (corr, xyz, coords)
// candidate field; fusing them loads each point's K candidates into
// registers once per iteration instead of twice (+ once per pyramid level).
//
// Geometry: one wave per point (block = 256 = 4 points), lanes own
// ceil(K/64) <= 8 candidates in statically-indexed VGPRs.
//
// forward outputs:
//   voxel (B, L*27, N): per-level 3^3 mean of corr over quantised offsets
//     (27 statically-indexed accumulators per lane, wave butterfly reduce);
//   knn (B, 4, k, N): [corr; rel-xyz] of the k nearest candidates
//     (wave argmin rounds, ties to the smallest index);
//   knn_idx (B, N, k) for backward.
// backward (d corr only; quantisation/selection indices are constants to
// autograd, reference corr.py:52-62 and coords detached per iteration):
//   d corr[j] = sum_l [valid_l(j)] g_vox[l*27+cell_l(j)] / cnt_l(cell)
//             + [j selected at slot t] g_knn[0, t].
#include <hip/hip_runtime.h>
#include "common.h"

#define R 3
#define CELLS 27
#define MAXL 4
#define MAXC 8   // K <= 512
#define MAXKN 64 // selected neighbours <= 64

__global__ __launch_bounds__(256) void pv_corr_fused_fwd_kernel(
    const float *__restrict__ corr,    // (B, N, K)
    const float *__restrict__ xyz,     // (B, N, K, 3)
    const float *__restrict__ coords,  // (B, N, 3)
    float *__restrict__ vox,           // (B, L*27, N)
    float *__restrict__ knn,           // (B, 4, k, N)
    int *__restrict__ knn_idx,         // (B, N, k)
    int B, int N, int K, int L, int k, float base_scale) {
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

  // ---- voxel pyramid (candidates re-read per phase: L1-resident;
  // caching them in VGPRs or keeping the kNN distances live across this
  // loop measured SLOWER via occupancy)
  for (int l = 0; l < L; ++l) {
    const float inv_r = 1.0f / (base_scale * (float)(1 << l));
    float s[CELLS], c[CELLS];
#pragma unroll
    for (int q = 0; q < CELLS; ++q) {
      s[q] = 0.f;
      c[q] = 0.f;
    }
    for (int j = lane; j < K; j += WAVE) {
      {
        const float dx = rintf((cand_xyz[j * 3 + 0] - cx) * inv_r);
        const float dy = rintf((cand_xyz[j * 3 + 1] - cy) * inv_r);
        const float dz = rintf((cand_xyz[j * 3 + 2] - cz) * inv_r);
        const bool valid = fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) &&
                           fabsf(dz) <= (R / 2);
        const int cell = ((int)dx + R / 2) * (R * R) + ((int)dy + R / 2) * R +
                         ((int)dz + R / 2);
        const float v = cand_corr[j];
#pragma unroll
        for (int q = 0; q < CELLS; ++q) {
          const bool hit = valid && (cell == q);
          s[q] += hit ? v : 0.f;
          c[q] += hit ? 1.f : 0.f;
        }
      }
    }
#pragma unroll
    for (int q = 0; q < CELLS; ++q) {
      s[q] = wave_sum(s[q]);
      c[q] = wave_sum(c[q]);
    }
    if (lane < CELLS) {
      float sv = 0.f, cc = 0.f;
#pragma unroll
      for (int q = 0; q < CELLS; ++q)
        if (lane == q) {
          sv = s[q];
          cc = c[q];
        }
      vox[((long)b * L * CELLS + l * CELLS + lane) * N + n] = sv / fmaxf(cc, 1.f);
    }
  }

  // ---- kNN branch (argmin rounds on register distances, computed here so
  // they are not live during the voxel phase)
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
  float *dst = knn + ((long)b * 4) * k * N + n;
  const long ch_stride = (long)k * N;
  int *idst = knn_idx + p * k;
  for (int t = 0; t < k; ++t) {
    float best = INFINITY;
    int bslot = 0;
#pragma unroll
    for (int s_ = 0; s_ < MAXC; ++s_)
      if (d[s_] < best) {
        best = d[s_];
        bslot = s_;
      }
    int bidx = lane + bslot * WAVE;
    if (best == INFINITY) bidx = 0x7fffffff;
    wave_argmin(best, bidx);
    if (bidx != 0x7fffffff && (bidx % WAVE) == lane) {
      const int s_ = bidx / WAVE;
      const int j = bidx;
      dst[(long)t * N] = cand_corr[j];
      dst[(long)t * N + ch_stride] = cand_xyz[j * 3 + 0] - cx;
      dst[(long)t * N + 2 * ch_stride] = cand_xyz[j * 3 + 1] - cy;
      dst[(long)t * N + 3 * ch_stride] = cand_xyz[j * 3 + 2] - cz;
      idst[t] = bidx;
#pragma unroll
      for (int ss = 0; ss < MAXC; ++ss)
        if (ss == s_) d[ss] = INFINITY;
    }
  }
}

__global__ __launch_bounds__(256) void pv_corr_fused_bwd_kernel(
    const float *__restrict__ g_vox,   // (B, L*27, N)
    const float *__restrict__ g_knn,   // (B, 4, k, N)
    const float *__restrict__ xyz,     // (B, N, K, 3)
    const float *__restrict__ coords,  // (B, N, 3)
    const int *__restrict__ knn_idx,   // (B, N, k)
    float *__restrict__ gcorr,         // (B, N, K)
    int B, int N, int K, int L, int k, float base_scale) {
  __shared__ int s_sel[4][MAXKN];
  __shared__ float s_g0[4][MAXKN];

  const long p = (long)blockIdx.x * 4 + wave_id();
  const bool active = p < (long)B * N;
  const int b = active ? (int)(p / N) : 0;
  const int n = active ? (int)(p % N) : 0;
  const int lane = lane_id();
  const int w = wave_id();

  // knn selections + channel-0 grads into LDS (per wave); every thread
  // reaches the barrier (inactive tail waves included)
  if (active && lane < k) {
    s_sel[w][lane] = knn_idx[p * k + lane];
    s_g0[w][lane] = g_knn[(((long)b * 4) * k + lane) * N + n];
  }
  __syncthreads();
  if (!active) return;

  const float cx = coords[p * 3 + 0];
  const float cy = coords[p * 3 + 1];
  const float cz = coords[p * 3 + 2];
  const float *cand_xyz = xyz + p * K * 3;

  // per-level counts + this lane's cell's (count, g_vox)
  float cnt_mine[MAXL], g_mine[MAXL];
#pragma unroll
  for (int l = 0; l < MAXL; ++l) {
    cnt_mine[l] = 0.f;
    g_mine[l] = 0.f;
  }
  for (int l = 0; l < L; ++l) {
    const float inv_r = 1.0f / (base_scale * (float)(1 << l));
    float c[CELLS];
#pragma unroll
    for (int q = 0; q < CELLS; ++q) c[q] = 0.f;
    for (int j = lane; j < K; j += WAVE) {
      const float dx = rintf((cand_xyz[j * 3 + 0] - cx) * inv_r);
      const float dy = rintf((cand_xyz[j * 3 + 1] - cy) * inv_r);
      const float dz = rintf((cand_xyz[j * 3 + 2] - cz) * inv_r);
      const bool valid = fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) &&
                         fabsf(dz) <= (R / 2);
      const int cell = ((int)dx + R / 2) * (R * R) + ((int)dy + R / 2) * R +
                       ((int)dz + R / 2);
#pragma unroll
      for (int q = 0; q < CELLS; ++q) c[q] += (valid && cell == q) ? 1.f : 0.f;
    }
#pragma unroll
    for (int q = 0; q < CELLS; ++q) c[q] = wave_sum(c[q]);
    float cc = 0.f;
#pragma unroll
    for (int q = 0; q < CELLS; ++q)
      if (lane == q) cc = c[q];
#pragma unroll
    for (int ll = 0; ll < MAXL; ++ll)
      if (ll == l) {
        cnt_mine[ll] = cc;
        g_mine[ll] = (lane < CELLS)
                         ? g_vox[((long)b * L * CELLS + l * CELLS + lane) * N + n]
                         : 0.f;
      }
  }

  for (int j = lane; j < K; j += WAVE) {
    const float ox = cand_xyz[j * 3 + 0] - cx;
    const float oy = cand_xyz[j * 3 + 1] - cy;
    const float oz = cand_xyz[j * 3 + 2] - cz;
    float g = 0.f;
#pragma unroll
    for (int l = 0; l < MAXL; ++l) {
      if (l < L) {
        const float inv_r = 1.0f / (base_scale * (float)(1 << l));
        const float dx = rintf(ox * inv_r);
        const float dy = rintf(oy * inv_r);
        const float dz = rintf(oz * inv_r);
        const bool valid = fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) &&
                           fabsf(dz) <= (R / 2);
        const int cell = valid ? ((int)dx + R / 2) * (R * R) +
                                     ((int)dy + R / 2) * R + ((int)dz + R / 2)
                               : 0;
        const float gq = __shfl(g_mine[l], cell, WAVE);
        const float cq = __shfl(cnt_mine[l], cell, WAVE);
        g += valid ? gq / fmaxf(cq, 1.f) : 0.f;
      }
    }
    // knn contribution: selected slots are unique within the row
    for (int t2 = 0; t2 < k; ++t2) g += (s_sel[w][t2] == j) ? s_g0[w][t2] : 0.f;
    gcorr[p * K + j] = g;
  }
}

void launch_pv_corr_fused_fwd(const float *corr, const float *xyz,
                              const float *coords, float *vox, float *knn,
                              int *knn_idx, int B, int N, int K, int L, int k,
                              float base_scale, hipStream_t stream) {
  const long pts = (long)B * N;
  hipLaunchKernelGGL(pv_corr_fused_fwd_kernel, dim3((pts + 3) / 4), dim3(256),
                     0, stream, corr, xyz, coords, vox, knn, knn_idx, B, N, K,
                     L, k, base_scale);
}

void launch_pv_corr_fused_bwd(const float *g_vox, const float *g_knn,
                              const float *xyz, const float *coords,
                              const int *knn_idx, float *gcorr, int B, int N,
                              int K, int L, int k, float base_scale,
                              hipStream_t stream) {
  const long pts = (long)B * N;
  hipLaunchKernelGGL(pv_corr_fused_bwd_kernel, dim3((pts + 3) / 4), dim3(256),
                     0, stream, g_vox, g_knn, xyz, coords, knn_idx, gcorr, B,
                     N, K, L, k, base_scale);
}
