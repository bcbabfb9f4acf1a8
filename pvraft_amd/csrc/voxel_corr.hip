// K4: voxel correlation pyramid, forward + backward (capability of
// reference model/corr.py:47-73, which goes through torch_scatter
// scatter_add twice per level).  Single kernel per direction:
//
//   out[b, l*27 + cell, n] = mean of corr[b,n,j] over candidates j whose
//   quantised offset round((xyz[b,n,j] - coords[b,n]) / (base*2^l)) lands
//   in cell of the 3x3x3 cube (count clamped >= 1).
//
// Geometry: one wave per point (block = 256 threads = 4 points), lanes
// stride the K candidates.  Per-cell partial sums/counts live in 27
// statically-indexed VGPRs per lane (27-way unrolled compare-select -- no
// LDS atomics, no scratch), then a 6-step butterfly reduction across the
// wave; lanes 0..26 write the cells.  Quantisation indices are constants
// to autograd (reference corr.py:52-62): backward only produces
// d corr = g_out[cell] / count via the same recomputation.
#include <hip/hip_runtime.h>
#include "common.h"

#define R 3
#define CELLS 27  // R^3
#define MAXL 4    // compile-time cap on pyramid levels (model uses 3)

__global__ __launch_bounds__(256) void voxel_corr_fwd_kernel(
    const float *__restrict__ corr,    // (B, N, K)
    const float *__restrict__ xyz,     // (B, N, K, 3)
    const float *__restrict__ coords,  // (B, N, 3)
    float *__restrict__ out,           // (B, L*27, N)
    int B, int N, int K, int L, float base_scale) {
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

  for (int l = 0; l < L; ++l) {
    const float inv_r = 1.0f / (base_scale * (float)(1 << l));
    float s[CELLS];
    float c[CELLS];
#pragma unroll
    for (int q = 0; q < CELLS; ++q) {
      s[q] = 0.f;
      c[q] = 0.f;
    }
    for (int j = lane; j < K; j += WAVE) {
      const float dx = rintf((cand_xyz[j * 3 + 0] - cx) * inv_r);
      const float dy = rintf((cand_xyz[j * 3 + 1] - cy) * inv_r);
      const float dz = rintf((cand_xyz[j * 3 + 2] - cz) * inv_r);
      const bool valid =
          fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) && fabsf(dz) <= (R / 2);
      const int cell =
          ((int)dx + R / 2) * (R * R) + ((int)dy + R / 2) * R + ((int)dz + R / 2);
      const float v = cand_corr[j];
#pragma unroll
      for (int q = 0; q < CELLS; ++q) {
        const bool hit = valid && (cell == q);
        s[q] += hit ? v : 0.f;
        c[q] += hit ? 1.f : 0.f;
      }
    }
#pragma unroll
    for (int q = 0; q < CELLS; ++q) {
      s[q] = wave_sum(s[q]);
      c[q] = wave_sum(c[q]);
    }
    if (lane < CELLS) {
      float sv = 0.f, cv = 0.f;
#pragma unroll
      for (int q = 0; q < CELLS; ++q)
        if (lane == q) {
          sv = s[q];
          cv = c[q];
        }
      out[((long)b * L * CELLS + l * CELLS + lane) * N + n] = sv / fmaxf(cv, 1.f);
    }
  }
}

__global__ __launch_bounds__(256) void voxel_corr_bwd_kernel(
    const float *__restrict__ gout,    // (B, L*27, N)
    const float *__restrict__ xyz,     // (B, N, K, 3)
    const float *__restrict__ coords,  // (B, N, 3)
    float *__restrict__ gcorr,         // (B, N, K)
    int B, int N, int K, int L, float base_scale) {
  const long p = (long)blockIdx.x * 4 + wave_id();
  if (p >= (long)B * N) return;
  const int b = (int)(p / N);
  const int n = (int)(p % N);
  const int lane = lane_id();

  const float cx = coords[p * 3 + 0];
  const float cy = coords[p * 3 + 1];
  const float cz = coords[p * 3 + 2];
  const float *cand_xyz = xyz + p * K * 3;

  // per-level: this lane's cell's (count, gout) -- lane q holds cell q
  float cnt_mine[MAXL];
  float g_mine[MAXL];
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
      const bool valid =
          fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) && fabsf(dz) <= (R / 2);
      const int cell =
          ((int)dx + R / 2) * (R * R) + ((int)dy + R / 2) * R + ((int)dz + R / 2);
#pragma unroll
      for (int q = 0; q < CELLS; ++q) c[q] += (valid && cell == q) ? 1.f : 0.f;
    }
#pragma unroll
    for (int q = 0; q < CELLS; ++q) c[q] = wave_sum(c[q]);
    float cv = 0.f;
#pragma unroll
    for (int q = 0; q < CELLS; ++q)
      if (lane == q) cv = c[q];
#pragma unroll
    for (int ll = 0; ll < MAXL; ++ll)
      if (ll == l) {
        cnt_mine[ll] = cv;
        g_mine[ll] =
            (lane < CELLS)
                ? gout[((long)b * L * CELLS + l * CELLS + lane) * N + n]
                : 0.f;
      }
  }

  for (int j = lane; j < K; j += WAVE) {
    const float x = cand_xyz[j * 3 + 0] - cx;
    const float y = cand_xyz[j * 3 + 1] - cy;
    const float z = cand_xyz[j * 3 + 2] - cz;
    float g = 0.f;
#pragma unroll
    for (int l = 0; l < MAXL; ++l) {
      if (l < L) {
        const float inv_r = 1.0f / (base_scale * (float)(1 << l));
        const float dx = rintf(x * inv_r);
        const float dy = rintf(y * inv_r);
        const float dz = rintf(z * inv_r);
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
    gcorr[p * K + j] = g;
  }
}

void launch_voxel_corr_fwd(const float *corr, const float *xyz,
                           const float *coords, float *out, int B, int N,
                           int K, int L, float base_scale, hipStream_t stream) {
  const long pts = (long)B * N;
  hipLaunchKernelGGL(voxel_corr_fwd_kernel, dim3((pts + 3) / 4), dim3(256), 0,
                     stream, corr, xyz, coords, out, B, N, K, L, base_scale);
}

void launch_voxel_corr_bwd(const float *gout, const float *xyz,
                           const float *coords, float *gcorr, int B, int N,
                           int K, int L, float base_scale, hipStream_t stream) {
  const long pts = (long)B * N;
  hipLaunchKernelGGL(voxel_corr_bwd_kernel, dim3((pts + 3) / 4), dim3(256), 0,
                     stream, gout, xyz, coords, gcorr, B, N, K, L, base_scale);
}
