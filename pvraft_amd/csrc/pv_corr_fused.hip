// K4+K5 fused: the per-GRU-iteration point-voxel correlation lookup
// (reference model/corr.py:44-93) as ONE kernel per direction.  Both the
// voxel pyramid and the kNN branch read the same (corr, xyz, coords)
// candidate field; fusing them keeps the field's HBM traffic to one pass
// per phase and the intermediate (B,N,K) distance / (B,N,27) scatter
// tensors never exist.
//
// Geometry: one wave per point (block = 256 = 4 points).  Each lane owns
// K/64 (<= 8) candidates in REGISTERS (relative xyz + corr + distance,
// statically indexed): the round-1 version staged the whole field in LDS
// and re-read it from every phase, which made the kernel LDS-op bound
// (~9k LDS operations per point).  Register residency leaves LDS for the
// things that need cross-lane communication only: the 27-cell voxel
// accumulators and the per-level counts.
//
// forward outputs:
//   voxel (B, L*27, N): per-level 3^3 mean of corr over quantised offsets;
//   knn (B, 4, k, N): [corr; rel-xyz] of the k nearest candidates
//     (wave extraction rounds; tie order arbitrary -- all consumers are
//     order-invariant);
//   knn_idx (B, N, k) for backward.
// backward (d corr only; quantisation/selection indices are constants to
// autograd, reference corr.py:52-62 and coords detached per iteration):
//   d corr[j] = sum_l [valid_l(j)] g_vox[l*27+cell_l(j)] / cnt_l(cell)
//             + [j selected at slot t] g_knn[0, t]
// with the kNN term resolved by an LDS scatter over the candidate row
// (selected slots are unique within a row).
#include <hip/hip_runtime.h>
#include "common.h"

#define R 3
#define CELLS 27
#define MAXL 4
#define MAXC 8    // K <= 512 candidate slots per lane
#define MAXKN 64  // selected neighbours <= 64
#define MAXK 512  // candidate field width (backward LDS scatter row)

__global__ __launch_bounds__(256) void pv_corr_fused_fwd_kernel(
    const float *__restrict__ corr,    // (B, N, K)
    const float *__restrict__ xyz,     // (B, N, K, 3)
    const float *__restrict__ coords,  // (B, N, 3)
    float *__restrict__ vox,           // (B, L*27, N)
    float *__restrict__ knn,           // (B, 4, k, N)
    int *__restrict__ knn_idx,         // (B, N, k)
    int B, int N, int K, int L, int k, float base_scale) {
  __shared__ float s_sum[4][CELLS];
  __shared__ float s_cnt[4][CELLS];

  const long p = (long)blockIdx.x * 4 + wave_id();
  if (p >= (long)B * N) return;
  const int b = (int)(p / N);
  const int n = (int)(p % N);
  const int lane = lane_id();
  const int w = wave_id();

  const float cx = coords[p * 3 + 0];
  const float cy = coords[p * 3 + 1];
  const float cz = coords[p * 3 + 2];
  const float *cand_xyz = xyz + p * K * 3;
  const float *cand_corr = corr + p * K;

  // this lane's candidates, register-resident (statically indexed)
  float rx[MAXC], ry[MAXC], rz[MAXC], rc[MAXC], d[MAXC];
#pragma unroll
  for (int t = 0; t < MAXC; ++t) {
    const int j = lane + t * WAVE;
    if (j < K) {
      rx[t] = cand_xyz[j * 3 + 0] - cx;
      ry[t] = cand_xyz[j * 3 + 1] - cy;
      rz[t] = cand_xyz[j * 3 + 2] - cz;
      rc[t] = cand_corr[j];
      d[t] = rx[t] * rx[t] + ry[t] * ry[t] + rz[t] * rz[t];
    } else {
      rx[t] = ry[t] = rz[t] = rc[t] = 0.f;
      d[t] = INFINITY;
    }
  }

  // ---- voxel pyramid: per-wave LDS histogram per level
  for (int l = 0; l < L; ++l) {
    const float inv_r = 1.0f / (base_scale * (float)(1 << l));
    if (lane < CELLS) {
      s_sum[w][lane] = 0.f;
      s_cnt[w][lane] = 0.f;
    }
    __threadfence_block();
#pragma unroll
    for (int t = 0; t < MAXC; ++t) {
      if (lane + t * WAVE < K) {
        const float dx = rintf(rx[t] * inv_r);
        const float dy = rintf(ry[t] * inv_r);
        const float dz = rintf(rz[t] * inv_r);
        if (fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) &&
            fabsf(dz) <= (R / 2)) {
          const int cell = ((int)dx + R / 2) * (R * R) +
                           ((int)dy + R / 2) * R + ((int)dz + R / 2);
          atomicAdd(&s_sum[w][cell], rc[t]);
          atomicAdd(&s_cnt[w][cell], 1.f);
        }
      }
    }
    __threadfence_block();
    if (lane < CELLS)
      vox[((long)b * L * CELLS + l * CELLS + lane) * N + n] =
          s_sum[w][lane] / fmaxf(s_cnt[w][lane], 1.f);
    // next level's zeroing is ordered behind these reads by instruction
    // order within the wave (LDS ops issue in order from one wave)
  }

  // ---- kNN branch: radix rank select of the k smallest distances
  // (inverted keys: larger key = smaller distance), then ballot-prefix
  // emission -- each lane owns its candidates' data in registers and
  // writes its own winners.  Slot order is arbitrary; every consumer
  // (conv + max-pool over k, index-scatter backward) is order-invariant.
  float *dst = knn + ((long)b * 4) * k * N + n;
  const long ch_stride = (long)k * N;
  int *idst = knn_idx + p * k;
  unsigned kv[MAXC];
#pragma unroll
  for (int t = 0; t < MAXC; ++t)
    kv[t] = (lane + t * WAVE) < K ? ~fkey(d[t]) : 0u;
  const unsigned kt = wave_rank_key(kv, k);
  int base = 0;
  for (int pass = 0; pass < 2; ++pass) {
#pragma unroll
    for (int t = 0; t < MAXC; ++t) {
      const bool elig = (lane + t * WAVE) < K &&
                        (pass == 0 ? kv[t] > kt : kv[t] == kt);
      const unsigned long long m = __ballot(elig);
      if (elig) {
        const int slot = base + __popcll(m & ((1ull << lane) - 1ull));
        if (slot < k) {
          dst[(long)slot * N] = rc[t];
          dst[(long)slot * N + ch_stride] = rx[t];
          dst[(long)slot * N + 2 * ch_stride] = ry[t];
          dst[(long)slot * N + 3 * ch_stride] = rz[t];
          idst[slot] = lane + t * WAVE;
        }
      }
      base += __popcll(m);
    }
    if (base >= k) break;
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
  __shared__ float s_cnt[4][MAXL][CELLS];
  __shared__ float s_gv[4][MAXL][CELLS];
  __shared__ float s_kg[4][MAXK];  // per-candidate kNN grad (scattered)

  const long p = (long)blockIdx.x * 4 + wave_id();
  if (p >= (long)B * N) return;
  const int b = (int)(p / N);
  const int n = (int)(p % N);
  const int lane = lane_id();
  const int w = wave_id();

  const float cx = coords[p * 3 + 0];
  const float cy = coords[p * 3 + 1];
  const float cz = coords[p * 3 + 2];
  const float *cand_xyz = xyz + p * K * 3;

  // register-resident relative candidate xyz (statically indexed)
  float rx[MAXC], ry[MAXC], rz[MAXC];
#pragma unroll
  for (int t = 0; t < MAXC; ++t) {
    const int j = lane + t * WAVE;
    if (j < K) {
      rx[t] = cand_xyz[j * 3 + 0] - cx;
      ry[t] = cand_xyz[j * 3 + 1] - cy;
      rz[t] = cand_xyz[j * 3 + 2] - cz;
    } else {
      rx[t] = ry[t] = rz[t] = 1e30f;  // lands outside every cell
    }
  }

  // scatter the kNN channel-0 grads onto their candidate slots (unique)
  for (int j = lane; j < K; j += WAVE) s_kg[w][j] = 0.f;
  __threadfence_block();
  if (lane < k) {
    const int sel = knn_idx[p * k + lane];
    const float g0 = g_knn[(((long)b * 4) * k + lane) * N + n];
    s_kg[w][sel] = g0;  // slots unique within the row: plain store
  }

  // per-level cell counts (LDS histogram) + g_vox row into LDS
  for (int l = 0; l < L; ++l) {
    const float inv_r = 1.0f / (base_scale * (float)(1 << l));
    if (lane < CELLS) {
      s_cnt[w][l][lane] = 0.f;
      s_gv[w][l][lane] =
          g_vox[((long)b * L * CELLS + l * CELLS + lane) * N + n];
    }
    __threadfence_block();
#pragma unroll
    for (int t = 0; t < MAXC; ++t) {
      if (lane + t * WAVE < K) {
        const float dx = rintf(rx[t] * inv_r);
        const float dy = rintf(ry[t] * inv_r);
        const float dz = rintf(rz[t] * inv_r);
        if (fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) &&
            fabsf(dz) <= (R / 2))
          atomicAdd(&s_cnt[w][l][((int)dx + R / 2) * (R * R) +
                                 ((int)dy + R / 2) * R + ((int)dz + R / 2)],
                    1.f);
      }
    }
  }
  __threadfence_block();

#pragma unroll
  for (int t = 0; t < MAXC; ++t) {
    const int j = lane + t * WAVE;
    if (j >= K) continue;
    float g = s_kg[w][j];
#pragma unroll
    for (int l = 0; l < MAXL; ++l) {
      if (l < L) {
        const float inv_r = 1.0f / (base_scale * (float)(1 << l));
        const float dx = rintf(rx[t] * inv_r);
        const float dy = rintf(ry[t] * inv_r);
        const float dz = rintf(rz[t] * inv_r);
        if (fabsf(dx) <= (R / 2) && fabsf(dy) <= (R / 2) &&
            fabsf(dz) <= (R / 2)) {
          const int cell = ((int)dx + R / 2) * (R * R) + ((int)dy + R / 2) * R +
                           ((int)dz + R / 2);
          g += s_gv[w][l][cell] / fmaxf(s_cnt[w][l][cell], 1.f);
        }
      }
    }
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
