// K6: SetConv stage 1 (edge conv -> GroupNorm -> act -> max-pool-K),
// restructured for CDNA4 instead of translated.
//
// Reference semantics (model/flot/gconv.py:64-75): for every point n and
// each of its K neighbours nb = idx[n, j],
//   edge[c, j, n] = concat(feats[nb] - feats[n], xyz[nb] - xyz[n])[c]
//   x1[m, j, n]   = sum_c W[m, c] * edge[c, j, n]          (fc1, 1x1 conv)
//   y[m, n]       = max_j act(GN(x1)[m, j, n])
//
// The round-1 implementation materialised edge (B, C+3, K, N) and x1
// (B, M, K, N) (~100 MB each at the flagship shape) and ran fc1 as a GEMM
// over K*N edges.  But fc1 is LINEAR in the edge vector, and the edge
// vector is a DIFFERENCE of per-point vectors, so with g = [feats; xyz]
// (per point) and Wg = W @ g:
//   x1[m, j, n] = Wg[m, idx[n, j]] - Wg[m, n]
// i.e. the entire pre-GN edge tensor is a GATHER-DIFFERENCE of a
// (B, N, M) tensor that is 32x smaller (3 MB at the flagship shape; it
// lives in one XCD's L2).  The GEMM shrinks 32x (over N points, not K*N
// edges) and runs through the MFMA pw GEMM path; the kernels here fuse the
// gather, GroupNorm statistics/normalisation, activation and K-max-pool
// directly on Wg -- the (B, *, K, N) tensors never exist, forward or
// backward.
//
// Layout: Wg is stored point-major, wg[b, n, m] ("WgT"), so a neighbour
// gather reads one contiguous M-vector and threads of the same point cover
// adjacent channels (coalesced).  Outputs y/argmax are point-major too;
// the caller transposes the 3 MB pooled result with the LDS-tiled
// transpose kernel (trivial next to the 100 MB it replaces).
//
// Backward is DETERMINISTIC (atomic-free on the data path): the gradient
// w.r.t. WgT at point p is
//   dWg[m, p] = sum_{edges e=(j,n): idx[n,j]=p} dx1[m, j, n]   (incoming)
//             - sum_j dx1[m, j, p]                             (centre)
// where dx1 is the standard GroupNorm+act+maxpool backward element,
// recomputed on the fly from WgT / argmax / saved stats (exact same
// formulas as gnmp_bwd_* in group_norm.hip).  The incoming sum walks the
// same inverse-adjacency CSR (order/offsets) the round-1 CSR backward
// used; every (p, m) output is written exactly once.
//
// GN statistics and their workspaces follow group_norm.hip exactly
// (multi-block partial sums -> 2 atomics/block/row into a persistent
// self-cleaning fp32 workspace -> shared finalize kernel), so numerics
// match the unfused gnmp path bit-for-bit in fp32.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

#define EG_THREADS 256

// shared finalize pass (mean/rstd from the summed workspace, re-zeroing
// it) lives in group_norm.hip
void launch_gn_finalize(float *, float *, float *, long, int, float,
                        hipStream_t);

template <typename T>
DEV_INLINE float ldg(const T *p) {
  return (float)*p;
}
template <typename T>
DEV_INLINE void stg(T *p, float v) {
  *p = (T)v;
}

// ---------------------------------------------------------------- forward

// pass 1: partial sum/sumsq per (b, group) over all (c, j, n) gather-diff
// elements.  Thread (p_l, c) covers channel c of one point per iteration;
// LDS bins bound the global atomics to 2 per (block, group).
template <typename T>
__global__ __launch_bounds__(EG_THREADS) void egnmp_fwd_reduce_kernel(
    const T *__restrict__ wg,      // (B, N, M)
    const int *__restrict__ idx,   // (B, N, K)
    float *__restrict__ ws,        // (B*G, 2) zeroed
    long N, int K, int M, int G) {
  const int b = blockIdx.z;
  const int ppb = EG_THREADS / M;          // points per block iteration
  const int p_l = (int)threadIdx.x / M;    // local point slot
  const int c = (int)threadIdx.x % M;
  const bool active = p_l < ppb;
  const int Cg = M / G;
  const int g = c / Cg;

  __shared__ float bins[8 * 2];  // G <= 8
  if (threadIdx.x < (unsigned)(G * 2)) bins[threadIdx.x] = 0.f;
  __syncthreads();

  const T *wgb = wg + (long)b * N * M;
  const int *idxb = idx + (long)b * N * K;
  float s = 0.f, ss = 0.f;
  if (active) {
    for (long n = (long)blockIdx.x * ppb + p_l; n < N;
         n += (long)gridDim.x * ppb) {
      const float center = ldg(wgb + n * M + c);
      const int *row = idxb + n * K;
      for (int j = 0; j < K; ++j) {
        const float v = ldg(wgb + (long)row[j] * M + c) - center;
        s += v;
        ss += v * v;
      }
    }
    atomicAdd(&bins[g * 2 + 0], s);
    atomicAdd(&bins[g * 2 + 1], ss);
  }
  __syncthreads();
  if (threadIdx.x < (unsigned)(G * 2))
    atomicAdd(&ws[(long)(b * G) * 2 + threadIdx.x],
              bins[threadIdx.x]);
}

// pass 2 (finalize) is shared with group_norm.hip: launch_gn_finalize.

// pass 3: normalized+act gather-diff values, max over j -> pooled yT
// (B, N, M) + u8 argmax
template <typename T, int ACT>
__global__ __launch_bounds__(EG_THREADS) void egnmp_fwd_apply_kernel(
    const T *__restrict__ wg, const int *__restrict__ idx,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    T *__restrict__ y,            // (B, N, M)
    unsigned char *__restrict__ am,  // (B, N, M)
    long N, int K, int M, int G, float slope,
    const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int ppb = EG_THREADS / M;
  const int p_l = (int)threadIdx.x / M;
  const int c = (int)threadIdx.x % M;
  if (p_l >= ppb) return;
  const int Cg = M / G;
  const int row = b * G + c / Cg;
  const float m = mean[row];
  const float r = rstd[row];
  const float ga = gamma[c], be = beta[c];
  if (ACT == 2) slope = *slope_ptr;

  const T *wgb = wg + (long)b * N * M;
  const int *idxb = idx + (long)b * N * K;
  for (long n = (long)blockIdx.x * ppb + p_l; n < N;
       n += (long)gridDim.x * ppb) {
    const float center = ldg(wgb + n * M + c);
    const int *irow = idxb + n * K;
    float best = -INFINITY;
    int bk = 0;
    for (int j = 0; j < K; ++j) {
      float v = (ldg(wgb + (long)irow[j] * M + c) - center - m) * r * ga + be;
      if (ACT >= 1) v = v > 0.f ? v : v * slope;
      if (v > best) {
        best = v;
        bk = j;
      }
    }
    stg(y + ((long)b * N + n) * M + c, best);
    am[((long)b * N + n) * M + c] = (unsigned char)bk;
  }
}

// ---------------------------------------------------------------- backward

// pass 1: row sums {sum dxhat, sum dxhat*xhat} and channel sums
// {sum dy_act, sum dy_act*xhat} over the POOLED domain (only the argmax
// element of each (n, c) carries dy).  LDS bins bound global atomics.
template <typename T, int ACT>
__global__ __launch_bounds__(EG_THREADS) void egnmp_bwd_reduce_kernel(
    const T *__restrict__ dy,     // (B, N, M) pooled grad (point-major)
    const T *__restrict__ wg, const int *__restrict__ idx,
    const unsigned char *__restrict__ am, const float *__restrict__ mean,
    const float *__restrict__ rstd, const float *__restrict__ gamma,
    const float *__restrict__ beta, float *__restrict__ row_ws,
    float *__restrict__ chan_ws, float *__restrict__ slope_ws, long N, int K,
    int M, int G, float slope, const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int ppb = EG_THREADS / M;
  const int p_l = (int)threadIdx.x / M;
  const int c = (int)threadIdx.x % M;
  const bool active = p_l < ppb;
  const int Cg = M / G;
  const int g = c / Cg;
  const float m = mean[b * G + g];
  const float r = rstd[b * G + g];
  const float ga = gamma[c], be = beta[c];
  if (ACT == 2) slope = *slope_ptr;

  extern __shared__ float sbins[];  // [G*2 rows | M*2 chans | 1 slope]
  for (unsigned i = threadIdx.x; i < (unsigned)(G * 2 + M * 2 + 1);
       i += EG_THREADS)
    sbins[i] = 0.f;
  __syncthreads();

  const T *wgb = wg + (long)b * N * M;
  const int *idxb = idx + (long)b * N * K;
  float sum_dx = 0.f, sum_dxx = 0.f, c_dg = 0.f, c_db = 0.f, d_sl = 0.f;
  if (active) {
    for (long n = (long)blockIdx.x * ppb + p_l; n < N;
         n += (long)gridDim.x * ppb) {
      const long pi = ((long)b * N + n) * M + c;
      const int k = am[pi];
      const int nb = idxb[n * K + k];
      const float v = ldg(wgb + (long)nb * M + c) - ldg(wgb + n * M + c);
      const float xhat = (v - m) * r;
      float gv = ldg(dy + pi);
      if (ACT >= 1) {
        const float pre = xhat * ga + be;
        if (ACT == 2 && pre <= 0.f) d_sl += gv * pre;
        gv = pre > 0.f ? gv : gv * slope;
      }
      c_db += gv;
      c_dg += gv * xhat;
      const float dxhat = gv * ga;
      sum_dx += dxhat;
      sum_dxx += dxhat * xhat;
    }
    atomicAdd(&sbins[g * 2 + 0], sum_dx);
    atomicAdd(&sbins[g * 2 + 1], sum_dxx);
    atomicAdd(&sbins[G * 2 + c * 2 + 0], c_db);
    atomicAdd(&sbins[G * 2 + c * 2 + 1], c_dg);
    if (ACT == 2) atomicAdd(&sbins[G * 2 + M * 2], d_sl);
  }
  __syncthreads();
  for (unsigned i = threadIdx.x; i < (unsigned)(G * 2); i += EG_THREADS)
    atomicAdd(&row_ws[(long)(b * G) * 2 + i], sbins[i]);
  for (unsigned i = threadIdx.x; i < (unsigned)(M * 2); i += EG_THREADS)
    atomicAdd(&chan_ws[i], sbins[G * 2 + i]);
  if (ACT == 2 && threadIdx.x == 0) atomicAdd(slope_ws, sbins[G * 2 + M * 2]);
}

// pass 2: dWgT (B, N, M), deterministic.  dx1 elements are recomputed on
// the fly with the exact gnmp_bwd_apply formula; the incoming sum walks
// the inverse-adjacency CSR (edge id = j*N + n, sorted by target).
template <typename T, int ACT>
__global__ __launch_bounds__(EG_THREADS) void egnmp_bwd_apply_kernel(
    const T *__restrict__ dy, const T *__restrict__ wg,
    const int *__restrict__ idx, const unsigned char *__restrict__ am,
    const int *__restrict__ order,    // (B, K*N) edge ids sorted by target
    const int *__restrict__ offsets,  // (B, N+1)
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    const float *__restrict__ row_ws, T *__restrict__ dwg, long N, int K,
    int M, int G, long row_len, float slope,
    const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int ppb = EG_THREADS / M;
  const int p_l = (int)threadIdx.x / M;
  const int c = (int)threadIdx.x % M;
  if (p_l >= ppb) return;
  const int Cg = M / G;
  const int row = b * G + c / Cg;
  const float m = mean[row];
  const float r = rstd[row];
  const float ga = gamma[c], be = beta[c];
  if (ACT == 2) slope = *slope_ptr;
  const float inv_n = 1.0f / (float)row_len;
  const float s1 = row_ws[row * 2 + 0];
  const float s2 = row_ws[row * 2 + 1];

  const T *wgb = wg + (long)b * N * M;
  const T *dyb = dy + (long)b * N * M;
  const unsigned char *amb = am + (long)b * N * M;
  const int *idxb = idx + (long)b * N * K;
  const int *ordb = order + (long)b * N * K;
  const int *offb = offsets + (long)b * (N + 1);

  for (long p = (long)blockIdx.x * ppb + p_l; p < N;
       p += (long)gridDim.x * ppb) {
    const float wg_p = ldg(wgb + p * M + c);
    // centre term: sum_j dx1[c, j, p]
    float acc = 0.f;
    {
      const int ksel = amb[p * M + c];
      float g0 = ldg(dyb + p * M + c);
      const int *irow = idxb + p * K;
      for (int j = 0; j < K; ++j) {
        const float v = ldg(wgb + (long)irow[j] * M + c) - wg_p;
        const float xhat = (v - m) * r;
        float dxhat = 0.f;
        if (j == ksel) {
          float gs = g0;
          if (ACT >= 1) {
            const float pre = xhat * ga + be;
            gs = pre > 0.f ? gs : gs * slope;
          }
          dxhat = gs * ga;
        }
        acc -= (dxhat - (s1 + xhat * s2) * inv_n) * r;
      }
    }
    // incoming term: edges whose neighbour is p
    const int lo = offb[p], hi = offb[p + 1];
    for (int t = lo; t < hi; ++t) {
      const int e = ordb[t];
      const int j = e / (int)N;
      const long n = e % (int)N;
      const float v = wg_p - ldg(wgb + n * M + c);
      const float xhat = (v - m) * r;
      float dxhat = 0.f;
      if (j == (int)amb[n * M + c]) {
        float gs = ldg(dyb + n * M + c);
        if (ACT >= 1) {
          const float pre = xhat * ga + be;
          gs = pre > 0.f ? gs : gs * slope;
        }
        dxhat = gs * ga;
      }
      acc += (dxhat - (s1 + xhat * s2) * inv_n) * r;
    }
    stg(dwg + ((long)b * N + p) * M + c, acc);
  }
}

// --------------------------------------------------------------- launchers

static int eg_chunks(long N, int ppb, int B, long cap_atomics) {
  // enough blocks to fill the chip, but bounded so the per-block workspace
  // atomics stay cheap (cap_atomics blocks hit each row address)
  long want = 2048 / (B > 0 ? B : 1);
  if (want > cap_atomics) want = cap_atomics;
  long blocks = (N + ppb - 1) / ppb;
  if (want > blocks) want = blocks;
  if (want < 1) want = 1;
  return (int)want;
}

template <typename T>
void egnmp_fwd_impl(const T *wg, const int *idx, float *ws, float *mean,
                    float *rstd, const float *gamma, const float *beta,
                    T *y, unsigned char *am, int B, long N, int K, int M,
                    int G, float eps, int act, float slope,
                    const float *slope_ptr, hipStream_t stream) {
  const int ppb = EG_THREADS / M;
  const dim3 rgrid(eg_chunks(N, ppb, B, 64), 1, B);
  hipLaunchKernelGGL(egnmp_fwd_reduce_kernel<T>, rgrid, dim3(EG_THREADS), 0,
                     stream, wg, idx, ws, N, K, M, G);
  launch_gn_finalize(ws, mean, rstd, (long)(M / G) * K * N, B * G, eps,
                     stream);
  const dim3 agrid(eg_chunks(N, ppb, B, 1 << 20), 1, B);
#define EG_FWD(A)                                                             \
  hipLaunchKernelGGL((egnmp_fwd_apply_kernel<T, A>), agrid, dim3(EG_THREADS), \
                     0, stream, wg, idx, mean, rstd, gamma, beta, y, am, N,   \
                     K, M, G, slope, slope_ptr)
  if (act == 2) EG_FWD(2);
  else if (act == 1) EG_FWD(1);
  else EG_FWD(0);
#undef EG_FWD
}

template <typename T>
void egnmp_bwd_impl(const T *dy, const T *wg, const int *idx,
                    const unsigned char *am, const int *order,
                    const int *offsets, const float *mean, const float *rstd,
                    const float *gamma, const float *beta, float *row_ws,
                    float *chan_ws, float *slope_ws, T *dwg, int B, long N,
                    int K, int M, int G, int act, float slope,
                    const float *slope_ptr, hipStream_t stream) {
  const int ppb = EG_THREADS / M;
  const dim3 rgrid(eg_chunks(N, ppb, B, 64), 1, B);
  const dim3 agrid(eg_chunks(N, ppb, B, 1 << 20), 1, B);
  const size_t shmem = (size_t)(G * 2 + M * 2 + 1) * sizeof(float);
  const long row_len = (long)(M / G) * K * N;
#define EG_BWD(A)                                                              \
  do {                                                                         \
    hipLaunchKernelGGL((egnmp_bwd_reduce_kernel<T, A>), rgrid,                 \
                       dim3(EG_THREADS), shmem, stream, dy, wg, idx, am, mean, \
                       rstd, gamma, beta, row_ws, chan_ws, slope_ws, N, K, M,  \
                       G, slope, slope_ptr);                                   \
    hipLaunchKernelGGL((egnmp_bwd_apply_kernel<T, A>), agrid,                  \
                       dim3(EG_THREADS), 0, stream, dy, wg, idx, am, order,    \
                       offsets, mean, rstd, gamma, beta, row_ws, dwg, N, K, M, \
                       G, row_len, slope, slope_ptr);                          \
  } while (0)
  if (act == 2) EG_BWD(2);
  else if (act == 1) EG_BWD(1);
  else EG_BWD(0);
#undef EG_BWD
}

void launch_egnmp_fwd(const void *wg, const int *idx, float *ws, float *mean,
                      float *rstd, const float *gamma, const float *beta,
                      void *y, unsigned char *am, int B, long N, int K, int M,
                      int G, float eps, int act, float slope,
                      const float *slope_ptr, bool bf16, hipStream_t stream) {
  if (bf16)
    egnmp_fwd_impl<__hip_bfloat16>((const __hip_bfloat16 *)wg, idx, ws, mean,
                                   rstd, gamma, beta, (__hip_bfloat16 *)y, am,
                                   B, N, K, M, G, eps, act, slope, slope_ptr,
                                   stream);
  else
    egnmp_fwd_impl<float>((const float *)wg, idx, ws, mean, rstd, gamma, beta,
                          (float *)y, am, B, N, K, M, G, eps, act, slope,
                          slope_ptr, stream);
}

void launch_egnmp_bwd(const void *dy, const void *wg, const int *idx,
                      const unsigned char *am, const int *order,
                      const int *offsets, const float *mean, const float *rstd,
                      const float *gamma, const float *beta, float *row_ws,
                      float *chan_ws, float *slope_ws, void *dwg, int B,
                      long N, int K, int M, int G, int act, float slope,
                      const float *slope_ptr, bool bf16, hipStream_t stream) {
  if (bf16)
    egnmp_bwd_impl<__hip_bfloat16>(
        (const __hip_bfloat16 *)dy, (const __hip_bfloat16 *)wg, idx, am, order,
        offsets, mean, rstd, gamma, beta, row_ws, chan_ws, slope_ws,
        (__hip_bfloat16 *)dwg, B, N, K, M, G, act, slope, slope_ptr, stream);
  else
    egnmp_bwd_impl<float>((const float *)dy, (const float *)wg, idx, am, order,
                          offsets, mean, rstd, gamma, beta, row_ws, chan_ws,
                          slope_ws, (float *)dwg, B, N, K, M, G, act, slope,
                          slope_ptr, stream);
}
