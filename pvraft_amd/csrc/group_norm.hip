// K9: GroupNorm (+ fused LeakyReLU) forward/backward for CDNA4.
//
// Why custom: ATen's GroupNorm forward launches ONE workgroup per
// (batch, group) row -- B=2, G=8 => 16 workgroups on a 256-CU chip, and it
// measured 30% of the PV-RAFT train step.  Here both directions use
// multi-workgroup reductions (fp32 accumulation, block partials combined
// with a handful of atomics into a small workspace) so the chip is filled
// regardless of B.
//
// Layout: x (B, C, S) contiguous (S = flattened spatial, e.g. K*N), G
// groups, group g = channels [g*Cg, (g+1)*Cg) -- a group's data is one
// contiguous block of Cg*S elements, reduced as a flat row.  Each
// workgroup owns a CONTIGUOUS chunk of one row, so a thread's channel
// changes only every ~S/256 iterations and per-channel partials flush with
// O(channels-touched) atomics, not O(elements).
// dtype: fp32 or bf16 IO (template), fp32 math everywhere.
//
// act: 0 = identity, 1 = LeakyReLU(slope) fused into the normalize pass
// (backward recomputes the pre-activation sign from xhat, gamma, beta).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

#define GN_THREADS 256

template <typename T>
DEV_INLINE float ld(const T *p) {
  return (float)*p;
}

template <typename T>
DEV_INLINE void st(T *p, float v) {
  *p = (T)v;
}

template <int ACT>
DEV_INLINE float act_slope(float slope, const float *slope_ptr) {
  return ACT == 2 ? *slope_ptr : slope;
}

DEV_INLINE float block_sum(float v) {
  v = wave_sum(v);
  __shared__ float sh[GN_THREADS / WAVE];
  if (lane_id() == 0) sh[wave_id()] = v;
  __syncthreads();
  float t = 0.f;
  if (threadIdx.x < GN_THREADS / WAVE) t = sh[threadIdx.x];
  __syncthreads();  // sh reused across calls
  return wave_sum(t);  // valid in wave 0
}

// chunk bounds for block `chunk` of `blocks_per_row` over a row of row_len
DEV_INLINE void chunk_range(long row_len, int blocks_per_row, int chunk,
                            long &lo, long &hi) {
  const long per = (row_len + blocks_per_row - 1) / blocks_per_row;
  lo = (long)chunk * per;
  hi = min(lo + per, row_len);
}

// ---------------------------------------------------------------- forward

// pass 1: partial sum/sumsq per row chunk -> ws[row] = {sum, sumsq}
template <typename T>
__global__ __launch_bounds__(GN_THREADS) void gn_fwd_reduce_kernel(
    const T *__restrict__ x, float *__restrict__ ws, long row_len, int rows,
    int blocks_per_row) {
  const int row = blockIdx.x / blocks_per_row;
  if (row >= rows) return;
  long lo, hi;
  chunk_range(row_len, blocks_per_row, blockIdx.x % blocks_per_row, lo, hi);
  const T *base = x + (long)row * row_len;
  float s = 0.f, ss = 0.f;
  for (long i = lo + threadIdx.x; i < hi; i += GN_THREADS) {
    const float v = ld(base + i);
    s += v;
    ss += v * v;
  }
  s = block_sum(s);
  ss = block_sum(ss);
  if (threadIdx.x == 0) {
    atomicAdd(&ws[row * 2 + 0], s);
    atomicAdd(&ws[row * 2 + 1], ss);
  }
}

// pass 2: rows threads -> mean/rstd
__global__ void gn_fwd_finalize_kernel(const float *__restrict__ ws,
                                       float *__restrict__ mean,
                                       float *__restrict__ rstd, long row_len,
                                       int rows, float eps) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= rows) return;
  const float m = ws[row * 2 + 0] / (float)row_len;
  const float var = ws[row * 2 + 1] / (float)row_len - m * m;
  mean[row] = m;
  rstd[row] = rsqrtf(fmaxf(var, 0.f) + eps);
}

// pass 3: y = act((x - mean) * rstd * gamma + beta)
template <typename T, int ACT>
__global__ __launch_bounds__(GN_THREADS) void gn_fwd_apply_kernel(
    const T *__restrict__ x, T *__restrict__ y, const float *__restrict__ mean,
    const float *__restrict__ rstd, const float *__restrict__ gamma,
    const float *__restrict__ beta, long S, int C, int G, long total,
    float slope, const float *__restrict__ slope_ptr) {
  const int Cg = C / G;
  if (ACT >= 1) slope = act_slope<ACT>(slope, slope_ptr);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long c = (i / S) % C;
    const long row = i / (S * Cg);  // == b * G + g
    float v = (ld(x + i) - mean[row]) * rstd[row] * gamma[c] + beta[c];
    if (ACT >= 1) v = v > 0.f ? v : v * slope;
    st(y + i, v);
  }
}

// ---------------------------------------------------------------- backward

// partial sums: per-row {sum dxhat, sum dxhat*xhat}, per-channel
// {sum dy_norm, sum dy_norm*xhat}.  Grid (spatial chunks, C, B): each
// block covers ONE channel's spatial slice, reduces its four partials
// block-wide and issues exactly four global atomics -- per-element or
// per-thread atomic flushes onto the tiny (C,2)/(rows,2) workspaces
// serialize catastrophically (measured 100-400x slower).
template <typename T, int ACT>
__global__ __launch_bounds__(GN_THREADS) void gn_bwd_reduce_kernel(
    const T *__restrict__ dy, const T *__restrict__ x,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    float *__restrict__ row_ws,   // (rows, 2)
    float *__restrict__ chan_ws,  // (C, 2)
    float *__restrict__ slope_ws, // (1,) d slope accumulator (ACT == 2)
    long S, int C, int G, float slope, const float *__restrict__ slope_ptr) {
  const int c = blockIdx.y;
  const int b = blockIdx.z;
  const int Cg = C / G;
  const int row = b * G + c / Cg;
  const float m = mean[row];
  const float r = rstd[row];
  const float ga = gamma[c], be = beta[c];
  if (ACT >= 1) slope = act_slope<ACT>(slope, slope_ptr);
  const T *xb = x + ((long)b * C + c) * S;
  const T *dyb = dy + ((long)b * C + c) * S;

  float sum_dx = 0.f, sum_dxx = 0.f, c_dg = 0.f, c_db = 0.f, d_sl = 0.f;
  for (long i = (long)blockIdx.x * GN_THREADS + threadIdx.x; i < S;
       i += (long)gridDim.x * GN_THREADS) {
    const float xhat = (ld(xb + i) - m) * r;
    float g = ld(dyb + i);
    if (ACT >= 1) {
      const float pre = xhat * ga + be;
      if (ACT == 2 && pre <= 0.f) d_sl += g * pre;
      g = pre > 0.f ? g : g * slope;
    }
    c_db += g;
    c_dg += g * xhat;
    const float dxhat = g * ga;
    sum_dx += dxhat;
    sum_dxx += dxhat * xhat;
  }
  sum_dx = block_sum(sum_dx);
  sum_dxx = block_sum(sum_dxx);
  c_db = block_sum(c_db);
  c_dg = block_sum(c_dg);
  if (ACT == 2) d_sl = block_sum(d_sl);
  if (threadIdx.x == 0) {
    atomicAdd(&row_ws[row * 2 + 0], sum_dx);
    atomicAdd(&row_ws[row * 2 + 1], sum_dxx);
    atomicAdd(&chan_ws[c * 2 + 0], c_db);
    atomicAdd(&chan_ws[c * 2 + 1], c_dg);
    if (ACT == 2) atomicAdd(slope_ws, d_sl);
  }
}

template <typename T, int ACT>
__global__ __launch_bounds__(GN_THREADS) void gn_bwd_apply_kernel(
    const T *__restrict__ dy, const T *__restrict__ x,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    const float *__restrict__ row_ws, T *__restrict__ dx, long S, int C,
    int G, long row_len, long total, float slope,
    const float *__restrict__ slope_ptr) {
  const int Cg = C / G;
  const float inv_n = 1.0f / (float)row_len;
  if (ACT >= 1) slope = act_slope<ACT>(slope, slope_ptr);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long c = (i / S) % C;
    const long row = i / (S * Cg);
    const float r = rstd[row];
    const float xhat = (ld(x + i) - mean[row]) * r;
    float g = ld(dy + i);
    if (ACT >= 1) {
      const float pre = xhat * gamma[c] + beta[c];
      g = pre > 0.f ? g : g * slope;
    }
    const float dxhat = g * gamma[c];
    const float v =
        (dxhat - (row_ws[row * 2 + 0] + xhat * row_ws[row * 2 + 1]) * inv_n) * r;
    st(dx + i, v);
  }
}

// ------------------------------------------------- GN + act + max-pool(K)
//
// SetConv stage 1 is GN -> LeakyReLU -> max over the K neighbour axis
// (reference gconv.py:71-75).  Fusing the pool into the normalize pass
// avoids materialising the (B, C, K, N) activation (208 MB at the default
// config) and the separate pool forward/backward kernels.  Statistics are
// over the FULL (K, N) spatial extent (pre-pool, matching the reference);
// the fwd stats passes above are reused unchanged.
//
// y[b,c,n] = max_k act(xhat[b,c,k,n]*gamma[c]+beta[c]), argmax saved (u8).
// Backward: dy lives on the pooled (B,C,N) domain; dxhat is nonzero only
// at argmax positions, but the GN mean/var coupling makes dx dense.

template <typename T, int ACT>
__global__ __launch_bounds__(GN_THREADS) void gnmp_fwd_apply_kernel(
    const T *__restrict__ x, T *__restrict__ y, unsigned char *__restrict__ am,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta, long N,
    int K, int C, int G, long total_out, float slope,
    const float *__restrict__ slope_ptr) {
  const int Cg = C / G;
  if (ACT >= 1) slope = act_slope<ACT>(slope, slope_ptr);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total_out;
       i += (long)gridDim.x * blockDim.x) {
    const long n = i % N;
    const long c = i / N % C;
    const long b = i / (N * C);
    const long row = b * G + c / Cg;
    const float m = mean[row];
    const float r = rstd[row];
    const float ga = gamma[c], be = beta[c];
    const T *base = x + ((b * C + c) * K) * N + n;
    float best = -INFINITY;
    int bk = 0;
    for (int k = 0; k < K; ++k) {
      float v = (ld(base + (long)k * N) - m) * r * ga + be;
      if (ACT >= 1) v = v > 0.f ? v : v * slope;
      if (v > best) {
        best = v;
        bk = k;
      }
    }
    st(y + i, best);
    am[i] = (unsigned char)bk;
  }
}

// backward pass 1: row/channel sums over the POOLED domain (selected
// positions only carry dxhat / dgamma / dbeta).  Grid (N chunks, C, B):
// one channel per block, four block-reduced partials, four atomics.
template <typename T, int ACT>
__global__ __launch_bounds__(GN_THREADS) void gnmp_bwd_reduce_kernel(
    const T *__restrict__ dy, const T *__restrict__ x,
    const unsigned char *__restrict__ am, const float *__restrict__ mean,
    const float *__restrict__ rstd, const float *__restrict__ gamma,
    const float *__restrict__ beta, float *__restrict__ row_ws,
    float *__restrict__ chan_ws, float *__restrict__ slope_ws, long N, int K,
    int C, int G, float slope, const float *__restrict__ slope_ptr) {
  const int c = blockIdx.y;
  const int b = blockIdx.z;
  const int Cg = C / G;
  const int row = b * G + c / Cg;
  const float m = mean[row];
  const float r = rstd[row];
  const float ga = gamma[c], be = beta[c];
  if (ACT >= 1) slope = act_slope<ACT>(slope, slope_ptr);
  const T *xb = x + ((long)b * C + c) * K * N;
  const long pooled_base = ((long)b * C + c) * N;

  float sum_dx = 0.f, sum_dxx = 0.f, c_dg = 0.f, c_db = 0.f, d_sl = 0.f;
  for (long n = (long)blockIdx.x * GN_THREADS + threadIdx.x; n < N;
       n += (long)gridDim.x * GN_THREADS) {
    const int k = am[pooled_base + n];
    const float xhat = (ld(xb + (long)k * N + n) - m) * r;
    float g = ld(dy + pooled_base + n);
    if (ACT >= 1) {
      const float pre = xhat * ga + be;
      if (ACT == 2 && pre <= 0.f) d_sl += g * pre;
      g = pre > 0.f ? g : g * slope;
    }
    c_db += g;
    c_dg += g * xhat;
    const float dxhat = g * ga;
    sum_dx += dxhat;
    sum_dxx += dxhat * xhat;
  }
  sum_dx = block_sum(sum_dx);
  sum_dxx = block_sum(sum_dxx);
  c_db = block_sum(c_db);
  c_dg = block_sum(c_dg);
  if (ACT == 2) d_sl = block_sum(d_sl);
  if (threadIdx.x == 0) {
    atomicAdd(&row_ws[row * 2 + 0], sum_dx);
    atomicAdd(&row_ws[row * 2 + 1], sum_dxx);
    atomicAdd(&chan_ws[c * 2 + 0], c_db);
    atomicAdd(&chan_ws[c * 2 + 1], c_dg);
    if (ACT == 2) atomicAdd(slope_ws, d_sl);
  }
}

// backward pass 2: dense dx over the full (B, C, K, N) domain.
// Grid (N chunks, C, B), one thread per n looping k: am/dy read once per
// pooled position instead of once per (k, n), and no per-element divides.
template <typename T, int ACT>
__global__ __launch_bounds__(GN_THREADS) void gnmp_bwd_apply_kernel(
    const T *__restrict__ dy, const T *__restrict__ x,
    const unsigned char *__restrict__ am, const float *__restrict__ mean,
    const float *__restrict__ rstd, const float *__restrict__ gamma,
    const float *__restrict__ beta, const float *__restrict__ row_ws,
    T *__restrict__ dx, long N, int K, int C, int G, long row_len,
    float slope, const float *__restrict__ slope_ptr) {
  const int c = blockIdx.y;
  const int b = blockIdx.z;
  const int Cg = C / G;
  const int row = b * G + c / Cg;
  const float m = mean[row];
  const float r = rstd[row];
  const float ga = gamma[c], be = beta[c];
  if (ACT >= 1) slope = act_slope<ACT>(slope, slope_ptr);
  const float inv_n = 1.0f / (float)row_len;
  const float s1 = row_ws[row * 2 + 0];
  const float s2 = row_ws[row * 2 + 1];
  const long base = ((long)b * C + c) * K * N;
  const long pooled_base = ((long)b * C + c) * N;
  for (long n = (long)blockIdx.x * GN_THREADS + threadIdx.x; n < N;
       n += (long)gridDim.x * GN_THREADS) {
    const int ksel = am[pooled_base + n];
    float g = ld(dy + pooled_base + n);
    for (int k = 0; k < K; ++k) {
      const long i = base + (long)k * N + n;
      const float xhat = (ld(x + i) - m) * r;
      float dxhat = 0.f;
      if (k == ksel) {
        float gs = g;
        if (ACT >= 1) {
          const float pre = xhat * ga + be;
          gs = pre > 0.f ? gs : gs * slope;
        }
        dxhat = gs * ga;
      }
      st(dx + i, (dxhat - (s1 + xhat * s2) * inv_n) * r);
    }
  }
}

// --------------------------------------------------------------- launchers

static int pick_blocks_per_row(long row_len, int rows) {
  int bpr = 1;
  while ((long)bpr * rows < 2048 && (long)bpr * GN_THREADS < row_len) bpr *= 2;
  return bpr;
}

template <typename T>
void gn_fwd_impl(const T *x, T *y, float *ws, float *mean, float *rstd,
                 const float *gamma, const float *beta, int rows, long row_len,
                 long S, int C, int G, float eps, int act, float slope,
                 const float *slope_ptr, hipStream_t stream) {
  const long total = (long)rows * row_len;
  const int bpr = pick_blocks_per_row(row_len, rows);
  hipLaunchKernelGGL(gn_fwd_reduce_kernel<T>, dim3(rows * bpr),
                     dim3(GN_THREADS), 0, stream, x, ws, row_len, rows, bpr);
  hipLaunchKernelGGL(gn_fwd_finalize_kernel, dim3((rows + 255) / 256), dim3(256),
                     0, stream, ws, mean, rstd, row_len, rows, eps);
  const int apply_blocks = (int)min((total + GN_THREADS - 1) / GN_THREADS, (long)65535);
#define GN_FWD_APPLY(A)                                                        \
  hipLaunchKernelGGL((gn_fwd_apply_kernel<T, A>), dim3(apply_blocks),          \
                     dim3(GN_THREADS), 0, stream, x, y, mean, rstd, gamma,     \
                     beta, S, C, G, total, slope, slope_ptr)
  if (act == 2) GN_FWD_APPLY(2);
  else if (act == 1) GN_FWD_APPLY(1);
  else GN_FWD_APPLY(0);
#undef GN_FWD_APPLY
}

static int pick_chunks(long spatial, long bc) {
  // enough (chunk, C, B) blocks to fill the chip, but no more than the
  // spatial extent supports
  long want = 4096 / (bc > 0 ? bc : 1);
  long cap = (spatial + GN_THREADS - 1) / GN_THREADS;
  long chunks = want < 1 ? 1 : want;
  if (chunks > cap) chunks = cap;
  return (int)(chunks < 1 ? 1 : chunks);
}

template <typename T>
void gn_bwd_impl(const T *dy, const T *x, const float *mean, const float *rstd,
                 const float *gamma, const float *beta, float *row_ws,
                 float *chan_ws, float *slope_ws, T *dx, int rows, long row_len,
                 long S, int C, int G, int act, float slope,
                 const float *slope_ptr, hipStream_t stream) {
  const long total = (long)rows * row_len;
  const int B = rows / G;
  const dim3 rgrid(pick_chunks(S, (long)B * C), C, B);
  const int apply_blocks = (int)min((total + GN_THREADS - 1) / GN_THREADS, (long)65535);
#define GN_BWD(A)                                                              \
  do {                                                                         \
    hipLaunchKernelGGL((gn_bwd_reduce_kernel<T, A>), rgrid, dim3(GN_THREADS),  \
                       0, stream, dy, x, mean, rstd, gamma, beta, row_ws,      \
                       chan_ws, slope_ws, S, C, G, slope, slope_ptr);          \
    hipLaunchKernelGGL((gn_bwd_apply_kernel<T, A>), dim3(apply_blocks),        \
                       dim3(GN_THREADS), 0, stream, dy, x, mean, rstd, gamma,  \
                       beta, row_ws, dx, S, C, G, row_len, total, slope,       \
                       slope_ptr);                                             \
  } while (0)
  if (act == 2) GN_BWD(2);
  else if (act == 1) GN_BWD(1);
  else GN_BWD(0);
#undef GN_BWD
}

template <typename T>
void gnmp_fwd_impl(const T *x, T *y, unsigned char *am, float *ws, float *mean,
                   float *rstd, const float *gamma, const float *beta, int rows,
                   long row_len, long N, int K, int C, int G, float eps,
                   int act, float slope, const float *slope_ptr,
                   hipStream_t stream) {
  const int B = rows / G;
  const long total_out = (long)B * C * N;
  const int bpr = pick_blocks_per_row(row_len, rows);
  hipLaunchKernelGGL(gn_fwd_reduce_kernel<T>, dim3(rows * bpr),
                     dim3(GN_THREADS), 0, stream, x, ws, row_len, rows, bpr);
  hipLaunchKernelGGL(gn_fwd_finalize_kernel, dim3((rows + 255) / 256), dim3(256),
                     0, stream, ws, mean, rstd, row_len, rows, eps);
  const int blocks = (int)min((total_out + GN_THREADS - 1) / GN_THREADS, (long)65535);
#define GNMP_FWD(A)                                                            \
  hipLaunchKernelGGL((gnmp_fwd_apply_kernel<T, A>), dim3(blocks),              \
                     dim3(GN_THREADS), 0, stream, x, y, am, mean, rstd, gamma, \
                     beta, N, K, C, G, total_out, slope, slope_ptr)
  if (act == 2) GNMP_FWD(2);
  else if (act == 1) GNMP_FWD(1);
  else GNMP_FWD(0);
#undef GNMP_FWD
}

template <typename T>
void gnmp_bwd_impl(const T *dy, const T *x, const unsigned char *am,
                   const float *mean, const float *rstd, const float *gamma,
                   const float *beta, float *row_ws, float *chan_ws,
                   float *slope_ws, T *dx, int rows, long row_len, long N,
                   int K, int C, int G, int act, float slope,
                   const float *slope_ptr, hipStream_t stream) {
  const int B = rows / G;
  const dim3 rgrid(pick_chunks(N, (long)B * C), C, B);
#define GNMP_BWD(A)                                                            \
  do {                                                                         \
    hipLaunchKernelGGL((gnmp_bwd_reduce_kernel<T, A>), rgrid,                  \
                       dim3(GN_THREADS), 0, stream, dy, x, am, mean, rstd,     \
                       gamma, beta, row_ws, chan_ws, slope_ws, N, K, C, G,     \
                       slope, slope_ptr);                                      \
    hipLaunchKernelGGL((gnmp_bwd_apply_kernel<T, A>), rgrid,                   \
                       dim3(GN_THREADS), 0, stream, dy, x, am, mean, rstd,     \
                       gamma, beta, row_ws, dx, N, K, C, G, row_len, slope,    \
                       slope_ptr);                                             \
  } while (0)
  if (act == 2) GNMP_BWD(2);
  else if (act == 1) GNMP_BWD(1);
  else GNMP_BWD(0);
#undef GNMP_BWD
}

// type-erased entry points (bindings.cpp is host-compiled, no HIP types)
void launch_gnmp_fwd(const void *x, void *y, unsigned char *am, float *ws,
                     float *mean, float *rstd, const float *gamma,
                     const float *beta, int rows, long row_len, long N, int K,
                     int C, int G, float eps, int act, float slope,
                     const float *slope_ptr, bool bf16, hipStream_t stream) {
  if (bf16)
    gnmp_fwd_impl<__hip_bfloat16>((const __hip_bfloat16 *)x,
                                  (__hip_bfloat16 *)y, am, ws, mean, rstd,
                                  gamma, beta, rows, row_len, N, K, C, G, eps,
                                  act, slope, slope_ptr, stream);
  else
    gnmp_fwd_impl<float>((const float *)x, (float *)y, am, ws, mean, rstd,
                         gamma, beta, rows, row_len, N, K, C, G, eps, act,
                         slope, slope_ptr, stream);
}

void launch_gnmp_bwd(const void *dy, const void *x, const unsigned char *am,
                     const float *mean, const float *rstd, const float *gamma,
                     const float *beta, float *row_ws, float *chan_ws,
                     float *slope_ws, void *dx, int rows, long row_len, long N,
                     int K, int C, int G, int act, float slope,
                     const float *slope_ptr, bool bf16, hipStream_t stream) {
  if (bf16)
    gnmp_bwd_impl<__hip_bfloat16>(
        (const __hip_bfloat16 *)dy, (const __hip_bfloat16 *)x, am, mean, rstd,
        gamma, beta, row_ws, chan_ws, slope_ws, (__hip_bfloat16 *)dx, rows,
        row_len, N, K, C, G, act, slope, slope_ptr, stream);
  else
    gnmp_bwd_impl<float>((const float *)dy, (const float *)x, am, mean, rstd,
                         gamma, beta, row_ws, chan_ws, slope_ws, (float *)dx,
                         rows, row_len, N, K, C, G, act, slope, slope_ptr,
                         stream);
}

void launch_gn_fwd(const void *x, void *y, float *ws, float *mean, float *rstd,
                   const float *gamma, const float *beta, int rows,
                   long row_len, long S, int C, int G, float eps, int act,
                   float slope, const float *slope_ptr, bool bf16,
                   hipStream_t stream) {
  if (bf16)
    gn_fwd_impl<__hip_bfloat16>((const __hip_bfloat16 *)x, (__hip_bfloat16 *)y,
                                ws, mean, rstd, gamma, beta, rows, row_len, S,
                                C, G, eps, act, slope, slope_ptr, stream);
  else
    gn_fwd_impl<float>((const float *)x, (float *)y, ws, mean, rstd, gamma,
                       beta, rows, row_len, S, C, G, eps, act, slope,
                       slope_ptr, stream);
}

void launch_gn_bwd(const void *dy, const void *x, const float *mean,
                   const float *rstd, const float *gamma, const float *beta,
                   float *row_ws, float *chan_ws, float *slope_ws, void *dx,
                   int rows, long row_len, long S, int C, int G, int act,
                   float slope, const float *slope_ptr, bool bf16,
                   hipStream_t stream) {
  if (bf16)
    gn_bwd_impl<__hip_bfloat16>((const __hip_bfloat16 *)dy,
                                (const __hip_bfloat16 *)x, mean, rstd, gamma,
                                beta, row_ws, chan_ws, slope_ws,
                                (__hip_bfloat16 *)dx, rows, row_len, S, C, G,
                                act, slope, slope_ptr, stream);
  else
    gn_bwd_impl<float>((const float *)dy, (const float *)x, mean, rstd, gamma,
                       beta, row_ws, chan_ws, slope_ws, (float *)dx, rows,
                       row_len, S, C, G, act, slope, slope_ptr, stream);
}
