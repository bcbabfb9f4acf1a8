// K8: fused kNN-correlation branch head: Conv2d(4->C) -> GroupNorm ->
// PReLU -> max over k, in one kernel pipeline.
//
// Reference semantics (model/corr.py:75-98 kNN branch; knn_conv defined
// at corr.py:23-29): raw per-candidate features [corr; rel-xyz]
// (B, 4, K, N) run through a 1x1 conv to C=64 channels, GroupNorm(8),
// PReLU, then max-pool over the K candidates.  Run as a GEMM the conv materialises a (B, C, K, N)
// activation (~33 M elements at the flagship shape) that GroupNorm and
// the pool each re-read -- ~1.5 ms/step of traffic + launches for a
// 4-wide contraction.
//
// Here the conv is evaluated INLINE: the edge vector has only 4
// components, so v[c, j, n] = W[c, :4] @ raw[:, j, n] + cb[c] is 4 FMAs
// from wave-uniform (SGPR) weights.  The reduce pass streams raw once,
// tracks per-(n, c) extremes of v over j (act(GN(v)) is piecewise
// monotone in v -- same argument as edge_gnmp) plus the GroupNorm
// sums, and the shared egnmp pick/finalize kernels produce the pooled
// output; the (B, C, K, N) tensor never exists, forward or backward.
//
// Channels are processed in CH=16 chunks (blockIdx.y): full-C state
// would need ~300 VGPRs/thread, so each chunk re-streams raw (4 MB; L2-
// resident after the first chunk) and owns 2 of the 8 groups.  All
// reductions are deterministic via per-block scratch partials folded by
// egnmp_sum_partials (no global atomics).
//
// Backward (one pass + small folds, no CSR -- the domain has no
// cross-point coupling): recomputes v on the fly, forms the GroupNorm
// backward element dv, contracts d_raw = W^T dv into per-chunk partial
// buffers (summed by a tiny combine kernel), and bins dW / dcb /
// dgamma / dbeta / dslope into scratch partials.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

#define KG_THREADS 256
// channels per chunk (blockIdx.y): the forward uses 8 (more blocks, the
// reduce is latency-bound at 1 wave/SIMD otherwise), the backward 16
// (each chunk owns a d_raw partial buffer -- fewer chunks, less combine
// traffic)

// shared folding / finalize / pick infrastructure (edge_gnmp.hip,
// group_norm.hip)
void launch_gn_finalize_scratch(const float *, long, float *, float *, long,
                                int, float, hipStream_t);
__global__ void egnmp_sum_partials_kernel(const float *__restrict__,
                                          float *__restrict__, long, int);
void launch_gnmp_pick(const float *, const float *, const unsigned char *,
                      const unsigned char *, const float *, const float *,
                      const float *, const float *, void *, unsigned char *,
                      float *, long, long, int, int, int, float,
                      const float *, bool, hipStream_t);

// ---------------------------------------------------------------- forward

template <int KG_CH>
__global__ __launch_bounds__(KG_THREADS) void kg_fwd_reduce_kernel(
    const float *__restrict__ raw,  // (B, 4, K, N)
    const float *__restrict__ W,    // (C, 4)
    const float *__restrict__ cb,   // (C) conv bias
    float *__restrict__ scratch,    // (B*G*2, gridX*chunks*B)
    float *__restrict__ vmax, float *__restrict__ vmin,  // (B, N, C) fp32
    unsigned char *__restrict__ amax, unsigned char *__restrict__ amin,
    long N, int K, int C, int G) {
  const int b = blockIdx.z;
  const int B = gridDim.z;
  const int chunk = blockIdx.y;
  const int c0 = chunk * KG_CH;
  const int Cg = C / G;
  const int g0 = c0 / Cg;                       // first group touched
  const int NG = (c0 + KG_CH - 1) / Cg - g0 + 1;  // groups touched (<=8)
  const int n_out = B * G * 2;

  // weights are wave-uniform: the compiler keeps these in SGPRs
  float wr[KG_CH][4], br[KG_CH];
#pragma unroll
  for (int c = 0; c < KG_CH; ++c) {
#pragma unroll
    for (int e = 0; e < 4; ++e) wr[c][e] = W[(c0 + c) * 4 + e];
    br[c] = cb[c0 + c];
  }

  __shared__ float bins[8 * 2];  // (NG<=8, 2): this block's group sums
  if (threadIdx.x < (unsigned)(NG * 2)) bins[threadIdx.x] = 0.f;
  __syncthreads();

  const float *rawb = raw + (long)b * 4 * K * N;
  float s[8] = {}, ss[8] = {};  // per local group (launcher checks NG<=8)

  for (long n = (long)blockIdx.x * KG_THREADS + threadIdx.x; n < N;
       n += (long)gridDim.x * KG_THREADS) {
    float vmx[KG_CH], vmn[KG_CH], vs[KG_CH];
    int jmx[KG_CH], jmn[KG_CH];
#pragma unroll
    for (int c = 0; c < KG_CH; ++c) {
      vmx[c] = -INFINITY;
      vmn[c] = INFINITY;
      vs[c] = 0.f;
      jmx[c] = 0;
      jmn[c] = 0;
    }
    for (int j = 0; j < K; ++j) {
      const float r0 = rawb[((long)0 * K + j) * N + n];
      const float r1 = rawb[((long)1 * K + j) * N + n];
      const float r2 = rawb[((long)2 * K + j) * N + n];
      const float r3 = rawb[((long)3 * K + j) * N + n];
#pragma unroll
      for (int c = 0; c < KG_CH; ++c) {
        const float v = wr[c][0] * r0 + wr[c][1] * r1 + wr[c][2] * r2 +
                        wr[c][3] * r3 + br[c];
        vs[c] += v;
        ss[(c0 + c) / Cg - g0] += v * v;
        if (v > vmx[c]) {
          vmx[c] = v;
          jmx[c] = j;
        }
        if (v < vmn[c]) {
          vmn[c] = v;
          jmn[c] = j;
        }
      }
    }
#pragma unroll
    for (int c = 0; c < KG_CH; ++c) s[(c0 + c) / Cg - g0] += vs[c];
    const long pi = ((long)b * N + n) * C + c0;
#pragma unroll
    for (int c4 = 0; c4 < KG_CH / 4; ++c4) {
      *(float4 *)(vmax + pi + c4 * 4) =
          make_float4(vmx[c4 * 4], vmx[c4 * 4 + 1], vmx[c4 * 4 + 2],
                      vmx[c4 * 4 + 3]);
      *(float4 *)(vmin + pi + c4 * 4) =
          make_float4(vmn[c4 * 4], vmn[c4 * 4 + 1], vmn[c4 * 4 + 2],
                      vmn[c4 * 4 + 3]);
      uchar4 ax, an;
      ax.x = (unsigned char)jmx[c4 * 4];
      ax.y = (unsigned char)jmx[c4 * 4 + 1];
      ax.z = (unsigned char)jmx[c4 * 4 + 2];
      ax.w = (unsigned char)jmx[c4 * 4 + 3];
      an.x = (unsigned char)jmn[c4 * 4];
      an.y = (unsigned char)jmn[c4 * 4 + 1];
      an.z = (unsigned char)jmn[c4 * 4 + 2];
      an.w = (unsigned char)jmn[c4 * 4 + 3];
      *(uchar4 *)(amax + pi + c4 * 4) = ax;
      *(uchar4 *)(amin + pi + c4 * 4) = an;
    }
  }
#pragma unroll
  for (int g = 0; g < 8; ++g)
    if (g < NG) {
      atomicAdd(&bins[g * 2 + 0], s[g]);
      atomicAdd(&bins[g * 2 + 1], ss[g]);
    }
  __syncthreads();
  // scratch layout: (B*G*2 rows, gridX*chunks*B cols); rows not covered
  // by this chunk get zero so the fold stays correct
  const long cols = (long)gridDim.x * gridDim.y * B;
  const long col = ((long)blockIdx.x * gridDim.y + chunk) * B + b;
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += KG_THREADS) {
    const int row_b = i / (G * 2);
    const int row_g = (i % (G * 2)) / 2;
    const int half = i & 1;
    float v = 0.f;
    if (row_b == b && row_g >= g0 && row_g < g0 + NG)
      v = bins[(row_g - g0) * 2 + half];
    scratch[(long)i * cols + col] = v;
  }
}

// ---------------------------------------------------------------- backward

// pass 1: GN backward sums over the POOLED domain (elementwise -- the
// argmax element's pre-GN value is saved in vsel).  Bins layout matches
// the egnmp backward ws: [B*G*2 | C*2 | 1].
template <typename T>
__global__ __launch_bounds__(KG_THREADS) void kg_bwd_reduce_kernel(
    const T *__restrict__ dyT,               // (B, N, C) pooled grad
    const float *__restrict__ vsel,          // (B, N, C)
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    float *__restrict__ scratch, long N, int C, int G,
    const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int B = gridDim.z;
  const float slope = *slope_ptr;
  const int Cg = C / G;
  const int tpc = C / 4;
  const int ppb = KG_THREADS / tpc;
  const int p_l = (int)threadIdx.x / tpc;
  const int c4 = (int)threadIdx.x % tpc;
  const bool active = p_l < ppb;
  const int n_out = B * G * 2 + C * 2 + 1;

  extern __shared__ float sbins[];
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += KG_THREADS)
    sbins[i] = 0.f;
  __syncthreads();

  float sum_dx[4] = {}, sum_dxx[4] = {}, c_dg[4] = {}, c_db[4] = {};
  float d_sl = 0.f;
  float m[4], r[4], ga[4], be[4];
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    const int c = c4 * 4 + e;
    m[e] = mean[b * G + c / Cg];
    r[e] = rstd[b * G + c / Cg];
    ga[e] = gamma[c];
    be[e] = beta[c];
  }
  if (active) {
    for (long n = (long)blockIdx.x * ppb + p_l; n < N;
         n += (long)gridDim.x * ppb) {
      const long pi = ((long)b * N + n) * C + c4 * 4;
      const float4 sv = *(const float4 *)(vsel + pi);
      const float svv[4] = {sv.x, sv.y, sv.z, sv.w};
      const T *dp = dyT + pi;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float xhat = (svv[e] - m[e]) * r[e];
        float gv = (float)dp[e];
        const float pre = xhat * ga[e] + be[e];
        if (pre <= 0.f) d_sl += gv * pre;
        gv = pre > 0.f ? gv : gv * slope;
        c_db[e] += gv;
        c_dg[e] += gv * xhat;
        const float dxhat = gv * ga[e];
        sum_dx[e] += dxhat;
        sum_dxx[e] += dxhat * xhat;
      }
    }
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int c = c4 * 4 + e;
      const int g = c / Cg;
      atomicAdd(&sbins[(b * G + g) * 2 + 0], sum_dx[e]);
      atomicAdd(&sbins[(b * G + g) * 2 + 1], sum_dxx[e]);
      atomicAdd(&sbins[B * G * 2 + c * 2 + 0], c_db[e]);
      atomicAdd(&sbins[B * G * 2 + c * 2 + 1], c_dg[e]);
    }
    atomicAdd(&sbins[B * G * 2 + C * 2], d_sl);
  }
  __syncthreads();
  const long col = (long)blockIdx.x * B + b;
  const long stride = (long)gridDim.x * B;
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += KG_THREADS)
    scratch[i * stride + col] = sbins[i];
}

// pass 2: d_raw partials per channel chunk + dW / dcb partials.
// dv[c, j, n] = (dxhat - (s1 + xhat*s2)/len) * rstd, dxhat only at the
// pooled argmax j; d_raw[:, j, n] = sum_c W[c, :] * dv[c]; dW[c, :] +=
// dv[c] * raw[:, j, n]; dcb[c] += dv[c].
template <typename T, int KG_CH, int JCH>
__global__ __launch_bounds__(KG_THREADS) void kg_bwd_apply_kernel(
    const T *__restrict__ dyT,               // (B, N, C)
    const float *__restrict__ raw,           // (B, 4, K, N)
    const float *__restrict__ W, const float *__restrict__ cb,
    const unsigned char *__restrict__ am,    // (B, N, C) pooled argmax
    const float *__restrict__ vsel, const float *__restrict__ mean,
    const float *__restrict__ rstd, const float *__restrict__ gamma,
    const float *__restrict__ beta, const float *__restrict__ ws,
    float *__restrict__ draw_part,  // (chunks, B, 4, K, N)
    float *__restrict__ wscratch,   // (C*4 + C, gridX*chunks*B)
    long N, int K, int C, int G, long row_len,
    const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int B = gridDim.z;
  const int chunk = blockIdx.y % ((C + KG_CH - 1) / KG_CH);
  const int jc = blockIdx.y / ((C + KG_CH - 1) / KG_CH);
  const int c0 = chunk * KG_CH;
  // j-range owned by this block: the per-edge outputs partition cleanly
  // over j (draw writes disjoint slices; dW partials fold per block)
  const int jn = (K + JCH - 1) / JCH;
  const int j_lo = jc * jn;
  const int j_hi = min(K, j_lo + jn);
  const int Cg = C / G;
  const float slope = *slope_ptr;
  const float inv_n = 1.0f / (float)row_len;
  const int n_out = C * 4 + C;

  float wr[KG_CH][4];
#pragma unroll
  for (int c = 0; c < KG_CH; ++c)
#pragma unroll
    for (int e = 0; e < 4; ++e) wr[c][e] = W[(c0 + c) * 4 + e];
  float m[KG_CH], r[KG_CH], ga[KG_CH], be[KG_CH], s1[KG_CH], s2[KG_CH];
#pragma unroll
  for (int c = 0; c < KG_CH; ++c) {
    const int cc = c0 + c;
    const int row = b * G + cc / Cg;
    m[c] = mean[row];
    r[c] = rstd[row];
    ga[c] = gamma[cc];
    be[c] = beta[cc];
    s1[c] = ws[row * 2 + 0];
    s2[c] = ws[row * 2 + 1];
  }

  __shared__ float bins[KG_CH * 5];  // dW (CH,4) + dcb (CH)
  for (unsigned i = threadIdx.x; i < (unsigned)(KG_CH * 5); i += KG_THREADS)
    bins[i] = 0.f;
  __syncthreads();

  const float *rawb = raw + (long)b * 4 * K * N;
  float *drawb = draw_part + ((long)chunk * B + b) * 4 * K * N;
  float dw[KG_CH][4] = {}, db[KG_CH] = {};

  for (long n = (long)blockIdx.x * KG_THREADS + threadIdx.x; n < N;
       n += (long)gridDim.x * KG_THREADS) {
    const long pi = ((long)b * N + n) * C + c0;
    // pooled-gradient terms for this chunk's channels
    float gsel[KG_CH];
    int ksel[KG_CH];
#pragma unroll
    for (int c4 = 0; c4 < KG_CH / 4; ++c4) {
      const uchar4 aq = *(const uchar4 *)(am + pi + c4 * 4);
      const float4 sv = *(const float4 *)(vsel + pi + c4 * 4);
      const int ks[4] = {aq.x, aq.y, aq.z, aq.w};
      const float svv[4] = {sv.x, sv.y, sv.z, sv.w};
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int c = c4 * 4 + e;
        float gv = (float)dyT[pi + c];
        const float pre = (svv[e] - m[c]) * r[c] * ga[c] + be[c];
        gv = pre > 0.f ? gv : gv * slope;
        gsel[c] = gv * ga[c];
        ksel[c] = ks[e];
      }
    }
    for (int j = j_lo; j < j_hi; ++j) {
      const float r0 = rawb[((long)0 * K + j) * N + n];
      const float r1 = rawb[((long)1 * K + j) * N + n];
      const float r2 = rawb[((long)2 * K + j) * N + n];
      const float r3 = rawb[((long)3 * K + j) * N + n];
      float d0 = 0.f, d1 = 0.f, d2 = 0.f, d3 = 0.f;
#pragma unroll
      for (int c = 0; c < KG_CH; ++c) {
        const float v = wr[c][0] * r0 + wr[c][1] * r1 + wr[c][2] * r2 +
                        wr[c][3] * r3 + cb[c0 + c];
        const float xhat = (v - m[c]) * r[c];
        const float dxhat = (j == ksel[c]) ? gsel[c] : 0.f;
        const float dv = (dxhat - (s1[c] + xhat * s2[c]) * inv_n) * r[c];
        d0 += wr[c][0] * dv;
        d1 += wr[c][1] * dv;
        d2 += wr[c][2] * dv;
        d3 += wr[c][3] * dv;
        dw[c][0] += dv * r0;
        dw[c][1] += dv * r1;
        dw[c][2] += dv * r2;
        dw[c][3] += dv * r3;
        db[c] += dv;
      }
      drawb[((long)0 * K + j) * N + n] = d0;
      drawb[((long)1 * K + j) * N + n] = d1;
      drawb[((long)2 * K + j) * N + n] = d2;
      drawb[((long)3 * K + j) * N + n] = d3;
    }
  }
#pragma unroll
  for (int c = 0; c < KG_CH; ++c) {
#pragma unroll
    for (int e = 0; e < 4; ++e) atomicAdd(&bins[c * 4 + e], dw[c][e]);
    atomicAdd(&bins[KG_CH * 4 + c], db[c]);
  }
  __syncthreads();
  // wscratch rows: [C*4 dW | C dcb]; this chunk owns rows c0..c0+CH
  const long cols = (long)gridDim.x * gridDim.y * B;
  const long col = ((long)blockIdx.x * gridDim.y + blockIdx.y) * B + b;
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += KG_THREADS) {
    float v = 0.f;
    if (i < (unsigned)(C * 4)) {
      const int c = i / 4;
      if (c >= c0 && c < c0 + KG_CH) v = bins[(c - c0) * 4 + (i & 3)];
    } else {
      const int c = i - C * 4;
      if (c >= c0 && c < c0 + KG_CH) v = bins[KG_CH * 4 + (c - c0)];
    }
    wscratch[(long)i * cols + col] = v;
  }
}

// sum the per-chunk d_raw partial buffers into the output
__global__ void kg_draw_combine_kernel(const float *__restrict__ draw_part,
                                       float *__restrict__ draw, long per_b,
                                       int chunks, int B) {
  const long total = per_b * B;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / per_b;
    const long off = i % per_b;
    float acc = 0.f;
    for (int c = 0; c < chunks; ++c)
      acc += draw_part[((long)c * B + b) * per_b + off];
    draw[i] = acc;
  }
}

// --------------------------------------------------------------- launchers

void launch_kg_fwd(const float *raw, const float *W, const float *cb,
                   float *scratch, float *ws, float *mean, float *rstd,
                   const float *gamma, const float *beta, float *vmax,
                   float *vmin, unsigned char *amax, unsigned char *amin,
                   void *y, unsigned char *am, float *vsel,
                   int B, long N, int K, int C, int G, float eps,
                   const float *slope_ptr, bool bf16, int nblk,
                   hipStream_t stream) {
  const int chunks = C / 4;
  const dim3 grid(nblk, chunks, B);
  hipLaunchKernelGGL(kg_fwd_reduce_kernel<4>, grid, dim3(KG_THREADS), 0,
                     stream, raw, W, cb, scratch, vmax, vmin, amax, amin, N,
                     K, C, G);
  (void)ws;
  launch_gn_finalize_scratch(scratch, (long)nblk * chunks * B, mean, rstd,
                             (long)(C / G) * K * N, B * G, eps, stream);
  launch_gnmp_pick(vmax, vmin, amax, amin, mean, rstd, gamma, beta, y, am,
                   vsel, (long)B * N * C, (long)N * C, C, G, 2, 0.f,
                   slope_ptr, bf16, stream);
}

void launch_kg_bwd(const void *dyT, const float *raw, const float *W,
                   const float *cb, const unsigned char *am,
                   const float *vsel, const float *mean, const float *rstd,
                   const float *gamma, const float *beta, float *scratch,
                   float *ws, float *wscratch, float *ws2, float *draw_part,
                   float *draw, int B, long N, int K, int C, int G,
                   const float *slope_ptr, bool bf16, int nblk,
                   hipStream_t stream) {
  const int chunks = C / 8;
  const long row_len = (long)(C / G) * K * N;
  {
    const dim3 rgrid(nblk, 1, B);
    const int n_out = B * G * 2 + C * 2 + 1;
    if (bf16)
      hipLaunchKernelGGL(kg_bwd_reduce_kernel<__hip_bfloat16>, rgrid,
                         dim3(KG_THREADS), (size_t)n_out * sizeof(float),
                         stream, (const __hip_bfloat16 *)dyT, vsel, mean,
                         rstd, gamma, beta, scratch, N, C, G, slope_ptr);
    else
      hipLaunchKernelGGL(kg_bwd_reduce_kernel<float>, rgrid,
                         dim3(KG_THREADS), (size_t)n_out * sizeof(float),
                         stream, (const float *)dyT, vsel, mean, rstd,
                         gamma, beta, scratch, N, C, G, slope_ptr);
    const int wpb = KG_THREADS / WAVE;
    hipLaunchKernelGGL(egnmp_sum_partials_kernel,
                       dim3((n_out + wpb - 1) / wpb), dim3(KG_THREADS), 0,
                       stream, scratch, ws, (long)nblk * B, n_out);
  }
  {
    const dim3 agrid(nblk, chunks * 4, B);
    if (bf16)
      hipLaunchKernelGGL((kg_bwd_apply_kernel<__hip_bfloat16, 8, 4>), agrid,
                         dim3(KG_THREADS), 0, stream,
                         (const __hip_bfloat16 *)dyT, raw, W, cb, am, vsel,
                         mean, rstd, gamma, beta, ws, draw_part, wscratch, N,
                         K, C, G, row_len, slope_ptr);
    else
      hipLaunchKernelGGL((kg_bwd_apply_kernel<float, 8, 4>), agrid,
                         dim3(KG_THREADS), 0, stream, (const float *)dyT,
                         raw, W, cb, am, vsel, mean, rstd, gamma, beta, ws,
                         draw_part, wscratch, N, K, C, G, row_len,
                         slope_ptr);
    const int n_out = C * 4 + C;
    const int wpb = KG_THREADS / WAVE;
    hipLaunchKernelGGL(egnmp_sum_partials_kernel,
                       dim3((n_out + wpb - 1) / wpb), dim3(KG_THREADS), 0,
                       stream, wscratch, ws2, (long)nblk * chunks * 4 * B,
                       n_out);
    const long per_b = (long)4 * K * N;
    long blocks = (per_b * B + KG_THREADS - 1) / KG_THREADS;
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(kg_draw_combine_kernel, dim3((unsigned)blocks),
                       dim3(KG_THREADS), 0, stream, draw_part, draw, per_b,
                       chunks, B);
  }
}
