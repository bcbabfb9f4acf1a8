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
// gather reads a contiguous M-vector; every thread owns a 4-channel quad
// (8 B bf16 / 16 B fp32 vector loads) and threads of one point cover
// adjacent quads (coalesced).  Outputs y/argmax are point-major too; the
// caller transposes the 3 MB pooled result with the LDS-tiled transpose
// kernel (trivial next to the 100 MB it replaces).
//
// Reductions are DETERMINISTIC: each block writes its partial sums to a
// per-block scratch slot (no global atomics); the forward GN stats fold
// straight from that scratch in the finalize kernel, the backward sums
// through one wave-per-output fold.
//
// Backward data path is deterministic too: the gradient w.r.t. WgT at
// point p is
//   dWg[m, p] = sum_{edges e=(j,n): idx[n,j]=p} dx1[m, j, n]   (incoming)
//             - sum_j dx1[m, j, p]                             (centre)
// where dx1 is the standard GroupNorm+act+maxpool backward element
// (exact same formulas as gnmp_bwd_* in group_norm.hip).  The CENTRE sum
// closes over j analytically (only j = argmax carries dy, recovered from
// the saved pre-GN extreme; sum_j xhat_j derives from the per-point
// gather sum the forward reduce stores) -- no K-loop in the apply pass.
// The INCOMING sum walks the inverse-adjacency CSR via the (order_n,
// order_j) side arrays -- source point and neighbour slot per ordered
// edge, no id decomposition, dy a scalar load only on the argmax hit;
// every (p, m) output is written exactly once.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

#define EG_THREADS 256

// shared finalize pass (mean/rstd folded straight from the per-block
// partial scratch) lives in group_norm.hip
void launch_gn_finalize_scratch(const float *, long, float *, float *, long,
                                int, float, hipStream_t);

template <typename T>
DEV_INLINE float ldg(const T *p) {
  return (float)*p;
}
template <typename T>
DEV_INLINE void stg(T *p, float v) {
  *p = (T)v;
}

// 4-channel quad vector (8 B for bf16, 16 B for fp32)
template <typename T>
struct alignas(4 * sizeof(T)) Quad {
  T v[4];
};

// ---------------------------------------------------------------- forward

// pass 1: per-block partial sum/sumsq per group over all (c, j, n)
// gather-diff elements -> scratch[(out)*(gridX*B) + gx*B + b], out in
// [0, G*2).  Deterministic (no atomics past the LDS bins).
// ... and, per (n, c), the EXTREMES of v over j with their arg j: the
// activation(GN(v)) is piecewise monotone in v (pre = v*(rstd*gamma) +
// const; LeakyReLU/PReLU monotone per sign region), so the K-max-pool
// resolves later from just (vmax, vmin) -- the apply pass needs NO gather
// sweep at all (it was a second full pass over the 50M gathered values).
template <typename T>
__global__ __launch_bounds__(EG_THREADS) void egnmp_fwd_reduce_kernel(
    const T *__restrict__ wg,      // (B, N, M)
    const int *__restrict__ idx,   // (B, N, K)
    float *__restrict__ scratch,   // (B*G*2, gridX*B)
    float *__restrict__ vmax, float *__restrict__ vmin,  // (B, N, M) fp32:
    // the backward recovers activation-branch signs from these; a bf16
    // round here flips the slope branch near zero (0.9 max grad error)
    unsigned char *__restrict__ amax,
    unsigned char *__restrict__ amin,                  // (B, N, M)
    float *__restrict__ vsum,      // (B, N, M): sum_j v per point (fp32)
    long N, int K, int M, int G) {
  const int b = blockIdx.z;
  const int B = gridDim.z;
  const int tpc = M / 4;                 // threads per point (quad each)
  const int ppb = EG_THREADS / tpc;      // points per block iteration
  const int p_l = (int)threadIdx.x / tpc;
  const int c4 = (int)threadIdx.x % tpc;
  const bool active = p_l < ppb;
  const int Cg = M / G;
  const int n_out = B * G * 2;

  extern __shared__ float bins[];  // (B*G*2); only this block's b is used
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += EG_THREADS)
    bins[i] = 0.f;
  __syncthreads();

  const T *wgb = wg + (long)b * N * M;
  const int *idxb = idx + (long)b * N * K;
  float s[4] = {0.f, 0.f, 0.f, 0.f}, ss[4] = {0.f, 0.f, 0.f, 0.f};
  if (active) {
    for (long n = (long)blockIdx.x * ppb + p_l; n < N;
         n += (long)gridDim.x * ppb) {
      const Quad<T> cq = *(const Quad<T> *)(wgb + n * M + c4 * 4);
      const int *row = idxb + n * K;
      float vmx[4], vmn[4], ps[4] = {0.f, 0.f, 0.f, 0.f};
      int jmx[4] = {0, 0, 0, 0}, jmn[4] = {0, 0, 0, 0};
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        vmx[e] = -INFINITY;
        vmn[e] = INFINITY;
      }
      for (int j = 0; j < K; ++j) {
        const Quad<T> nq = *(const Quad<T> *)(wgb + (long)row[j] * M + c4 * 4);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const float v = (float)nq.v[e] - (float)cq.v[e];
          ps[e] += v;
          ss[e] += v * v;
          if (v > vmx[e]) {
            vmx[e] = v;
            jmx[e] = j;
          }
          if (v < vmn[e]) {
            vmn[e] = v;
            jmn[e] = j;
          }
        }
      }
#pragma unroll
      for (int e = 0; e < 4; ++e) s[e] += ps[e];
      uchar4 ax, an;
      ax.x = (unsigned char)jmx[0]; ax.y = (unsigned char)jmx[1];
      ax.z = (unsigned char)jmx[2]; ax.w = (unsigned char)jmx[3];
      an.x = (unsigned char)jmn[0]; an.y = (unsigned char)jmn[1];
      an.z = (unsigned char)jmn[2]; an.w = (unsigned char)jmn[3];
      const long pi = ((long)b * N + n) * M + c4 * 4;
      *(float4 *)(vmax + pi) = make_float4(vmx[0], vmx[1], vmx[2], vmx[3]);
      *(float4 *)(vmin + pi) = make_float4(vmn[0], vmn[1], vmn[2], vmn[3]);
      *(uchar4 *)(amax + pi) = ax;
      *(uchar4 *)(amin + pi) = an;
      // per-point gather sum: lets the backward centre term close over j
      // (sum_j xhat_j derives from it) with no K-loop at all
      *(float4 *)(vsum + pi) =
          make_float4(ps[0], ps[1], ps[2], ps[3]);
    }
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int g = (c4 * 4 + e) / Cg;
      atomicAdd(&bins[(b * G + g) * 2 + 0], s[e]);
      atomicAdd(&bins[(b * G + g) * 2 + 1], ss[e]);
    }
  }
  __syncthreads();
  const long col = (long)blockIdx.x * B + b;
  const long stride = (long)gridDim.x * B;
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += EG_THREADS)
    scratch[i * stride + col] = bins[i];
}

// fold per-block partials into ws: one wave per output slot.  Slots that
// belong to another batch's rows were written as zero, so the sum is
// correct and deterministic (fixed grid -> fixed summation order).
__global__ void egnmp_sum_partials_kernel(const float *__restrict__ scratch,
                                          float *__restrict__ ws, long cols,
                                          int n_out) {
  const int out = blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (out >= n_out) return;
  const float *src = scratch + (long)out * cols;
  float acc = 0.f;
  for (long i = lane_id(); i < cols; i += WAVE) acc += src[i];
  acc = wave_sum(acc);
  if (lane_id() == 0) ws[out] = acc;
}

// pass 3: ELEMENTWISE pick from the tracked extremes -- for each (n, c)
// the pooled value is max(act(gn(vmax)), act(gn(vmin))) with the matching
// arg j (act(gn(v)) is piecewise monotone in v, so the max over all K
// values is attained at one of the two extremes, for either sign of
// gamma and any activation slope).
template <typename T, int ACT>
__global__ __launch_bounds__(EG_THREADS) void egnmp_fwd_pick_kernel(
    const float *__restrict__ vmax, const float *__restrict__ vmin,
    const unsigned char *__restrict__ amax,
    const unsigned char *__restrict__ amin,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    T *__restrict__ y,               // (B, N, M)
    unsigned char *__restrict__ am,  // (B, N, M)
    float *__restrict__ vsel,        // (B, N, M): chosen pre-GN extreme
    long total, long NM, int M, int G, float slope,
    const float *__restrict__ slope_ptr) {
  if (ACT == 2) slope = *slope_ptr;
  const int Cg = M / G;
  for (long i4 = (long)blockIdx.x * EG_THREADS + threadIdx.x; i4 * 4 < total;
       i4 += (long)gridDim.x * EG_THREADS) {
    const long i = i4 * 4;
    const long b = i / NM;
    const int c0 = (int)(i % M);
    const float4 qx = *(const float4 *)(vmax + i);
    const float4 qn = *(const float4 *)(vmin + i);
    const uchar4 ax = *(const uchar4 *)(amax + i);
    const uchar4 an = *(const uchar4 *)(amin + i);
    const float qxs[4] = {qx.x, qx.y, qx.z, qx.w};
    const float qns[4] = {qn.x, qn.y, qn.z, qn.w};
    const int axs[4] = {ax.x, ax.y, ax.z, ax.w};
    const int ans[4] = {an.x, an.y, an.z, an.w};
    Quad<T> oq;
    float sq[4];
    uchar4 aq;
    unsigned char out_j[4];
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int c = c0 + e;
      const int row = (int)b * G + c / Cg;
      const float m = mean[row];
      const float r = rstd[row];
      const float ga = gamma[c], be = beta[c];
      float vhi = (qxs[e] - m) * r * ga + be;
      float vlo = (qns[e] - m) * r * ga + be;
      if (ACT >= 1) {
        vhi = vhi > 0.f ? vhi : vhi * slope;
        vlo = vlo > 0.f ? vlo : vlo * slope;
      }
      const bool hi = vhi >= vlo;
      oq.v[e] = (T)(hi ? vhi : vlo);
      sq[e] = hi ? qxs[e] : qns[e];
      out_j[e] = (unsigned char)(hi ? axs[e] : ans[e]);
    }
    aq.x = out_j[0]; aq.y = out_j[1]; aq.z = out_j[2]; aq.w = out_j[3];
    *(Quad<T> *)(y + i) = oq;
    *(uchar4 *)(am + i) = aq;
    *(float4 *)(vsel + i) = make_float4(sq[0], sq[1], sq[2], sq[3]);
  }
}

// ---------------------------------------------------------------- backward

// pass 1: row sums {sum dxhat, sum dxhat*xhat}, channel sums
// {sum dy_act, sum dy_act*xhat} and d slope over the POOLED domain (only
// the argmax element of each (n, c) carries dy).  Partials land in
// scratch exactly like the forward reduce; out layout matches the ws
// layout [rows*2 | M*2 | 1] so one fold kernel serves both.
template <typename T, int ACT>
__global__ __launch_bounds__(EG_THREADS) void egnmp_bwd_reduce_kernel(
    const T *__restrict__ dy,  // (B, N, M) pooled grad (point-major)
    const T *__restrict__ wg, const int *__restrict__ idx,
    const unsigned char *__restrict__ am, const float *__restrict__ mean,
    const float *__restrict__ rstd, const float *__restrict__ gamma,
    const float *__restrict__ beta, float *__restrict__ scratch, long N,
    int K, int M, int G, float slope, const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int B = gridDim.z;
  const int tpc = M / 4;
  const int ppb = EG_THREADS / tpc;
  const int p_l = (int)threadIdx.x / tpc;
  const int c4 = (int)threadIdx.x % tpc;
  const bool active = p_l < ppb;
  const int Cg = M / G;
  if (ACT == 2) slope = *slope_ptr;

  extern __shared__ float sbins[];  // [G*2*B rows | M*2 chans | 1 slope]
  const int n_out = G * 2 * B + M * 2 + 1;
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += EG_THREADS)
    sbins[i] = 0.f;
  __syncthreads();

  const T *wgb = wg + (long)b * N * M;
  const int *idxb = idx + (long)b * N * K;
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
      const long pi = ((long)b * N + n) * M + c4 * 4;
      const uchar4 aq = *(const uchar4 *)(am + pi);
      const Quad<T> gq = *(const Quad<T> *)(dy + pi);
      const Quad<T> cq = *(const Quad<T> *)(wgb + n * M + c4 * 4);
      const int ks[4] = {aq.x, aq.y, aq.z, aq.w};
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int nb = idxb[n * K + ks[e]];
        const float v =
            ldg(wgb + (long)nb * M + c4 * 4 + e) - (float)cq.v[e];
        const float xhat = (v - m[e]) * r[e];
        float gv = (float)gq.v[e];
        if (ACT >= 1) {
          const float pre = xhat * ga[e] + be[e];
          if (ACT == 2 && pre <= 0.f) d_sl += gv * pre;
          gv = pre > 0.f ? gv : gv * slope;
        }
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
      atomicAdd(&sbins[G * 2 * B + c * 2 + 0], c_db[e]);
      atomicAdd(&sbins[G * 2 * B + c * 2 + 1], c_dg[e]);
    }
    if (ACT == 2) atomicAdd(&sbins[G * 2 * B + M * 2], d_sl);
  }
  __syncthreads();
  const long col = (long)blockIdx.x * B + b;
  const long stride = (long)gridDim.x * B;
  for (unsigned i = threadIdx.x; i < (unsigned)n_out; i += EG_THREADS)
    scratch[i * stride + col] = sbins[i];
}

// pass 2: dWgT (B, N, M), deterministic.  The centre term
// -sum_j dx1[c, j, p] closes over j analytically: only j = am carries dy
// (recovered from the saved pre-GN extreme vsel), and sum_j xhat_j
// derives from the per-point gather sum the forward reduce stored
// (vsum) -- so no K-loop and no idx reads here.  The incoming term walks
// the inverse-adjacency CSR (ordn/ordj side arrays), with dy a scalar
// load only on the (rare) argmax hit.
template <typename T, int ACT>
__global__ __launch_bounds__(EG_THREADS) void egnmp_bwd_apply_kernel(
    const T *__restrict__ dy, const T *__restrict__ wg,
    const unsigned char *__restrict__ am,
    const float *__restrict__ vsel,   // (B, N, M): chosen pre-GN extreme
    const float *__restrict__ vsum,   // (B, N, M): sum_j v per point
    const int *__restrict__ offsets,  // (B, N+1)
    const int *__restrict__ ordn,     // (B, K*N): source n per ordered edge
    const unsigned char *__restrict__ ordj,  // (B, K*N): its neighbour slot
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    const float *__restrict__ row_ws, T *__restrict__ dwg, long N, int K,
    int M, int G, long row_len, float slope,
    const float *__restrict__ slope_ptr) {
  const int b = blockIdx.z;
  const int tpc = M / 4;
  const int ppb = EG_THREADS / tpc;
  const int p_l = (int)threadIdx.x / tpc;
  const int c4 = (int)threadIdx.x % tpc;
  if (p_l >= ppb) return;
  const int Cg = M / G;
  const float inv_n = 1.0f / (float)row_len;
  float m[4], r[4], ga[4], be[4], s1[4], s2[4];
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    const int c = c4 * 4 + e;
    const int row = b * G + c / Cg;
    m[e] = mean[row];
    r[e] = rstd[row];
    ga[e] = gamma[c];
    be[e] = beta[c];
    s1[e] = row_ws[row * 2 + 0];
    s2[e] = row_ws[row * 2 + 1];
  }
  if (ACT == 2) slope = *slope_ptr;

  const T *wgb = wg + (long)b * N * M;
  const T *dyb = dy + (long)b * N * M;
  const unsigned char *amb = am + (long)b * N * M;
  const float *vselb = vsel + (long)b * N * M;
  const float *vsumb = vsum + (long)b * N * M;
  const int *ordnb = ordn + (long)b * N * K;
  const unsigned char *ordjb = ordj + (long)b * N * K;
  const int *offb = offsets + (long)b * (N + 1);

  for (long p = (long)blockIdx.x * ppb + p_l; p < N;
       p += (long)gridDim.x * ppb) {
    const Quad<T> pq = *(const Quad<T> *)(wgb + p * M + c4 * 4);
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    // centre term, closed over j:
    //   -sum_j (dxhat_j - (s1 + xhat_j*s2)*inv_n)*r
    //   = -dxhat_am*r + (K*s1 + s2*sum_j xhat_j)*inv_n*r,
    //   sum_j xhat_j = (vsum - K*mean)*rstd
    {
      const Quad<T> gq = *(const Quad<T> *)(dyb + p * M + c4 * 4);
      const float4 sv = *(const float4 *)(vselb + p * M + c4 * 4);
      const float4 vs = *(const float4 *)(vsumb + p * M + c4 * 4);
      const float sqs[4] = {sv.x, sv.y, sv.z, sv.w};
      const float vss[4] = {vs.x, vs.y, vs.z, vs.w};
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float xhat_am = (sqs[e] - m[e]) * r[e];
        float gs = (float)gq.v[e];
        if (ACT >= 1) {
          const float pre = xhat_am * ga[e] + be[e];
          gs = pre > 0.f ? gs : gs * slope;
        }
        const float sum_xhat = (vss[e] - (float)K * m[e]) * r[e];
        acc[e] = -gs * ga[e] * r[e] +
                 ((float)K * s1[e] + sum_xhat * s2[e]) * inv_n * r[e];
      }
    }
    // incoming term: edges whose neighbour is p.  The dy(argmax) part
    // fires iff this edge's slot j is the pooled argmax of (n, c) -- the
    // CSR side arrays (ordn, ordj) give (n, j) with no id decomposition,
    // and dy is a scalar load only on the (rare) argmax hit.  Exact under
    // duplicate idx entries: only the recorded slot fires.
    const int lo = offb[p], hi = offb[p + 1];
    for (int t = lo; t < hi; ++t) {
      const long n = ordnb[t];
      const int j = ordjb[t];
      const Quad<T> nq = *(const Quad<T> *)(wgb + n * M + c4 * 4);
      const uchar4 aq = *(const uchar4 *)(amb + n * M + c4 * 4);
      const int ks[4] = {aq.x, aq.y, aq.z, aq.w};
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float v = (float)pq.v[e] - (float)nq.v[e];
        const float xhat = (v - m[e]) * r[e];
        float dxhat = 0.f;
        if (j == ks[e]) {
          float gs = ldg(dyb + n * M + c4 * 4 + e);
          if (ACT >= 1) {
            const float pre = xhat * ga[e] + be[e];
            gs = pre > 0.f ? gs : gs * slope;
          }
          dxhat = gs * ga[e];
        }
        acc[e] += (dxhat - (s1[e] + xhat * s2[e]) * inv_n) * r[e];
      }
    }
    Quad<T> oq;
#pragma unroll
    for (int e = 0; e < 4; ++e) oq.v[e] = (T)acc[e];
    *(Quad<T> *)(dwg + ((long)b * N + p) * M + c4 * 4) = oq;
  }
}

// --------------------------------------------------------------- launchers

static int eg_chunks(long N, int ppb, int B, int cap) {
  long want = cap / (B > 0 ? B : 1);
  long blocks = (N + ppb - 1) / ppb;
  if (want > blocks) want = blocks;
  if (want < 1) want = 1;
  return (int)want;
}

template <typename T>
void egnmp_fwd_impl(const T *wg, const int *idx, float *scratch, float *ws,
                    float *mean, float *rstd, const float *gamma,
                    const float *beta, float *vmax, float *vmin,
                    unsigned char *amax,
                    unsigned char *amin, float *vsum, T *y,
                    unsigned char *am, float *vsel,
                    int B, long N, int K, int M, int G, float eps, int act,
                    float slope, const float *slope_ptr, int rchunks,
                    hipStream_t stream) {
  const dim3 rgrid(rchunks, 1, B);
  const int n_out_f = B * G * 2;
  hipLaunchKernelGGL(egnmp_fwd_reduce_kernel<T>, rgrid, dim3(EG_THREADS),
                     (size_t)n_out_f * sizeof(float), stream, wg, idx,
                     scratch, vmax, vmin, amax, amin, vsum, N, K, M, G);
  (void)ws;
  launch_gn_finalize_scratch(scratch, (long)rchunks * B, mean, rstd,
                             (long)(M / G) * K * N, B * G, eps, stream);
  const long total = (long)B * N * M;
  long pb = (total / 4 + EG_THREADS - 1) / EG_THREADS;
  if (pb > 2048) pb = 2048;
  if (pb < 1) pb = 1;
#define EG_FWD(A)                                                             \
  hipLaunchKernelGGL((egnmp_fwd_pick_kernel<T, A>), dim3((unsigned)pb),       \
                     dim3(EG_THREADS), 0, stream, vmax, vmin, amax, amin,     \
                     mean, rstd, gamma, beta, y, am, vsel, total,             \
                     N * M, M, G, slope, slope_ptr)
  if (act == 2) EG_FWD(2);
  else if (act == 1) EG_FWD(1);
  else EG_FWD(0);
#undef EG_FWD
}

template <typename T>
void egnmp_bwd_impl(const T *dy, const T *wg, const int *idx,
                    const unsigned char *am, const float *vsel,
                    const float *vsum, const int *offsets,
                    const int *ordn, const unsigned char *ordj,
                    const float *mean,
                    const float *rstd, const float *gamma, const float *beta,
                    float *scratch, float *ws, T *dwg, int B, long N, int K,
                    int M, int G, int act, float slope,
                    const float *slope_ptr, int rchunks,
                    hipStream_t stream) {
  const int tpc = M / 4;
  const int ppb = EG_THREADS / tpc;
  const dim3 rgrid(rchunks, 1, B);
  const dim3 agrid(eg_chunks(N, ppb, B, 2048), 1, B);
  const int n_out = B * G * 2 + M * 2 + 1;
  const size_t shmem = (size_t)n_out * sizeof(float);
  const long row_len = (long)(M / G) * K * N;
  const int waves_per_block = EG_THREADS / WAVE;
#define EG_BWD(A)                                                              \
  do {                                                                         \
    hipLaunchKernelGGL((egnmp_bwd_reduce_kernel<T, A>), rgrid,                 \
                       dim3(EG_THREADS), shmem, stream, dy, wg, idx, am, mean, \
                       rstd, gamma, beta, scratch, N, K, M, G, slope,          \
                       slope_ptr);                                             \
    hipLaunchKernelGGL(egnmp_sum_partials_kernel,                              \
                       dim3((n_out + waves_per_block - 1) / waves_per_block),  \
                       dim3(EG_THREADS), 0, stream, scratch, ws,               \
                       (long)rchunks * B, n_out);                              \
    hipLaunchKernelGGL((egnmp_bwd_apply_kernel<T, A>), agrid,                  \
                       dim3(EG_THREADS), 0, stream, dy, wg, am, vsel, vsum,    \
                       offsets, ordn, ordj, mean, rstd, gamma, beta, ws, dwg,  \
                       N, K, M, G, row_len, slope, slope_ptr);                 \
  } while (0)
  if (act == 2) EG_BWD(2);
  else if (act == 1) EG_BWD(1);
  else EG_BWD(0);
#undef EG_BWD
}

// Standalone pick launcher for sibling fused ops (csrc/knn_gnmp.hip):
// elementwise max(act(gn(vmax)), act(gn(vmin))) + argmax + selected
// pre-GN value, over any (B, N, M) extremes tensors.
void launch_gnmp_pick(const float *vmax, const float *vmin,
                      const unsigned char *amax, const unsigned char *amin,
                      const float *mean, const float *rstd,
                      const float *gamma, const float *beta, void *y,
                      unsigned char *am, float *vsel, long total, long NM,
                      int M, int G, int act, float slope,
                      const float *slope_ptr, bool bf16,
                      hipStream_t stream) {
  long pb = (total / 4 + EG_THREADS - 1) / EG_THREADS;
  if (pb > 2048) pb = 2048;
  if (pb < 1) pb = 1;
#define EG_PICK(T, A)                                                         \
  hipLaunchKernelGGL((egnmp_fwd_pick_kernel<T, A>), dim3((unsigned)pb),       \
                     dim3(EG_THREADS), 0, stream, vmax, vmin, amax, amin,     \
                     mean, rstd, gamma, beta, (T *)y, am, vsel, total, NM,    \
                     M, G, slope, slope_ptr)
  if (bf16) {
    if (act == 2) EG_PICK(__hip_bfloat16, 2);
    else if (act == 1) EG_PICK(__hip_bfloat16, 1);
    else EG_PICK(__hip_bfloat16, 0);
  } else {
    if (act == 2) EG_PICK(float, 2);
    else if (act == 1) EG_PICK(float, 1);
    else EG_PICK(float, 0);
  }
#undef EG_PICK
}

int egnmp_reduce_chunks(long N, int M, int B) {
  const int ppb = EG_THREADS / (M / 4);
  long want = 1024 / (B > 0 ? B : 1);
  long blocks = (N + ppb - 1) / ppb;
  if (want > blocks) want = blocks;
  if (want < 1) want = 1;
  return (int)want;
}

void launch_egnmp_fwd(const void *wg, const int *idx, float *scratch,
                      float *ws, float *mean, float *rstd, const float *gamma,
                      const float *beta, float *vmax, float *vmin,
                      unsigned char *amax, unsigned char *amin, float *vsum,
                      void *y, unsigned char *am, float *vsel, int B, long N,
                      int K, int M, int G, float eps, int act, float slope,
                      const float *slope_ptr, bool bf16, int rchunks,
                      hipStream_t stream) {
  if (bf16)
    egnmp_fwd_impl<__hip_bfloat16>((const __hip_bfloat16 *)wg, idx, scratch,
                                   ws, mean, rstd, gamma, beta,
                                   vmax, vmin, amax, amin, vsum,
                                   (__hip_bfloat16 *)y, am,
                                   vsel, B, N, K,
                                   M, G, eps, act, slope, slope_ptr, rchunks,
                                   stream);
  else
    egnmp_fwd_impl<float>((const float *)wg, idx, scratch, ws, mean, rstd,
                          gamma, beta, vmax, vmin, amax,
                          amin, vsum, (float *)y, am, vsel, B, N, K,
                          M, G, eps, act, slope, slope_ptr, rchunks, stream);
}

void launch_egnmp_bwd(const void *dy, const void *wg, const int *idx,
                      const unsigned char *am, const float *vsel,
                      const float *vsum, const int *offsets,
                      const int *ordn, const unsigned char *ordj,
                      const float *mean,
                      const float *rstd, const float *gamma,
                      const float *beta, float *scratch, float *ws,
                      void *dwg, int B, long N, int K, int M, int G, int act,
                      float slope, const float *slope_ptr, bool bf16,
                      int rchunks, hipStream_t stream) {
  if (bf16)
    egnmp_bwd_impl<__hip_bfloat16>(
        (const __hip_bfloat16 *)dy, (const __hip_bfloat16 *)wg, idx, am,
        vsel, vsum,
        offsets, ordn, ordj, mean, rstd, gamma, beta, scratch, ws,
        (__hip_bfloat16 *)dwg, B, N, K, M, G, act, slope, slope_ptr, rchunks,
        stream);
  else
    egnmp_bwd_impl<float>((const float *)dy, (const float *)wg, idx, am,
                          vsel, vsum,
                          offsets, ordn, ordj, mean, rstd, gamma, beta,
                          scratch, ws, (float *)dwg, B, N, K, M, G, act,
                          slope, slope_ptr, rchunks, stream);
}
