// K2+K3 fused: all-pair correlation GEMM with streaming top-K truncation
// (reference model/corr.py:95-99 computes the full B x N x M matrix and
// model/corr.py:37 topk's it; the round-1 GPU path still materialised
// fp32 row-chunks for a separate selection kernel).
//
// Here the (N x M) matrix NEVER exists.  bf16 MFMA GEMM
// (v_mfma_f32_16x16x32_bf16, fp32 accumulate, * 1/sqrt(C)) with a
// sample-threshold streaming selection:
//
//   kernel A (sample): each query row computes its correlation against a
//     1-in-ST column sample (<= 1024 columns) and extracts two exact
//     sample order statistics by wave extraction rounds:
//       P_hi = kp_hi-th largest sample value, P_lo = kp_lo-th largest.
//     An order statistic of rank r estimates the full count above it with
//     sd ~ st*sqrt(r), so the ranks put ~4 sd between the expected direct
//     count and K on one side and the expected coverage and K on the
//     other.  For ST == 1 the sample is the full row and the thresholds
//     are exact.
//   kernel B (sweep): full GEMM sweep; values > P_hi are emitted straight
//     into the output slots; values in [P_lo, P_hi] land in a per-row
//     band workspace; everything else is rejected with no atomics.
//     Sample and sweep values are BITWISE equal (identical MFMA fragment
//     and k-block order), so the threshold counts are exact -- and the
//     kernel CHECKS them: rows whose direct count exceeded K, whose
//     coverage missed K, or whose band overflowed get their thresholds
//     adjusted from the EXACT counts and are re-swept (<= 4 rounds; only
//     workgroups owning failed rows re-sweep).  After the loop only
//     massive value ties (degenerate inputs) can remain short, and any K
//     of tied values is a correct top-K SET -- remaining slots pad from
//     the band head, like the round-1 kernel.
//   kernel C (select): a wave per row extracts the remaining
//     K - count(>P_hi) largest entries from the band.
//
// The top-K SET is unordered (reference sorted=True is a torch.topk
// detail; every consumer is order-invariant).
//
// Inputs are point-major (B, N, C)/(B, M, C) bf16 (the LDS-tiled
// transpose of the encoder's (B, C, N) maps); C % 32 == 0, C <= 256.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define CT_THREADS 256
#define CT_SAMP 1024   // max sample columns per row
#define CT_CAP 704     // band capacity per row
#define CT_RWA 16      // rows per workgroup, sample kernel
#define CT_RWB 32      // rows per workgroup, sweep kernel
#define CT_CPAD 8      // bf16 padding per LDS f1 row

// ------------------------------------------------------------- kernel A

template <int KBN>  // C / 32
__global__ __launch_bounds__(CT_THREADS) void corr_sample_kernel(
    const __hip_bfloat16 *__restrict__ f1t,  // (B, N, C)
    const __hip_bfloat16 *__restrict__ f2t,  // (B, M, C)
    float *__restrict__ thr,                 // (B, N, 2) {P_hi, P_lo}
    int N, int M, int st, int ns, int kp_hi, int kp_lo, float scale) {
  constexpr int C = KBN * 32;
  __shared__ __hip_bfloat16 s_f1[CT_RWA][C + CT_CPAD];
  __shared__ float s_samp[CT_RWA][CT_SAMP + 1];

  const int b = blockIdx.z;
  const int n0 = blockIdx.x * CT_RWA;
  const int lane = lane_id();
  const int wv = wave_id();

  for (int i = threadIdx.x; i < CT_RWA * C / 8; i += CT_THREADS) {
    const int r = i / (C / 8);
    const int c8 = (i % (C / 8)) * 8;
    bf16x8 v = (bf16x8)(__bf16)0.0f;
    if (n0 + r < N)
      v = *(const bf16x8 *)(f1t + ((long)b * N + n0 + r) * C + c8);
    *(bf16x8 *)&s_f1[r][c8] = v;
  }
  __syncthreads();

  const int frow = lane & 15;
  const int koff = (lane >> 4) * 8;
  for (int cf = wv; cf * 16 < ns; cf += CT_THREADS / WAVE) {
    const int j = cf * 16 + frow;  // sample index (B-frag row)
    const long m = (long)j * st;
    const bool ok = j < ns && m < M;
    const __hip_bfloat16 *f2r = f2t + ((long)b * M + (ok ? m : 0)) * C;
    bf16x8 bf[KBN];
#pragma unroll
    for (int kb = 0; kb < KBN; ++kb)
      bf[kb] = ok ? *(const bf16x8 *)(f2r + kb * 32 + koff) : (bf16x8)(__bf16)0.0f;
    f32x4 acc = (f32x4)(0.f);
#pragma unroll
    for (int kb = 0; kb < KBN; ++kb) {
      const bf16x8 afrag = *(const bf16x8 *)&s_f1[frow][kb * 32 + koff];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bf[kb], acc, 0, 0, 0);
    }
    const int jj = cf * 16 + (lane & 15);
    if (jj < ns) {
      const bool valid = (long)jj * st < M;
#pragma unroll
      for (int e = 0; e < 4; ++e)
        s_samp[(lane >> 4) * 4 + e][jj] = valid ? acc[e] * scale : -INFINITY;
    }
  }
  __syncthreads();

  // per-row exact sample order statistics via radix rank select (O(32*S)
  // wave-ops; extraction rounds at rank ~100 measured ~25x slower)
  for (int r = wv; r < CT_RWA; r += CT_THREADS / WAVE) {
    const int n = n0 + r;
    if (n >= N) continue;
    unsigned kv[CT_SAMP / WAVE];
#pragma unroll
    for (int s = 0; s < CT_SAMP / WAVE; ++s) {
      const int p = lane + s * WAVE;
      kv[s] = p < ns ? fkey(s_samp[r][p]) : 0u;  // 0 = smallest key
    }
    const unsigned khi = wave_rank_key(kv, kp_hi);
    const unsigned klo = wave_rank_key(kv, kp_lo);
    if (lane == 0) {
      thr[((long)b * N + n) * 2 + 0] = fkey_inv(khi);
      thr[((long)b * N + n) * 2 + 1] = fkey_inv(klo);
    }
  }
}

// ------------------------------------------------------------- kernel B

template <int KBN>
__global__ __launch_bounds__(CT_THREADS) void corr_sweep_kernel(
    const __hip_bfloat16 *__restrict__ f1t, const __hip_bfloat16 *__restrict__ f2t,
    const float *__restrict__ thr,
    float *__restrict__ out_v,   // (B, N, K)
    int *__restrict__ out_i,     // (B, N, K)
    float *__restrict__ band_v,  // (B*N, CAP)
    int *__restrict__ band_i,    // (B*N, CAP)
    int *__restrict__ cnt,       // (B*N, 2) {n_hi (<=K), n_band (<=CAP)}
    int N, int M, int K, float scale) {
  constexpr int C = KBN * 32;
  __shared__ __hip_bfloat16 s_f1[CT_RWB][C + CT_CPAD];
  __shared__ float s_phi[CT_RWB], s_plo[CT_RWB];
  __shared__ unsigned s_hi[CT_RWB], s_bd[CT_RWB];
  __shared__ unsigned s_active;

  const int b = blockIdx.z;
  const int n0 = blockIdx.x * CT_RWB;
  const int lane = lane_id();
  const int wv = wave_id();

  for (int i = threadIdx.x; i < CT_RWB * C / 8; i += CT_THREADS) {
    const int r = i / (C / 8);
    const int c8 = (i % (C / 8)) * 8;
    bf16x8 v = (bf16x8)(__bf16)0.0f;
    if (n0 + r < N)
      v = *(const bf16x8 *)(f1t + ((long)b * N + n0 + r) * C + c8);
    *(bf16x8 *)&s_f1[r][c8] = v;
  }
  for (int r = threadIdx.x; r < CT_RWB; r += CT_THREADS) {
    const int n = n0 + r;
    s_phi[r] = n < N ? thr[((long)b * N + n) * 2 + 0] : INFINITY;
    s_plo[r] = n < N ? thr[((long)b * N + n) * 2 + 1] : INFINITY;
    s_hi[r] = 0;
    s_bd[r] = 0;
  }
  if (threadIdx.x == 0) s_active = (unsigned)-1;
  __syncthreads();

  const int frow = lane & 15;
  const int koff = (lane >> 4) * 8;
  const int nfrag = (M + 15) / 16;
  for (int iter = 0; iter < 4 && s_active != 0u; ++iter) {
    // register copies of this lane's row thresholds/active bits: the LDS
    // counter atomics below alias LDS for the compiler, which would
    // otherwise re-read s_phi/s_plo/s_active per emitted element
    float rphi[CT_RWB / 16][4], rplo[CT_RWB / 16][4];
    bool ract[CT_RWB / 16][4];
#pragma unroll
    for (int rf = 0; rf < CT_RWB / 16; ++rf)
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int r = rf * 16 + (lane >> 4) * 4 + e;
        rphi[rf][e] = s_phi[r];
        rplo[rf][e] = s_plo[r];
        ract[rf][e] = (s_active >> r & 1u) != 0u;
      }
    for (int cf = wv; cf < nfrag; cf += CT_THREADS / WAVE) {
      const int j = cf * 16 + frow;  // column (B-frag row)
      const bool ok = j < M;
      const __hip_bfloat16 *f2r = f2t + ((long)b * M + (ok ? j : 0)) * C;
      bf16x8 bf[KBN];
#pragma unroll
      for (int kb = 0; kb < KBN; ++kb)
        bf[kb] = ok ? *(const bf16x8 *)(f2r + kb * 32 + koff)
                    : (bf16x8)(__bf16)0.0f;
      f32x4 acc[CT_RWB / 16];
#pragma unroll
      for (int rf = 0; rf < CT_RWB / 16; ++rf) acc[rf] = (f32x4)(0.f);
#pragma unroll
      for (int kb = 0; kb < KBN; ++kb)
#pragma unroll
        for (int rf = 0; rf < CT_RWB / 16; ++rf) {
          const bf16x8 afrag =
              *(const bf16x8 *)&s_f1[rf * 16 + frow][kb * 32 + koff];
          acc[rf] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bf[kb], acc[rf], 0, 0, 0);
        }
      const int jj = cf * 16 + (lane & 15);
      if (jj < M) {
#pragma unroll
        for (int rf = 0; rf < CT_RWB / 16; ++rf)
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            const int r = rf * 16 + (lane >> 4) * 4 + e;
            const long n = n0 + r;
            if (n >= N || !ract[rf][e]) continue;
            const float v = acc[rf][e] * scale;
            if (v > rphi[rf][e]) {
              const unsigned slot = atomicAdd(&s_hi[r], 1u);
              if (slot < (unsigned)K) {
                out_v[((long)b * N + n) * K + slot] = v;
                out_i[((long)b * N + n) * K + slot] = jj;
              }
            } else if (v >= rplo[rf][e]) {
              const unsigned p = atomicAdd(&s_bd[r], 1u);
              if (p < CT_CAP) {
                band_v[((long)b * N + n) * CT_CAP + p] = v;
                band_i[((long)b * N + n) * CT_CAP + p] = jj;
              }
            }
          }
      }
    }
    __syncthreads();
    // exact-count check: adjust failed rows' thresholds and re-sweep them
    for (int r = threadIdx.x; r < CT_RWB; r += CT_THREADS) {
      if (!(s_active >> r & 1u)) continue;
      if (n0 + r >= N) {
        atomicAnd(&s_active, ~(1u << r));
        continue;
      }
      const unsigned chi = s_hi[r], cbd = s_bd[r];
      const bool over_hi = chi > (unsigned)K;
      const bool under = chi + min(cbd, (unsigned)CT_CAP) < (unsigned)K;
      const bool over_bd = cbd > (unsigned)CT_CAP && chi + cbd >= (unsigned)K;
      if ((!over_hi && !under && !over_bd) || iter == 3) {
        // done -- or out of retries (massive ties): KEEP the last sweep's
        // counters/buffers; kernel C pads what is short
        atomicAnd(&s_active, ~(1u << r));
        continue;
      }
      const float span = s_phi[r] - s_plo[r] + 1e-6f * (1.f + fabsf(s_phi[r]));
      if (over_hi) {
        // too many strict accepts: push P_hi above them, band takes over
        s_plo[r] = s_phi[r];
        s_phi[r] = s_phi[r] + span * 4.f;
      } else if (under) {
        // coverage short of K: band region grows downward
        s_phi[r] = s_plo[r];
        s_plo[r] = s_plo[r] - span * 4.f;
      } else {
        // band overflowed its buffer: shrink the band span by the COUNT
        // ratio (cbd is the true count), with slack for the next sweep
        const float f =
            fminf(1.f, (float)(K - (int)chi + 64) / (float)cbd * 1.25f);
        s_plo[r] = s_phi[r] - (s_phi[r] - s_plo[r]) * f;
      }
      s_hi[r] = 0;
      s_bd[r] = 0;
    }
    __syncthreads();
  }
  for (int r = threadIdx.x; r < CT_RWB; r += CT_THREADS) {
    const long n = n0 + r;
    if (n < N) {
      cnt[((long)b * N + n) * 2 + 0] = (int)min(s_hi[r], (unsigned)K);
      cnt[((long)b * N + n) * 2 + 1] = (int)min(s_bd[r], (unsigned)CT_CAP);
    }
  }
}

// ------------------------------------------------------------- kernel C

__global__ __launch_bounds__(CT_THREADS) void corr_band_select_kernel(
    const float *__restrict__ band_v, const int *__restrict__ band_i,
    const int *__restrict__ cnt, float *__restrict__ out_v,
    int *__restrict__ out_i, long R, int K) {
  const int lane = lane_id();
  const long row = (long)blockIdx.x * (CT_THREADS / WAVE) + wave_id();
  if (row >= R) return;
  const int c_hi = cnt[row * 2 + 0];
  const int L = cnt[row * 2 + 1];
  const int need = K - c_hi;
  if (need <= 0) return;
  const int take = need < L ? need : L;

  float dv[CT_CAP / WAVE];
  unsigned kv[CT_CAP / WAVE];
  int iv[CT_CAP / WAVE];
#pragma unroll
  for (int s = 0; s < CT_CAP / WAVE; ++s) {
    const int p = lane + s * WAVE;
    dv[s] = p < L ? band_v[row * CT_CAP + p] : 0.f;
    kv[s] = p < L ? fkey(dv[s]) : 0u;
    iv[s] = p < L ? band_i[row * CT_CAP + p] : 0;
  }
  // take-th largest key, then ballot-prefix emission: strict winners
  // first, then just enough threshold ties
  const unsigned kt = wave_rank_key(kv, take);
  int base = c_hi;
  for (int pass = 0; pass < 2; ++pass) {
#pragma unroll
    for (int s = 0; s < CT_CAP / WAVE; ++s) {
      const bool elig =
          (lane + s * WAVE) < L && (pass == 0 ? kv[s] > kt : kv[s] == kt);
      const unsigned long long m = __ballot(elig);
      if (elig) {
        const int slot = base + __popcll(m & ((1ull << lane) - 1ull));
        if (slot < c_hi + take) {
          out_v[row * K + slot] = dv[s];
          out_i[row * K + slot] = iv[s];
        }
      }
      base += __popcll(m);
      if (base >= c_hi + take && pass == 1) break;
    }
    if (base >= c_hi + take) break;
  }
  // degenerate tie overflow: pad from the band head
  if (lane == 0)
    for (int t = c_hi + take; t < K; ++t) {
      out_v[row * K + t] = L > 0 ? band_v[row * CT_CAP] : -INFINITY;
      out_i[row * K + t] = L > 0 ? band_i[row * CT_CAP] : 0;
    }
}

// ------------------------------------------------------------- launcher

void launch_corr_topk(const void *f1t, const void *f2t, float *thr,
                      float *out_v, int *out_i, float *band_v, int *band_i,
                      int *cnt, int B, int N, int M, int C, int K,
                      float scale, hipStream_t stream) {
  const int st = M > CT_SAMP ? (M + CT_SAMP - 1) / CT_SAMP : 1;
  const int ns = (M + st - 1) / st;
  int kp_hi, kp_lo;
  if (st == 1) {
    kp_hi = kp_lo = K;
  } else {
    // rank-r sample order statistic -> full-count estimate sd ~ st*sqrt(r):
    // put ~4 sd between E[direct] and K, and E[coverage] and K
    const double q = (double)K / st;
    double x = (-4.0 + sqrt(16.0 + 4.0 * q)) / 2.0;
    kp_hi = (int)(x * x);
    // coverage margin 3 sd (4 sd made E[band] ~ CAP-sized and the band-
    // overflow retry path hot); the exact-count retry covers the tail
    x = (3.0 + sqrt(9.0 + 4.0 * q)) / 2.0;
    kp_lo = (int)(x * x) + 1;
    if (kp_hi < 1) kp_hi = 1;
    if (kp_lo <= kp_hi) kp_lo = kp_hi + 1;
    if (kp_lo > ns) kp_lo = ns;
    if (kp_hi > kp_lo) kp_hi = kp_lo;
  }
  const dim3 ga((N + CT_RWA - 1) / CT_RWA, 1, B);
  const dim3 gb((N + CT_RWB - 1) / CT_RWB, 1, B);
  const long R = (long)B * N;
  const int wpb = CT_THREADS / WAVE;
#define CT_DISPATCH(KBN)                                                       \
  do {                                                                         \
    hipLaunchKernelGGL((corr_sample_kernel<KBN>), ga, dim3(CT_THREADS), 0,     \
                       stream, (const __hip_bfloat16 *)f1t,                    \
                       (const __hip_bfloat16 *)f2t, thr, N, M, st, ns, kp_hi,  \
                       kp_lo, scale);                                          \
    hipLaunchKernelGGL((corr_sweep_kernel<KBN>), gb, dim3(CT_THREADS), 0,      \
                       stream, (const __hip_bfloat16 *)f1t,                    \
                       (const __hip_bfloat16 *)f2t, thr, out_v, out_i, band_v, \
                       band_i, cnt, N, M, K, scale);                           \
  } while (0)
  switch (C / 32) {
    case 1: CT_DISPATCH(1); break;
    case 2: CT_DISPATCH(2); break;
    case 3: CT_DISPATCH(3); break;
    case 4: CT_DISPATCH(4); break;
    case 5: CT_DISPATCH(5); break;
    case 6: CT_DISPATCH(6); break;
    case 7: CT_DISPATCH(7); break;
    default: CT_DISPATCH(8); break;
  }
#undef CT_DISPATCH
  hipLaunchKernelGGL(corr_band_select_kernel,
                     dim3((unsigned)((R + wpb - 1) / wpb)), dim3(CT_THREADS),
                     0, stream, band_v, band_i, cnt, out_v, out_i, R, K);
}
