// K1: fused kNN graph construction (capability of reference
// model/flot/graph.py:53-60, which materialises the full B x N x N distance
// matrix and argsorts it).
//
// Sample-threshold design (round 2; the round-1 histogram-select measured
// 454 us at B=2, N=8192 -- one LDS atomic per (candidate, query) with
// same-bin conflicts made it atomic-bound ~10x over its compute floor):
//
//   pass A: every query computes distances to a 1-in-ST sample of the
//           cloud (ST = N/1024, so <= 1024 samples); a wave extracts the
//           exact k'-th smallest sample distance (k' = ceil(2k/ST)) by
//           argmin rounds -> threshold tau.  E[|{d <= tau}|] = 2k over the
//           full cloud, so the collect buffer stays tiny.
//   pass B: one full sweep; candidates with d <= tau go to a per-query
//           (CAP=256) buffer -- ~2k LDS-atomic slot grabs per query
//           instead of N.  Everything else is rejected with ZERO atomics.
//   select: a wave extracts the exact k smallest from the buffer.
//
//   Underflow (buffer < k: ~1e-5 per query for k=32 by the Poisson tail)
//   widens tau 4x and re-sweeps the active queries; after 3 widenings the
//   remaining slots pad with the query itself (degenerate tie clouds --
//   same semantics as round 1; downstream consumers are set/order
//   invariant and edge features of self-pairs are zero).
//
//   For small clouds (ST == 1) the sample IS the cloud, tau is the exact
//   min(2k, N)-th distance, and no underflow is possible.
//
// Candidate tiles (TILE_PTS x 3 fp32) are staged in LDS and shared by
// QB=8 queries per workgroup; grid (ceil(N/QB), B) fills the chip.
//
// LDS is sized for 4 workgroups per CU (~38 KB of the 160 KB): the first
// cut staged the QB x SAMP sample-distance matrix in LDS too (33 KB),
// which capped residency at 2 WGs/CU and left the kernel latency-bound
// (PMC: SQ_WAIT_ANY ~8.6x SQ_BUSY_CYCLES) -- sample distances now go
// straight into the ranking wave's registers.
#include <hip/hip_runtime.h>
#include "common.h"

#define KNN_THREADS 256
#define TILE_PTS 1792
#define QB 8            // queries per workgroup
#define CAP 256         // collect-buffer capacity per query
#define SAMP 1024       // max sample size per query
#define KNN_MAXK 48     // model uses 32 (reference extractor.py:10)

__global__ __launch_bounds__(KNN_THREADS) void knn_select_kernel(
    const float *__restrict__ xyz,  // (B, N, 3)
    int *__restrict__ out_idx,      // (B, N, k)
    int N, int k) {
  __shared__ float s_tile[TILE_PTS * 3];
  __shared__ float s_bd[QB][CAP];
  __shared__ int s_bi[QB][CAP];
  __shared__ unsigned s_cnt[QB];  // collect counters
  __shared__ float s_tau[QB];
  __shared__ unsigned s_active;   // queries still needing a sweep
  __shared__ float s_q[QB][3];

  const int b = blockIdx.y;
  const int q0 = blockIdx.x * QB;
  const float *cloud = xyz + (long)b * N * 3;

  const int st = N > SAMP ? (N + SAMP - 1) / SAMP : 1;
  const int ns = (N + st - 1) / st;
  // E[collected] target: 2k exact for st==1; 4k for sampled thresholds (a
  // rank-r order statistic estimates the full count with sd ~ st*sqrt(r),
  // so aim well clear of both k and CAP; the exact-count retry loop below
  // repairs the tails)
  int kp = st == 1 ? 2 * k : (4 * k + st - 1) / st;
  if (kp > ns) kp = ns;
  if (kp < 1) kp = 1;

  if (threadIdx.x < QB) {
    const int q = q0 + threadIdx.x;
    const int qq = min(q, N - 1);
    s_q[threadIdx.x][0] = cloud[qq * 3 + 0];
    s_q[threadIdx.x][1] = cloud[qq * 3 + 1];
    s_q[threadIdx.x][2] = cloud[qq * 3 + 2];
    s_cnt[threadIdx.x] = 0;
  }
  if (threadIdx.x == 0) s_active = (1u << QB) - 1;
  __syncthreads();

  // register copies of the query coordinates: the LDS atomics below alias
  // LDS in the compiler's view, so s_q reads inside the hot loops would
  // otherwise be re-issued per (candidate, query) -- measured ~4x VALU/LDS
  // instruction bloat
  float qx[QB], qy[QB], qz[QB];
#pragma unroll
  for (int qi = 0; qi < QB; ++qi) {
    qx[qi] = s_q[qi][0];
    qy[qi] = s_q[qi][1];
    qz[qi] = s_q[qi][2];
  }

  // ---- pass A + exact k'-th smallest, fused per wave: the ranking
  // wave computes its queries' sample distances straight into the radix
  // key registers (p < ns implies p*st < N).  Query coords come from
  // s_q, not the qx registers: qi is a runtime index here and a
  // dynamically-indexed per-lane array would spill ALL of qx/qy/qz to
  // scratch.
  {
    const int lane = lane_id();
    for (int qi = wave_id(); qi < QB; qi += KNN_THREADS / WAVE) {
      const float sx = s_q[qi][0], sy = s_q[qi][1], sz = s_q[qi][2];
      unsigned kv[SAMP / WAVE];
#pragma unroll
      for (int s = 0; s < SAMP / WAVE; ++s) {
        const int p = lane + s * WAVE;
        unsigned key = 0u;
        if (p < ns) {
          const long c = (long)p * st;
          const float dx = cloud[c * 3 + 0] - sx;
          const float dy = cloud[c * 3 + 1] - sy;
          const float dz = cloud[c * 3 + 2] - sz;
          key = ~fkey(dx * dx + dy * dy + dz * dz);
        }
        kv[s] = key;
      }
      const unsigned kt = wave_rank_key(kv, kp);
      if (lane == 0) s_tau[qi] = fkey_inv(~kt);
    }
  }
  __syncthreads();

  // ---- pass B: collect d <= tau; retarget tau from the EXACT count on
  // underflow (< k) or buffer overflow (> CAP) and re-sweep
  for (int iter = 0; iter < 4 && s_active != 0; ++iter) {
    const unsigned active = s_active;
    float tau[QB];
#pragma unroll
    for (int qi = 0; qi < QB; ++qi)
      tau[qi] = (active >> qi & 1u) ? s_tau[qi] : -1.f;  // -1: accept none
    for (int t0 = 0; t0 < N; t0 += TILE_PTS) {
      const int tn = min(TILE_PTS, N - t0);
      __syncthreads();
      for (int i = threadIdx.x; i < tn * 3; i += KNN_THREADS)
        s_tile[i] = cloud[(long)t0 * 3 + i];
      __syncthreads();
      for (int c = threadIdx.x; c < tn; c += KNN_THREADS) {
        const float cx = s_tile[c * 3 + 0];
        const float cy = s_tile[c * 3 + 1];
        const float cz = s_tile[c * 3 + 2];
#pragma unroll
        for (int qi = 0; qi < QB; ++qi) {
          const float dx = cx - qx[qi];
          const float dy = cy - qy[qi];
          const float dz = cz - qz[qi];
          const float d = dx * dx + dy * dy + dz * dz;
          if (d <= tau[qi]) {
            const unsigned p = atomicAdd(&s_cnt[qi], 1u);
            if (p < CAP) {
              s_bd[qi][p] = d;
              s_bi[qi][p] = t0 + c;
            }
          }
        }
      }
    }
    __syncthreads();
    if (threadIdx.x < QB) {
      const int qi = threadIdx.x;
      if (active >> qi & 1u) {
        const unsigned c = s_cnt[qi];
        if ((c >= (unsigned)k && c <= (unsigned)CAP) || q0 + qi >= N ||
            (iter == 3)) {
          atomicAnd(&s_active, ~(1u << qi));
        } else {
          // retarget E[count] to 4k using count ~ tau^1.5 (3-D volume);
          // +epsilon escapes tau == 0 (coincident points)
          const float ratio = (4.f * k) / (float)max(c, 2u);
          s_tau[qi] = s_tau[qi] * __powf(ratio, 0.6667f) + 1e-30f;
          s_cnt[qi] = 0;
        }
      }
    }
    __syncthreads();
  }

  // ---- final: exact k smallest from each query's buffer
  const int lane = lane_id();
  for (int qi = wave_id(); qi < QB; qi += KNN_THREADS / WAVE) {
    const int q = q0 + qi;
    if (q >= N) continue;
    const int L = (int)min(s_cnt[qi], (unsigned)CAP);
    const int take = k < L ? k : L;
    float dv[CAP / WAVE];
    int iv[CAP / WAVE];
#pragma unroll
    for (int s = 0; s < CAP / WAVE; ++s) {
      const int p = lane + s * WAVE;
      dv[s] = p < L ? s_bd[qi][p] : INFINITY;
      iv[s] = p < L ? s_bi[qi][p] : q;
    }
    for (int r = 0; r < take; ++r) {
      int pay;
      wave_extract_min(dv, iv, pay);
      if (lane == 0) out_idx[((long)b * N + q) * k + r] = pay;
    }
    // degenerate clouds (ties beyond CAP / still-short buffer): pad with
    // the query itself -- a valid neighbour with zero edge features
    if (lane == 0)
      for (int r = take; r < k; ++r)
        out_idx[((long)b * N + q) * k + r] = q;
  }
}

void launch_knn_graph(const float *xyz, int *out_idx, int B, int N, int k,
                      hipStream_t stream) {
  dim3 grid((N + QB - 1) / QB, B);
  hipLaunchKernelGGL(knn_select_kernel, grid, dim3(KNN_THREADS), 0, stream,
                     xyz, out_idx, N, k);
}

// 30-bit Morton (Z-order) keys for spatial point relabeling: gathers all
// over the model (SetConv neighbour rows, correlation lookups) touch
// RANDOM point ids on unordered clouds, so every 8 B quad pulls its own
// cacheline.  Sorting points along a space-filling curve once per pair
// makes kNN neighbourhoods id-local, and the gather kernels L2/L1-hit.
__device__ __forceinline__ unsigned mg_part(unsigned v) {
  v &= 0x3ffu;
  v = (v | (v << 16)) & 0x030000FFu;
  v = (v | (v << 8)) & 0x0300F00Fu;
  v = (v | (v << 4)) & 0x030C30C3u;
  v = (v | (v << 2)) & 0x09249249u;
  return v;
}

__global__ void morton_keys_kernel(const float *__restrict__ xyz,
                                   const float *__restrict__ mn,   // (B, 3)
                                   const float *__restrict__ inv_ext,
                                   long *__restrict__ keys, long N) {
  const int b = blockIdx.y;
  const float mx0 = mn[b * 3 + 0], mx1 = mn[b * 3 + 1], mx2 = mn[b * 3 + 2];
  const float e0 = inv_ext[b * 3 + 0], e1 = inv_ext[b * 3 + 1],
              e2 = inv_ext[b * 3 + 2];
  for (long n = (long)blockIdx.x * blockDim.x + threadIdx.x; n < N;
       n += (long)gridDim.x * blockDim.x) {
    const float *p = xyz + ((long)b * N + n) * 3;
    unsigned q0 = (unsigned)fminf(fmaxf((p[0] - mx0) * e0, 0.f), 1023.f);
    unsigned q1 = (unsigned)fminf(fmaxf((p[1] - mx1) * e1, 0.f), 1023.f);
    unsigned q2 = (unsigned)fminf(fmaxf((p[2] - mx2) * e2, 0.f), 1023.f);
    keys[(long)b * N + n] =
        (long)((mg_part(q2) << 2) | (mg_part(q1) << 1) | mg_part(q0));
  }
}

void launch_morton_keys(const float *xyz, const float *mn,
                        const float *inv_ext, long *keys, int B, long N,
                        hipStream_t stream) {
  long blocks = (N + 255) / 256;
  if (blocks > 1024) blocks = 1024;
  hipLaunchKernelGGL(morton_keys_kernel, dim3((unsigned)blocks, B),
                     dim3(256), 0, stream, xyz, mn, inv_ext, keys, N);
}
