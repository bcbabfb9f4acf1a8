// K1: fused kNN graph construction (capability of reference
// model/flot/graph.py:53-60, which materialises the full B x N x N distance
// matrix and argsorts it).
//
// Histogram-select design (the insert-heap variants measured ~2.3 ms for
// B=2, N=8192: the first ~2k candidates of every query force a heap insert
// in SOME lane, serialising whole waves).  Distances are non-negative
// floats, so their bit patterns order monotonically:
//
//   pass 1: histogram each query's N squared distances into 256 bins by
//           float-exponent byte (bits >> 23); prefix-scan to find the
//           threshold bin T where the k-th smallest falls;
//   pass 1.5 (rare): if bin T is overfull (> CAP), refine it by the next 8
//           mantissa bits;
//   pass 2: candidates strictly below the threshold are accepted directly
//           (slot via LDS counter); candidates AT the threshold go to a
//           small boundary buffer; a per-query wave argmin selects the
//           remaining slots from it.
//
// No per-candidate branching storms: every pass is straight-line math +
// one LDS atomic.  Candidate tiles (TILE_PTS x 3 fp32) are staged in LDS
// and shared by QB=8 queries per workgroup; grid (ceil(N/QB), B) fills the
// chip.  Neighbour order within a query is arbitrary (downstream max-pools
// are order-invariant; reference relies on the set only).
#include <hip/hip_runtime.h>
#include "common.h"

#define KNN_THREADS 256
#define TILE_PTS 2048
#define QB 8          // queries per workgroup
#define CAP 256       // boundary-buffer capacity per query
#define KNN_MAXK 48   // model uses 32 (reference extractor.py:10)

DEV_INLINE unsigned dist_bits(float d) {
  return __float_as_uint(d);  // d >= 0 -> monotonic
}

__global__ __launch_bounds__(KNN_THREADS) void knn_select_kernel(
    const float *__restrict__ xyz,  // (B, N, 3)
    int *__restrict__ out_idx,      // (B, N, k)
    int N, int k) {
  __shared__ float s_tile[TILE_PTS * 3];
  __shared__ unsigned s_hist[QB][256];
  __shared__ float s_bd[QB][CAP];
  __shared__ int s_bi[QB][CAP];
  __shared__ unsigned s_acc[QB];   // accepted-slot counters
  __shared__ unsigned s_bcnt[QB];  // boundary counters
  __shared__ unsigned s_thr[QB];   // threshold bin T
  __shared__ unsigned s_sub[QB];   // refined sub-bin (0xffffffff = no refine)
  __shared__ int s_need[QB];       // slots to fill from the boundary buffer
  __shared__ float s_q[QB][3];

  const int b = blockIdx.y;
  const int q0 = blockIdx.x * QB;
  const float *cloud = xyz + (long)b * N * 3;

  if (threadIdx.x < QB) {
    const int q = q0 + threadIdx.x;
    const int qq = min(q, N - 1);
    s_q[threadIdx.x][0] = cloud[qq * 3 + 0];
    s_q[threadIdx.x][1] = cloud[qq * 3 + 1];
    s_q[threadIdx.x][2] = cloud[qq * 3 + 2];
    s_acc[threadIdx.x] = 0;
    s_bcnt[threadIdx.x] = 0;
    s_sub[threadIdx.x] = 0xffffffffu;
  }
  for (int i = threadIdx.x; i < QB * 256; i += KNN_THREADS)
    ((unsigned *)s_hist)[i] = 0;
  __syncthreads();

  // ---- pass 1: exponent-byte histogram
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
        const float dx = cx - s_q[qi][0];
        const float dy = cy - s_q[qi][1];
        const float dz = cz - s_q[qi][2];
        const float d = dx * dx + dy * dy + dz * dz;
        atomicAdd(&s_hist[qi][dist_bits(d) >> 23], 1u);
      }
    }
  }
  __syncthreads();

  // ---- threshold scan (thread qi walks its query's 256 bins)
  if (threadIdx.x < QB) {
    const int qi = threadIdx.x;
    unsigned cum = 0, T = 255;
    for (int bin = 0; bin < 256; ++bin) {
      const unsigned nxt = cum + s_hist[qi][bin];
      if (nxt >= (unsigned)k) {
        T = bin;
        break;
      }
      cum = nxt;
    }
    s_thr[qi] = T;
    s_acc[qi] = 0;
    s_need[qi] = k - (int)cum;  // slots to take from bin T
    // overfull threshold bin -> refine by the next 8 bits below the exponent
    if (s_hist[qi][T] > CAP - 8) s_sub[qi] = 0;  // mark: refine needed
  }
  __syncthreads();

  bool any_refine = false;
  for (int qi = 0; qi < QB; ++qi) any_refine |= (s_sub[qi] == 0u);
  if (any_refine) {
    // reuse the histograms for the sub-bins of each query's threshold bin
    for (int i = threadIdx.x; i < QB * 256; i += KNN_THREADS)
      ((unsigned *)s_hist)[i] = (((unsigned *)s_hist)[i] & 0u);
    __syncthreads();
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
          if (s_sub[qi] != 0u && s_sub[qi] != 0xfffffffeu) continue;
          const float dx = cx - s_q[qi][0];
          const float dy = cy - s_q[qi][1];
          const float dz = cz - s_q[qi][2];
          const float d = dx * dx + dy * dy + dz * dz;
          const unsigned bits = dist_bits(d);
          if ((bits >> 23) == s_thr[qi])
            atomicAdd(&s_hist[qi][(bits >> 15) & 0xff], 1u);
        }
      }
    }
    __syncthreads();
    if (threadIdx.x < QB && s_sub[threadIdx.x] == 0u) {
      const int qi = threadIdx.x;
      unsigned cum = 0, T2 = 255;
      const unsigned need = (unsigned)s_need[qi];
      for (int bin = 0; bin < 256; ++bin) {
        const unsigned nxt = cum + s_hist[qi][bin];
        if (nxt >= need) {
          T2 = bin;
          break;
        }
        cum = nxt;
      }
      s_sub[qi] = T2;
      s_need[qi] = (int)(need - cum);
    }
    __syncthreads();
  }

  // ---- pass 2: collect
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
        const int q = q0 + qi;
        if (q >= N) continue;
        const float dx = cx - s_q[qi][0];
        const float dy = cy - s_q[qi][1];
        const float dz = cz - s_q[qi][2];
        const float d = dx * dx + dy * dy + dz * dz;
        const unsigned bits = dist_bits(d);
        const unsigned bin = bits >> 23;
        const unsigned T = s_thr[qi];
        bool accept, boundary;
        if (s_sub[qi] == 0xffffffffu) {
          accept = bin < T;
          boundary = bin == T;
        } else {  // refined: threshold cut inside bin T at sub-bin T2
          const unsigned sub = (bits >> 15) & 0xff;
          accept = bin < T || (bin == T && sub < s_sub[qi]);
          boundary = bin == T && sub == s_sub[qi];
        }
        if (accept) {
          const unsigned slot = atomicAdd(&s_acc[qi], 1u);
          out_idx[((long)b * N + q) * k + slot] = t0 + c;
        } else if (boundary) {
          const unsigned p = atomicAdd(&s_bcnt[qi], 1u);
          if (p < CAP) {
            s_bd[qi][p] = d;
            s_bi[qi][p] = t0 + c;
          }
        }
      }
    }
  }
  __syncthreads();

  // ---- final: wave argmin rounds over each query's boundary buffer
  // waves take queries round-robin (4 waves, 8 queries)
  const int lane = lane_id();
  for (int qi = wave_id(); qi < QB; qi += KNN_THREADS / WAVE) {
    const int q = q0 + qi;
    if (q >= N) continue;
    const int L = (int)min(s_bcnt[qi], (unsigned)CAP);
    int need = s_need[qi];
    if (need > L) need = L;  // degenerate overflow: ties beyond CAP dropped
    int base = (int)s_acc[qi];
    // lanes own entries lane, lane+64, ... (CAP/WAVE = 4 slots max)
    float dv[CAP / WAVE];
    int iv[CAP / WAVE];
#pragma unroll
    for (int s = 0; s < CAP / WAVE; ++s) {
      const int p = lane + s * WAVE;
      dv[s] = p < L ? s_bd[qi][p] : INFINITY;
      iv[s] = p < L ? s_bi[qi][p] : 0x7fffffff;
    }
    int written = 0;
    for (int r = 0; r < need; ++r) {
      float best = INFINITY;
      int bslot = 0;
#pragma unroll
      for (int s = 0; s < CAP / WAVE; ++s)
        if (dv[s] < best) {
          best = dv[s];
          bslot = s;
        }
      int bidx = lane + bslot * WAVE;
      if (best == INFINITY) bidx = 0x7fffffff;
      float bv = best;
      wave_argmin(bv, bidx);
      if (bidx != 0x7fffffff && (bidx % WAVE) == lane) {
        const int s = bidx / WAVE;
        out_idx[((long)b * N + q) * k + base + r] = iv[s];
#pragma unroll
        for (int ss = 0; ss < CAP / WAVE; ++ss)
          if (ss == s) dv[ss] = INFINITY;
      }
      if (bidx != 0x7fffffff) ++written;
    }
    // degenerate overflow (ties beyond CAP): pad remaining slots with the
    // query itself (a valid neighbour; edge features become zero)
    if (lane == 0)
      for (int r = base + written; r < k; ++r)
        out_idx[((long)b * N + q) * k + r] = q;
  }
}

void launch_knn_graph(const float *xyz, int *out_idx, int B, int N, int k,
                      hipStream_t stream) {
  dim3 grid((N + QB - 1) / QB, B);
  hipLaunchKernelGGL(knn_select_kernel, grid, dim3(KNN_THREADS), 0, stream,
                     xyz, out_idx, N, k);
}
