// Shared helpers for the pvraft_amd CDNA4 (gfx950) kernels.
//
// Conventions: wave64 (CDNA wavefront), block = 256 threads = 4 waves
// unless stated otherwise.  All kernels are written directly for CDNA4 --
// no CUDA-compat shims, no warp32 assumptions.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

DEV_INLINE int lane_id() { return threadIdx.x & (WAVE - 1); }
DEV_INLINE int wave_id() { return threadIdx.x >> 6; }

// Butterfly sum across the 64 lanes of a wave; every lane ends with the sum.
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// Butterfly min with index across the wave: (val, idx) -> every lane holds
// the minimum value and the smallest index attaining it.
DEV_INLINE void wave_argmin(float &val, int &idx) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(val, off, WAVE);
    int oi = __shfl_xor(idx, off, WAVE);
    if (ov < val || (ov == val && oi < idx)) {
      val = ov;
      idx = oi;
    }
  }
}

DEV_INLINE int wave_sum_int(int v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// order-preserving float -> unsigned key: a < b  <=>  fkey(a) < fkey(b)
DEV_INLINE unsigned fkey(float v) {
  unsigned b = __float_as_uint(v);
  return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}
DEV_INLINE float fkey_inv(unsigned k) {
  return __uint_as_float((k & 0x80000000u) ? (k ^ 0x80000000u) : ~k);
}

// Exact r-th LARGEST key among the wave's register slots, by 32-step
// radix descent with ballot counting: O(32 * S) wave-ops, vs the
// O(r * S) of repeated argmax extraction (which measured ~50x slower at
// r ~ 200).  Returns K* such that count(key >= K*) >= r and
// count(key > K*) < r; K* is the key of an actual element.
template <int S>
DEV_INLINE unsigned wave_rank_key(const unsigned (&kv)[S], int r) {
  unsigned cur = 0;
#pragma unroll
  for (int bit = 31; bit >= 0; --bit) {
    const unsigned cand = cur | (1u << bit);
    int c = 0;
#pragma unroll
    for (int s = 0; s < S; ++s) c += kv[s] >= cand;
    if (wave_sum_int(c) >= r) cur = cand;
  }
  return cur;
}

// One extraction round of a wave-cooperative k-smallest selection over
// per-lane register slots.  CRITICAL: every slot access is statically
// indexed (unrolled compare/select) -- a single dynamically-indexed
// access would spill the whole array to scratch (global memory), which
// measured ~40x slower on these kernels.  Returns the minimum value
// across all lanes' live slots (INFINITY when exhausted), broadcasts its
// payload int to every lane, and kills the winning slot.
template <int S>
DEV_INLINE float wave_extract_min(float (&dv)[S], const int (&iv)[S],
                                  int &out_payload) {
  float best = INFINITY;
  int bslot = 0;
#pragma unroll
  for (int s = 0; s < S; ++s)
    if (dv[s] < best) {
      best = dv[s];
      bslot = s;
    }
  int bi = lane_id() | (bslot << 6);
  if (best == INFINITY) bi = 0x7fffffff;
  float bv = best;
  wave_argmin(bv, bi);
  const int wl = bi & 63;
  const int ws = bi >> 6;
  int pay = 0;
#pragma unroll
  for (int s = 0; s < S; ++s)
    if (s == ws) pay = iv[s];  // ws is wave-uniform; static select
  out_payload = __shfl(pay, wl, WAVE);
  if (bi != 0x7fffffff && wl == lane_id()) {
#pragma unroll
    for (int s = 0; s < S; ++s)
      if (s == ws) dv[s] = INFINITY;
  }
  return bv;
}

#define HIP_CHECK_LAST()                                          \
  do {                                                            \
    hipError_t e = hipGetLastError();                             \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",    \
                hipGetErrorString(e));                            \
  } while (0)
