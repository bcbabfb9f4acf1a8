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

#define HIP_CHECK_LAST()                                          \
  do {                                                            \
    hipError_t e = hipGetLastError();                             \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",    \
                hipGetErrorString(e));                            \
  } while (0)
