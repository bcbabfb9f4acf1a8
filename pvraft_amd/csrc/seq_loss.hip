// K10: fused gamma-weighted sequence loss (reference tools/loss.py:4-13).
//
// loss = sum_t gamma^(T-1-t) * masked-mean-L1(flow_t - gt)
//
// The eager formulation runs ~5 kernels per flow forward (+~6 backward)
// x 8 GRU iterations per step.  Here: ONE reduction pass over all T flows
// (their pointers travel by value in the kernel-argument block, so the
// launch works identically in eager mode and inside a hipGraph capture),
// a 1-thread finalize, and ONE backward kernel writing all T gradients.
//
// The (T totals + mask count) workspace is persistent and self-cleaning:
// finalize re-zeroes it after consuming (same pattern as the GroupNorm
// forward workspace -- calls are stream-ordered, replays included).
#include <hip/hip_runtime.h>
#include "common.h"

#define SL_MAXT 32
#define SL_THREADS 256

struct SlPtrs {
  const float *p[SL_MAXT];
};
struct SlGradPtrs {
  float *p[SL_MAXT];
};

namespace {

DEV_INLINE float block_sum_sl(float v, float *sh) {
  v = wave_sum(v);
  if (lane_id() == 0) sh[wave_id()] = v;
  __syncthreads();
  float t = 0.f;
  if (threadIdx.x < SL_THREADS / WAVE) t = sh[threadIdx.x];
  __syncthreads();
  return wave_sum(t);
}

}  // namespace

// ws layout: [0..T) per-flow masked |err| totals, [T] mask count
__global__ __launch_bounds__(SL_THREADS) void seq_loss_fwd_kernel(
    SlPtrs flows, const float *__restrict__ gt, const float *__restrict__ mask,
    float *__restrict__ ws, long BN, int T, int mask_stride) {
  __shared__ float sh[SL_THREADS / WAVE];
  float acc[SL_MAXT];
#pragma unroll
  for (int t = 0; t < SL_MAXT; ++t) acc[t] = 0.f;
  float cnt = 0.f;
  for (long i = (long)blockIdx.x * SL_THREADS + threadIdx.x; i < BN;
       i += (long)gridDim.x * SL_THREADS) {
    const float m = mask[i * mask_stride] > 0.f ? 1.f : 0.f;
    cnt += m;
    const float gx = gt[i * 3 + 0], gy = gt[i * 3 + 1], gz = gt[i * 3 + 2];
    for (int t = 0; t < T; ++t) {
      const float *f = flows.p[t];
      acc[t] += m * (fabsf(f[i * 3 + 0] - gx) + fabsf(f[i * 3 + 1] - gy) +
                     fabsf(f[i * 3 + 2] - gz));
    }
  }
  for (int t = 0; t < T; ++t) {
    const float s = block_sum_sl(acc[t], sh);
    if (threadIdx.x == 0) atomicAdd(&ws[t], s);
  }
  const float c = block_sum_sl(cnt, sh);
  if (threadIdx.x == 0) atomicAdd(&ws[SL_MAXT], c);
}

// 1 thread: loss scalar + count out; re-zero the workspace
__global__ void seq_loss_finalize_kernel(float *__restrict__ ws,
                                         float *__restrict__ loss,
                                         float *__restrict__ count_out, int T,
                                         float gamma) {
  float cnt = ws[SL_MAXT];
  float denom = fmaxf(cnt * 3.f, 1.f);
  float l = 0.f;
  for (int t = 0; t < T; ++t) {
    l += powf(gamma, (float)(T - 1 - t)) * ws[t] / denom;
    ws[t] = 0.f;
  }
  ws[SL_MAXT] = 0.f;
  *loss = l;
  *count_out = denom;
}

// dflow_t = dloss * gamma^(T-1-t) * m * sign(flow_t - gt) / denom
__global__ __launch_bounds__(SL_THREADS) void seq_loss_bwd_kernel(
    SlPtrs flows, SlGradPtrs grads, const float *__restrict__ gt,
    const float *__restrict__ mask, const float *__restrict__ dloss,
    const float *__restrict__ denom, long BN, int T, int mask_stride,
    float gamma) {
  const float g0 = *dloss / *denom;
  float w[SL_MAXT];
  for (int t = 0; t < T; ++t) w[t] = g0 * powf(gamma, (float)(T - 1 - t));
  for (long i = (long)blockIdx.x * SL_THREADS + threadIdx.x; i < BN;
       i += (long)gridDim.x * SL_THREADS) {
    const float m = mask[i * mask_stride] > 0.f ? 1.f : 0.f;
    const float gx = gt[i * 3 + 0], gy = gt[i * 3 + 1], gz = gt[i * 3 + 2];
    for (int t = 0; t < T; ++t) {
      const float *f = flows.p[t];
      float *d = grads.p[t];
      const float s = m * w[t];
      const float ex = f[i * 3 + 0] - gx, ey = f[i * 3 + 1] - gy,
                  ez = f[i * 3 + 2] - gz;
      d[i * 3 + 0] = s * (ex > 0.f ? 1.f : (ex < 0.f ? -1.f : 0.f));
      d[i * 3 + 1] = s * (ey > 0.f ? 1.f : (ey < 0.f ? -1.f : 0.f));
      d[i * 3 + 2] = s * (ez > 0.f ? 1.f : (ez < 0.f ? -1.f : 0.f));
    }
  }
}

static inline int sl_blocks(long BN) {
  long b = (BN + SL_THREADS - 1) / SL_THREADS;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

void launch_seq_loss_fwd(const float *const *flow_ptrs, const float *gt,
                         const float *mask, float *ws, float *loss,
                         float *count_out, long BN, int T, int mask_stride,
                         float gamma, hipStream_t stream) {
  SlPtrs fp{};
  for (int t = 0; t < T; ++t) fp.p[t] = flow_ptrs[t];
  hipLaunchKernelGGL(seq_loss_fwd_kernel, dim3(sl_blocks(BN)),
                     dim3(SL_THREADS), 0, stream, fp, gt, mask, ws, BN, T,
                     mask_stride);
  hipLaunchKernelGGL(seq_loss_finalize_kernel, dim3(1), dim3(1), 0, stream,
                     ws, loss, count_out, T, gamma);
}

void launch_seq_loss_bwd(const float *const *flow_ptrs, float *const *grad_ptrs,
                         const float *gt, const float *mask,
                         const float *dloss, const float *denom, long BN,
                         int T, int mask_stride, float gamma,
                         hipStream_t stream) {
  SlPtrs fp{};
  SlGradPtrs gp{};
  for (int t = 0; t < T; ++t) {
    fp.p[t] = flow_ptrs[t];
    gp.p[t] = grad_ptrs[t];
  }
  hipLaunchKernelGGL(seq_loss_bwd_kernel, dim3(sl_blocks(BN)),
                     dim3(SL_THREADS), 0, stream, fp, gp, gt, mask, dloss,
                     denom, BN, T, mask_stride, gamma);
}
