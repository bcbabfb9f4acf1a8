// Fused ConvGRU gate math (reference model/update.py:31-40).
//
// The GRU's elementwise soup -- sigmoid over the stacked z/r preactivations,
// r*h, tanh(q), and the (1-z)h + zq blend, each with its backward -- runs
// ~20 eager kernels per GRU iteration (x8 iterations per step).  The two
// gate GEMMs must stay separate (q's GEMM consumes r*h), so the fusion
// splits at the GEMM boundary into two elementwise kernels per direction:
//
//   zr  : pre_zr (B,2H,N), h (B,H,N) -> z, r, rh            (before q GEMM)
//   q   : pre_q  (B,H,N),  z, h      -> q, hnew             (after  q GEMM)
//
// All math in f32 regardless of storage type (bf16 under autocast); loads
// and stores are 16-byte vectors when the H*N plane allows it.  Pure
// elementwise + grid-stride, so the kernels are trivially hipGraph-safe.
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace {

template <typename T, int V>
struct alignas(sizeof(T) * V) Vec {
  T v[V];
};

template <typename T>
DEV_INLINE float to_f32(T x) {
  return static_cast<float>(x);
}
template <>
DEV_INLINE float to_f32<__hip_bfloat16>(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
template <typename T>
DEV_INLINE T from_f32(float x) {
  return static_cast<T>(x);
}
template <>
DEV_INLINE __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}

DEV_INLINE float sigmoidf_(float x) { return 1.0f / (1.0f + __expf(-x)); }

// ---------------------------------------------------------------- zr gates
// pre (B,2H,N): channels [0,H) = z preactivation, [H,2H) = r preactivation.
// Flat vector index e over the (B,H,N) plane; the pre tensor needs the
// batch row doubled: b = e / HN, pre_z at b*2*HN + (e % HN), pre_r +HN.
template <typename T, int V>
__global__ __launch_bounds__(256) void gru_zr_fwd_kernel(
    const Vec<T, V> *__restrict__ pre, const Vec<T, V> *__restrict__ h,
    Vec<T, V> *__restrict__ z, Vec<T, V> *__restrict__ r,
    Vec<T, V> *__restrict__ rh, long HN, long total) {
  for (long e = blockIdx.x * (long)blockDim.x + threadIdx.x; e < total;
       e += (long)gridDim.x * blockDim.x) {
    const long b = e / HN, o = e - b * HN;
    const Vec<T, V> pz = pre[b * 2 * HN + o];
    const Vec<T, V> pr = pre[b * 2 * HN + HN + o];
    const Vec<T, V> hv = h[e];
    Vec<T, V> zv, rv, rhv;
#pragma unroll
    for (int i = 0; i < V; ++i) {
      const float zi = sigmoidf_(to_f32(pz.v[i]));
      const float ri = sigmoidf_(to_f32(pr.v[i]));
      zv.v[i] = from_f32<T>(zi);
      rv.v[i] = from_f32<T>(ri);
      rhv.v[i] = from_f32<T>(ri * to_f32(hv.v[i]));
    }
    z[e] = zv;
    r[e] = rv;
    rh[e] = rhv;
  }
}

template <typename T, int V>
__global__ __launch_bounds__(256) void gru_zr_bwd_kernel(
    const Vec<T, V> *__restrict__ dz, const Vec<T, V> *__restrict__ drh,
    const Vec<T, V> *__restrict__ z, const Vec<T, V> *__restrict__ r,
    const Vec<T, V> *__restrict__ h, Vec<T, V> *__restrict__ dpre,
    Vec<T, V> *__restrict__ dh, long HN, long total) {
  for (long e = blockIdx.x * (long)blockDim.x + threadIdx.x; e < total;
       e += (long)gridDim.x * blockDim.x) {
    const long b = e / HN, o = e - b * HN;
    const Vec<T, V> dzv = dz[e], drhv = drh[e], zv = z[e], rv = r[e],
                    hv = h[e];
    Vec<T, V> dpz, dpr, dhv;
#pragma unroll
    for (int i = 0; i < V; ++i) {
      const float zi = to_f32(zv.v[i]), ri = to_f32(rv.v[i]);
      const float g = to_f32(drhv.v[i]);
      dpz.v[i] = from_f32<T>(to_f32(dzv.v[i]) * zi * (1.0f - zi));
      dpr.v[i] = from_f32<T>(g * to_f32(hv.v[i]) * ri * (1.0f - ri));
      dhv.v[i] = from_f32<T>(g * ri);
    }
    dpre[b * 2 * HN + o] = dpz;
    dpre[b * 2 * HN + HN + o] = dpr;
    dh[e] = dhv;
  }
}

// ------------------------------------------------------------ q gate+blend
template <typename T, int V>
__global__ __launch_bounds__(256) void gru_q_fwd_kernel(
    const Vec<T, V> *__restrict__ pre, const Vec<T, V> *__restrict__ z,
    const Vec<T, V> *__restrict__ h, Vec<T, V> *__restrict__ q,
    Vec<T, V> *__restrict__ hnew, long total) {
  for (long e = blockIdx.x * (long)blockDim.x + threadIdx.x; e < total;
       e += (long)gridDim.x * blockDim.x) {
    const Vec<T, V> pv = pre[e], zv = z[e], hv = h[e];
    Vec<T, V> qv, hn;
#pragma unroll
    for (int i = 0; i < V; ++i) {
      const float qi = tanhf(to_f32(pv.v[i]));
      const float zi = to_f32(zv.v[i]);
      qv.v[i] = from_f32<T>(qi);
      hn.v[i] = from_f32<T>((1.0f - zi) * to_f32(hv.v[i]) + zi * qi);
    }
    q[e] = qv;
    hnew[e] = hn;
  }
}

template <typename T, int V>
__global__ __launch_bounds__(256) void gru_q_bwd_kernel(
    const Vec<T, V> *__restrict__ dhnew, const Vec<T, V> *__restrict__ q,
    const Vec<T, V> *__restrict__ z, const Vec<T, V> *__restrict__ h,
    Vec<T, V> *__restrict__ dpre, Vec<T, V> *__restrict__ dz,
    Vec<T, V> *__restrict__ dh, long total) {
  for (long e = blockIdx.x * (long)blockDim.x + threadIdx.x; e < total;
       e += (long)gridDim.x * blockDim.x) {
    const Vec<T, V> gv = dhnew[e], qv = q[e], zv = z[e], hv = h[e];
    Vec<T, V> dpv, dzv, dhv;
#pragma unroll
    for (int i = 0; i < V; ++i) {
      const float g = to_f32(gv.v[i]), qi = to_f32(qv.v[i]),
                  zi = to_f32(zv.v[i]);
      dpv.v[i] = from_f32<T>(g * zi * (1.0f - qi * qi));
      dzv.v[i] = from_f32<T>(g * (qi - to_f32(hv.v[i])));
      dhv.v[i] = from_f32<T>(g * (1.0f - zi));
    }
    dpre[e] = dpv;
    dz[e] = dzv;
    dh[e] = dhv;
  }
}

inline int blocks_for(long total_vec) {
  long b = (total_vec + 255) / 256;
  if (b > 4096) b = 4096;  // grid-stride covers the rest
  if (b < 1) b = 1;
  return (int)b;
}

}  // namespace

#define DISPATCH_TV(bf16, HN, total, ...)                    \
  do {                                                       \
    const bool vec_ok = ((HN) % 8 == 0);                     \
    if (bf16) {                                              \
      if (vec_ok) {                                          \
        using T = __hip_bfloat16;                            \
        constexpr int V = 8;                                 \
        __VA_ARGS__;                                         \
      } else {                                               \
        using T = __hip_bfloat16;                            \
        constexpr int V = 1;                                 \
        __VA_ARGS__;                                         \
      }                                                      \
    } else {                                                 \
      if (vec_ok) {                                          \
        using T = float;                                     \
        constexpr int V = 4;                                 \
        __VA_ARGS__;                                         \
      } else {                                               \
        using T = float;                                     \
        constexpr int V = 1;                                 \
        __VA_ARGS__;                                         \
      }                                                      \
    }                                                        \
  } while (0)

void launch_gru_zr_fwd(const void *pre, const void *h, void *z, void *r,
                       void *rh, long B, long HN, bool bf16,
                       hipStream_t stream) {
  const long total = B * HN;
  DISPATCH_TV(bf16, HN, total, {
    const long tv = total / V;
    hipLaunchKernelGGL((gru_zr_fwd_kernel<T, V>), dim3(blocks_for(tv)),
                       dim3(256), 0, stream, (const Vec<T, V> *)pre,
                       (const Vec<T, V> *)h, (Vec<T, V> *)z, (Vec<T, V> *)r,
                       (Vec<T, V> *)rh, HN / V, tv);
  });
}

void launch_gru_zr_bwd(const void *dz, const void *drh, const void *z,
                       const void *r, const void *h, void *dpre, void *dh,
                       long B, long HN, bool bf16, hipStream_t stream) {
  const long total = B * HN;
  DISPATCH_TV(bf16, HN, total, {
    const long tv = total / V;
    hipLaunchKernelGGL((gru_zr_bwd_kernel<T, V>), dim3(blocks_for(tv)),
                       dim3(256), 0, stream, (const Vec<T, V> *)dz,
                       (const Vec<T, V> *)drh, (const Vec<T, V> *)z,
                       (const Vec<T, V> *)r, (const Vec<T, V> *)h,
                       (Vec<T, V> *)dpre, (Vec<T, V> *)dh, HN / V, tv);
  });
}

void launch_gru_q_fwd(const void *pre, const void *z, const void *h, void *q,
                      void *hnew, long B, long HN, bool bf16,
                      hipStream_t stream) {
  const long total = B * HN;
  DISPATCH_TV(bf16, HN, total, {
    const long tv = total / V;
    hipLaunchKernelGGL((gru_q_fwd_kernel<T, V>), dim3(blocks_for(tv)),
                       dim3(256), 0, stream, (const Vec<T, V> *)pre,
                       (const Vec<T, V> *)z, (const Vec<T, V> *)h,
                       (Vec<T, V> *)q, (Vec<T, V> *)hnew, tv);
  });
}

void launch_gru_q_bwd(const void *dhnew, const void *q, const void *z,
                      const void *h, void *dpre, void *dz, void *dh, long B,
                      long HN, bool bf16, hipStream_t stream) {
  const long total = B * HN;
  DISPATCH_TV(bf16, HN, total, {
    const long tv = total / V;
    hipLaunchKernelGGL((gru_q_bwd_kernel<T, V>), dim3(blocks_for(tv)),
                       dim3(256), 0, stream, (const Vec<T, V> *)dhnew,
                       (const Vec<T, V> *)q, (const Vec<T, V> *)z,
                       (const Vec<T, V> *)h, (Vec<T, V> *)dpre,
                       (Vec<T, V> *)dz, (Vec<T, V> *)dh, tv);
  });
}
