// K7 forward: hand-written MFMA 1x1-conv GEMM with a fused epilogue.
//
//   y[b, o, s] = act( sum_i W_i[o, :] @ x_i[b, :, s] + bias[o]
//                     + addend[b, o, s] )
//
// Every 1x1 conv in PV-RAFT is such a GEMM (reference gconv.py:26-33,
// corr.py:15-29, update.py:11-29,60-66), and the concat-free update block
// (model/update.py) evaluates sums of COLUMN-SLICE GEMMs plus carried
// addends.  Launched one-by-one through hipBLASLt each piece is a ~5 us
// launch-floor kernel plus separate bias/add/ReLU elementwise launches;
// this kernel evaluates the whole sum-with-epilogue in ONE launch:
// multi-operand accumulation (the MFMA accumulator carries across parts),
// fp32 bias, optional fp32/bf16 addend (a row-contiguous channel-slice
// view, e.g. the hoisted GRU gate contribution m[:, :2H]), and an
// optional ReLU.
//
// Geometry: 256 threads = 4 waves; each workgroup owns a 64-column tile
// of one batch's output and ALL Co rows (template NRF = row fragments,
// statically indexed accumulators -- dynamic indexing would spill to
// scratch).  Per part, the (Co x Ci) weight and the (Ci x 64) activation
// tile are staged in dynamic LDS (the activation transposed on the way
// in: x is channel-major, the B fragment wants k-contiguous columns),
// zero-padded to a 32-multiple k so ragged Ci (61, 3, 81...) needs no
// edge cases.  bf16 operands, fp32 accumulate
// (v_mfma_f32_16x16x32_bf16), bf16 store.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdlib>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define PF_THREADS 256
#define PF_MAXP 4       // max summed parts
// Output columns per workgroup is a TEMPLATE parameter: at the flagship
// S=8192, 64-column tiles give only ceil(S/64)*B = 256 workgroups -- one
// per CU, 4 waves resident, latency-bound (~10 us for a ~2 us problem).
// Narrower tiles split the FOUR waves between column tiles and
// row-fragment halves (same LDS traffic, more workgroups): TC=32 doubles
// the grid, TC=16 quadruples it.  The launcher picks the widest TC that
// still yields >= 2 workgroups per CU.

struct PwFwdPart {
  const void *w;  // (Co, Ci) bf16 row-major
  const void *x;  // (B, Ci, S) bf16
  int ci;
};

template <int NRF, int TC>
__global__ __launch_bounds__(PF_THREADS) void pw_fwd_kernel(
    PwFwdPart p0, PwFwdPart p1, PwFwdPart p2, PwFwdPart p3, int nparts,
    const float *__restrict__ bias,   // (Co) fp32 or null
    const void *__restrict__ addend,  // bf16/fp32 slice base or null
    long addend_bstride, int addend_fp32,
    __hip_bfloat16 *__restrict__ y,   // (B, Co, S)
    int Co, long S, int act, int pitch) {
  constexpr int COL16 = TC / 16;        // column tiles per WG
  constexpr int RFG = 4 / COL16;        // wave groups along rows
  constexpr int NRF_PER = (NRF + RFG - 1) / RFG;  // row frags per wave
  extern __shared__ __hip_bfloat16 smem[];
  __hip_bfloat16 *s_w = smem;                       // (NRF*16, pitch)
  __hip_bfloat16 *s_x = smem + (long)NRF * 16 * pitch;  // (TC, pitch)

  const int b = blockIdx.z;
  const long s0 = (long)blockIdx.x * TC;
  const int lane = lane_id();
  const int wv = wave_id();
  const int ct = wv % COL16;            // this wave's column tile
  const int rf0 = (wv / COL16) * NRF_PER;  // first row fragment

  f32x4 acc[NRF_PER];
#pragma unroll
  for (int rf = 0; rf < NRF_PER; ++rf) acc[rf] = (f32x4)(0.f);

#pragma unroll
  for (int pi = 0; pi < PF_MAXP; ++pi) {
    if (pi >= nparts) break;
    // static selection: a runtime-indexed local array would spill the
    // descriptors to scratch (measured 112 B of scratch, ~3x slowdown)
    const PwFwdPart pp = pi == 0 ? p0 : pi == 1 ? p1 : pi == 2 ? p2 : p3;
    const int ci = pp.ci;
    const int cip = (ci + 31) & ~31;
    const __hip_bfloat16 *w = (const __hip_bfloat16 *)pp.w;
    const __hip_bfloat16 *x =
        (const __hip_bfloat16 *)pp.x + (long)b * ci * S;
    __syncthreads();
    // stage W rows (k-contiguous, zero-padded in both dims)
    for (int i = threadIdx.x; i < NRF * 16 * (cip / 8); i += PF_THREADS) {
      const int r = i / (cip / 8);
      const int c8 = (i % (cip / 8)) * 8;
      bf16x8 v = (bf16x8)(__bf16)0.0f;
      if (r < Co) {
        if (c8 + 8 <= ci) {
          v = *(const bf16x8 *)(w + (long)r * ci + c8);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (c8 + e < ci)
              ((__bf16 *)&v)[e] = *(const __bf16 *)(w + (long)r * ci + c8 + e);
        }
      }
      *(bf16x8 *)&s_w[(long)r * pitch + c8] = v;
    }
    // stage the activation tile TRANSPOSED: read x[r][s0+c] coalesced,
    // write s_x[c][r]
    for (int i = threadIdx.x; i < cip * (TC / 4); i += PF_THREADS) {
      const int r = i / (TC / 4);         // channel (k)
      const int c4 = (i % (TC / 4)) * 4;  // column group
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const long s = s0 + c4 + e;
        const float v = (r < ci && s < S) ? (float)x[(long)r * S + s] : 0.f;
        s_x[(long)(c4 + e) * pitch + r] = (__hip_bfloat16)v;
      }
    }
    __syncthreads();
    const int frow = lane & 15;
    const int koff = (lane >> 4) * 8;
    for (int kb = 0; kb < cip; kb += 32) {
      const bf16x8 bfrag =
          *(const bf16x8 *)&s_x[(long)(ct * 16 + frow) * pitch + kb + koff];
#pragma unroll
      for (int rf = 0; rf < NRF_PER; ++rf) {
        if (rf0 + rf < NRF) {  // wave-uniform guard (keeps the unroll)
          const bf16x8 afrag = *(const bf16x8 *)
              &s_w[(long)((rf0 + rf) * 16 + frow) * pitch + kb + koff];
          acc[rf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                            acc[rf], 0, 0, 0);
        }
      }
    }
  }

  // epilogue: bias + addend + activation, bf16 store
  const long sc = s0 + ct * 16 + (lane & 15);
  if (sc >= S) return;
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int rf = 0; rf < NRF_PER; ++rf)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int o = (rf0 + rf) * 16 + crow + e;
      if (rf0 + rf < NRF && o < Co) {
        float v = acc[rf][e];
        if (bias != nullptr) v += bias[o];
        if (addend != nullptr) {
          const long ai = (long)b * addend_bstride + (long)o * S + sc;
          v += addend_fp32 ? ((const float *)addend)[ai]
                           : (float)((const __hip_bfloat16 *)addend)[ai];
        }
        if (act == 1) v = v > 0.f ? v : 0.f;
        y[((long)b * Co + o) * S + sc] = (__hip_bfloat16)v;
      }
    }
}

void launch_pw_fwd(const void **ws, const void **xs, const int *cis,
                   int nparts, const float *bias, const void *addend,
                   long addend_bstride, int addend_fp32, void *y, int B,
                   int Co, long S, int act, hipStream_t stream) {
  PwFwdPart p[PF_MAXP] = {};
  int cip_max = 32;
  for (int i = 0; i < nparts; ++i) {
    p[i] = PwFwdPart{ws[i], xs[i], cis[i]};
    const int cip = (cis[i] + 31) & ~31;
    if (cip > cip_max) cip_max = cip;
  }
  const int pitch = cip_max + 8;
  const int nrf = (Co + 15) / 16;
  // Tile-width policy: narrow tiles add workgroups (waves split between
  // column tiles and row-fragment groups) at the cost of re-staging the
  // weight tile per extra WG.  A paired same-box A/B measured the narrow
  // policy 0.55 ms/step FASTER at the flagship shape (13.93 vs 14.46/
  // 14.51 classic); PVRAFT_PW_TC=classic opts out.
  static const bool classic = [] {
    const char* e = getenv("PVRAFT_PW_TC");
    return e && e[0] == 'c';
  }();
  int tc = 64;
  if (!classic) {
    if ((long)B * ((S + 63) / 64) < 512) tc = 32;
    if ((long)B * ((S + 31) / 32) < 512) tc = 16;
  }
#define PF_LAUNCH_T(NRF, TC)                                                  \
  do {                                                                        \
    const dim3 grid((unsigned)((S + TC - 1) / TC), 1, B);                     \
    const size_t shmem =                                                      \
        (size_t)(nrf * 16 + TC) * pitch * sizeof(__hip_bfloat16);             \
    hipLaunchKernelGGL((pw_fwd_kernel<NRF, TC>), grid, dim3(PF_THREADS),      \
                       shmem, stream, p[0], p[1], p[2], p[3], nparts, bias,   \
                       addend, addend_bstride, addend_fp32,                   \
                       (__hip_bfloat16 *)y, Co, S, act, pitch);               \
  } while (0)
#define PF_LAUNCH(NRF)                                                        \
  do {                                                                        \
    if (tc == 64) PF_LAUNCH_T(NRF, 64);                                       \
    else if (tc == 32) PF_LAUNCH_T(NRF, 32);                                  \
    else PF_LAUNCH_T(NRF, 16);                                                \
  } while (0)
  switch (nrf) {
    case 1: PF_LAUNCH(1); break;
    case 2: PF_LAUNCH(2); break;
    case 3: PF_LAUNCH(3); break;
    case 4: PF_LAUNCH(4); break;
    case 5: PF_LAUNCH(5); break;
    case 6: PF_LAUNCH(6); break;
    case 7: PF_LAUNCH(7); break;
    case 8: PF_LAUNCH(8); break;
    case 12: PF_LAUNCH(12); break;
    default: PF_LAUNCH(16); break;
  }
#undef PF_LAUNCH
#undef PF_LAUNCH_T
}
