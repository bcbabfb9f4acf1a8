// K7: split-K weight gradient for 1x1 convolutions, MFMA (CDNA4, bf16).
//
//   dW[o, i] = sum_{b, s} dy[b, o, s] * x[b, i, s]
//
// i.e. a (Co x Ci) GEMM with a huge reduction dim (B*S ~ 16-524k) and tiny
// output.  hipBLASLt's heuristic picks a non-split-K kernel here (a handful
// of workgroups on 256 CUs, measured ~210 us for ~5 us of work), and the
// batched-matmul reformulation forces permute copies (~25 ms/step of
// aten::copy_).  This kernel reads both operands IN PLACE (row-major over
// s, k-contiguous) and splits the reduction across B x SCHUNKS workgroups.
//
// TWO-STAGE accumulation: stage 1 writes each block's fp32 partial tile to
// a persistent scratch with PLAIN stores (the padded (B*sc, to*64, ti*64)
// region is fully overwritten every call, so it is never zeroed and never
// atomically contended -- the earlier one-stage version did ~4096 global
// atomicAdds per block onto the tiny dW and needed a zero-fill launch per
// call, together ~40% of the kernel's time at the per-iteration shapes);
// stage 2 is a thin reduction producing dW (and the fused bias grad) with
// one thread per output element.
//
// Geometry: 256 threads = 4 waves per block; 64x64 output tile, each wave
// one 32x32 quadrant = 2x2 v_mfma_f32_16x16x32_bf16 fragments (16 fp32
// accumulators per lane).  K-loop stages 64x32 operand tiles through LDS
// (rows padded 16 B so the 16-lane fragment reads are bank-conflict-free).
//
// Fragment maps (gfx950, K=32): A and B operands are k-contiguous 8-vectors
// at row (lane & 15), k-offset (lane >> 4) * 8; C/D: col = lane & 15,
// row = (lane >> 4) * 4 + reg.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define WG_THREADS 256
#define TILE 64      // output tile edge
#define KB 32        // K-block per MFMA
#define LDS_PAD 8    // bf16 elements of row padding (16 B)

__global__ __launch_bounds__(WG_THREADS) void pw_wgrad_partial_kernel(
    const __hip_bfloat16 *__restrict__ dy,  // (B, Co, S)
    const __hip_bfloat16 *__restrict__ x,   // (B, Ci, S)
    float *__restrict__ part,               // (B*sc, to*64, ti*64)
    float *__restrict__ bias_part_out,      // (B*sc, to*64) or null
    int B, int Co, int Ci, long S, int schunks, int to64, int ti64) {
  __shared__ __hip_bfloat16 sA[TILE][KB + LDS_PAD];
  __shared__ __hip_bfloat16 sB[TILE][KB + LDS_PAD];

  const int tile_o = blockIdx.x * TILE;
  const int tile_i = blockIdx.y * TILE;
  const int b = blockIdx.z / schunks;
  const int chunk = blockIdx.z % schunks;
  // chunk bounds KB-aligned so in-range stage loads are whole 16B vectors
  const long per = (((S + schunks - 1) / schunks + KB - 1) / KB) * KB;
  const long s_lo = chunk * per;
  const long s_hi = min(s_lo + per, S);
  // NOTE: blocks with an empty range still fall through and store their
  // (zero) partial tile -- the scratch is never pre-zeroed.
  const bool vec_ok = (S % 8 == 0);

  const int lane = lane_id();
  const int wid = wave_id();
  const int wo = (wid >> 1) * 32;  // wave's quadrant offsets in the tile
  const int wi = (wid & 1) * 32;

  f32x4 acc[2][2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int bb = 0; bb < 2; ++bb) acc[a][bb] = (f32x4)(0.f);

  const __hip_bfloat16 *dyb = dy + ((long)b * Co) * S;
  const __hip_bfloat16 *xb = x + ((long)b * Ci) * S;

  // stage loop: 64 rows x 32 k each for A (dy rows) and B (x rows)
  // loads: 2048 elems per tile, 256 threads -> 8 contiguous bf16 per thread
  const int ldr = threadIdx.x / 4;        // row 0..63
  const int ldc = (threadIdx.x % 4) * 8;  // col 0,8,16,24
  float bias_part = 0.f;                  // this thread's dy row partial
  for (long s0 = s_lo; s0 < s_hi; s0 += KB) {
    __syncthreads();
    {
      const int o = tile_o + ldr;
      const long s = s0 + ldc;
      const bool full = vec_ok && (s + 8 <= s_hi);
      bf16x8 av = (bf16x8)(__bf16)0.0f;
      if (o < Co) {
        const __hip_bfloat16 *src = dyb + (long)o * S + s;
        if (full) {
          av = *(const bf16x8 *)src;  // one coalesced 16 B load
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (s + e < s_hi) ((__bf16 *)&av)[e] = *(const __bf16 *)(src + e);
        }
      }
      *(bf16x8 *)&sA[ldr][ldc] = av;
      if (bias_part_out != nullptr && tile_i == 0) {
#pragma unroll
        for (int e = 0; e < 8; ++e) bias_part += (float)((__bf16 *)&av)[e];
      }
      const int i = tile_i + ldr;
      bf16x8 bv = (bf16x8)(__bf16)0.0f;
      if (i < Ci) {
        const __hip_bfloat16 *src = xb + (long)i * S + s;
        if (full) {
          bv = *(const bf16x8 *)src;
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (s + e < s_hi) ((__bf16 *)&bv)[e] = *(const __bf16 *)(src + e);
        }
      }
      *(bf16x8 *)&sB[ldr][ldc] = bv;
    }
    __syncthreads();

    const int frow = lane & 15;
    const int foff = (lane >> 4) * 8;
#pragma unroll
    for (int a = 0; a < 2; ++a) {
      const bf16x8 afrag = *(const bf16x8 *)&sA[wo + a * 16 + frow][foff];
#pragma unroll
      for (int bb = 0; bb < 2; ++bb) {
        const bf16x8 bfrag = *(const bf16x8 *)&sB[wi + bb * 16 + frow][foff];
        acc[a][bb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                             acc[a][bb], 0, 0, 0);
      }
    }
  }

  // bias-grad partials: reduce this block's per-thread dy-row sums into
  // one row vector and store it (only the tile_i == 0 column of blocks)
  if (bias_part_out != nullptr && tile_i == 0) {
    __shared__ float s_bias[TILE];
    if (threadIdx.x < TILE) s_bias[threadIdx.x] = 0.f;
    __syncthreads();
    atomicAdd(&s_bias[ldr], bias_part);  // LDS, 4 lanes per row
    __syncthreads();
    if (threadIdx.x < TILE)
      bias_part_out[(long)blockIdx.z * to64 + tile_o + threadIdx.x] =
          s_bias[threadIdx.x];
  }

  // C/D map: col = lane&15, row = (lane>>4)*4 + reg; plain stores into the
  // padded partial tile (no guards: o_pad < to*64, i_pad < ti*64 always)
  float *pt = part + (long)blockIdx.z * to64 * ti64;
  const int crow = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int bb = 0; bb < 2; ++bb)
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int o = tile_o + wo + a * 16 + crow + e;
        const int i = tile_i + wi + bb * 16 + ccol;
        pt[(long)o * ti64 + i] = acc[a][bb][e];
      }
}

// stage 2: dw[o,i] = sum_z part[z,o,i]; threads with i==0 also reduce the
// bias partials.  Z = B*schunks <= a few hundred; reads are i-coalesced.
__global__ __launch_bounds__(WG_THREADS) void pw_wgrad_reduce_kernel(
    const float *__restrict__ part, const float *__restrict__ bias_part,
    float *__restrict__ dw, float *__restrict__ dbias, int Co, int Ci, int Z,
    int to64, int ti64) {
  const long t = (long)blockIdx.x * WG_THREADS + threadIdx.x;
  const long total = (long)Co * Ci;
  if (t < total) {
    const int o = (int)(t / Ci);
    const int i = (int)(t - (long)o * Ci);
    float s = 0.f;
    const long stride = (long)to64 * ti64;
    const float *p = part + (long)o * ti64 + i;
    for (int z = 0; z < Z; ++z) s += p[(long)z * stride];
    dw[t] = s;
  }
  if (dbias != nullptr && t < Co) {
    float s = 0.f;
    const float *p = bias_part + t;
    for (int z = 0; z < Z; ++z) s += p[(long)z * to64];
    dbias[t] = s;
  }
}

void launch_pw_wgrad(const void *dy, const void *x, float *dw, float *dbias,
                     float *part, float *bias_part, int B, int Co, int Ci,
                     long S, int schunks, hipStream_t stream) {
  const int to = (Co + TILE - 1) / TILE;
  const int ti = (Ci + TILE - 1) / TILE;
  const int to64 = to * TILE, ti64 = ti * TILE;
  dim3 grid(to, ti, B * schunks);
  hipLaunchKernelGGL(pw_wgrad_partial_kernel, grid, dim3(WG_THREADS), 0,
                     stream, (const __hip_bfloat16 *)dy,
                     (const __hip_bfloat16 *)x, part,
                     dbias != nullptr ? bias_part : nullptr, B, Co, Ci, S,
                     schunks, to64, ti64);
  const long total = (long)Co * Ci;
  hipLaunchKernelGGL(pw_wgrad_reduce_kernel,
                     dim3((unsigned)((total + WG_THREADS - 1) / WG_THREADS)),
                     dim3(WG_THREADS), 0, stream, part, bias_part, dw, dbias,
                     Co, Ci, B * schunks, to64, ti64);
}

int pw_wgrad_schunks(int B, int Co, int Ci, long S, int schunks_opt) {
  if (schunks_opt > 0) return schunks_opt;
  const int to = (Co + TILE - 1) / TILE;
  const int ti = (Ci + TILE - 1) / TILE;
  // atomic-free split: fill the chip, bounded by one KB-block per chunk
  long sc = 768 / ((long)to * ti * B) + 1;
  long cap = (S + KB - 1) / KB;
  if (sc > cap) sc = cap;
  if (sc > 256) sc = 256;
  if (sc < 1) sc = 1;
  return (int)sc;
}
