// K7: split-K weight gradient for 1x1 convolutions, MFMA (CDNA4, bf16).
//
//   dW[o, i] = sum_{b, s} dy[b, o, s] * x[b, i, s]
//
// i.e. a (Co x Ci) GEMM with a huge reduction dim (B*S ~ 16-524k) and tiny
// output.  hipBLASLt's heuristic picks a non-split-K kernel here (a handful
// of workgroups on 256 CUs, measured ~210 us for ~5 us of work), and the
// batched-matmul reformulation forces permute copies (~25 ms/step of
// aten::copy_).  This kernel reads both operands IN PLACE (row-major over
// s, k-contiguous) and splits the reduction across B x SCHUNKS workgroups,
// accumulating fp32 partial tiles into dW with one atomicAdd per output
// element per block.
//
// Geometry: 256 threads = 4 waves per block; 64x64 output tile, each wave
// one 32x32 quadrant = 2x2 v_mfma_f32_16x16x32_bf16 fragments (16 fp32
// accumulators per lane).  K-loop stages 64x32 operand tiles through LDS
// (rows padded 16 B so the 16-lane fragment reads are bank-conflict-free).
//
// Fragment maps (gfx950, K=32): A and B operands are k-contiguous 8-vectors
// at row (lane & 15), k-offset (lane >> 4) * 8; C/D: col = lane & 15,
// row = (lane >> 4) * 4 + reg.
//
// Two entry points share the tile body:
//  * pw_wgrad_kernel        -- one (dy, x, dw) problem per launch;
//  * pw_wgrad_batched_kernel-- MANY problems in ONE launch.  The training
//    step runs ~180 such weight gradients (the 8-iteration GRU loop's
//    conv stacks); launched one-by-one they are launch/atomic-floor-bound
//    (~20 us each for ~1 us of math).  The batched kernel takes a device
//    array of job descriptors plus a block->(job, tile, chunk) map and
//    accumulates DIRECTLY into each weight's fp32 grad buffer, so the
//    per-call zero-fill + autograd accumulate-add pairs disappear with
//    the launches.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define WG_THREADS 256
#define TILE 64      // output tile edge
#define KB 32        // K-block per MFMA
#define LDS_PAD 8    // bf16 elements of row padding (16 B)

#include "pw_wgrad_job.h"

// One 64x64 output tile x one (b, chunk) slice of the reduction.
__device__ __forceinline__ void pw_wgrad_tile(
    const __hip_bfloat16 *__restrict__ dy, const __hip_bfloat16 *__restrict__ x,
    float *__restrict__ dw, float *__restrict__ dbias, int Co, int Ci, long S,
    int schunks, int tile_o, int tile_i, int b, int chunk,
    __hip_bfloat16 (*sA)[KB + LDS_PAD], __hip_bfloat16 (*sB)[KB + LDS_PAD]) {
  // chunk bounds KB-aligned so in-range stage loads are whole 16B vectors
  const long per = (((S + schunks - 1) / schunks + KB - 1) / KB) * KB;
  const long s_lo = chunk * per;
  const long s_hi = min(s_lo + per, S);
  if (s_lo >= S) return;
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
      if (dbias != nullptr && tile_i == 0) {
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

  // bias gradient: reduce the per-thread dy-row partials (4 threads per
  // row share ldr) and accumulate -- replaces a separate slow ATen reduce
  if (dbias != nullptr && tile_i == 0) {
    __shared__ float s_bias[TILE];
    if (threadIdx.x < TILE) s_bias[threadIdx.x] = 0.f;
    __syncthreads();
    atomicAdd(&s_bias[ldr], bias_part);
    __syncthreads();
    if (threadIdx.x < TILE && tile_o + threadIdx.x < Co)
      atomicAdd(&dbias[tile_o + threadIdx.x], s_bias[threadIdx.x]);
  }

  // C/D map: col = lane&15, row = (lane>>4)*4 + reg
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
        if (o < Co && i < Ci) atomicAdd(&dw[(long)o * Ci + i], acc[a][bb][e]);
      }
}

__global__ __launch_bounds__(WG_THREADS) void pw_wgrad_kernel(
    const __hip_bfloat16 *__restrict__ dy, const __hip_bfloat16 *__restrict__ x,
    float *__restrict__ dw, float *__restrict__ dbias, int B, int Co, int Ci,
    long S, int schunks) {
  __shared__ __hip_bfloat16 sA[TILE][KB + LDS_PAD];
  __shared__ __hip_bfloat16 sB[TILE][KB + LDS_PAD];
  const int b = blockIdx.z / schunks;
  const int chunk = blockIdx.z % schunks;
  pw_wgrad_tile(dy, x, dw, dbias, Co, Ci, S, schunks, blockIdx.x * TILE,
                blockIdx.y * TILE, b, chunk, sA, sB);
}

// map[blockIdx.x] = job(12b) | bz(12b) | to(4b) | ti(4b)
__global__ __launch_bounds__(WG_THREADS) void pw_wgrad_batched_kernel(
    const PwWgradJob *__restrict__ jobs, const unsigned int *__restrict__ map) {
  __shared__ __hip_bfloat16 sA[TILE][KB + LDS_PAD];
  __shared__ __hip_bfloat16 sB[TILE][KB + LDS_PAD];
  const unsigned int m = map[blockIdx.x];
  const PwWgradJob j = jobs[m >> 20];
  const int bz = (m >> 8) & 0xFFF;
  const int to = (m >> 4) & 0xF;
  const int ti = m & 0xF;
  pw_wgrad_tile((const __hip_bfloat16 *)j.dy, (const __hip_bfloat16 *)j.x,
                j.dw, j.dbias, j.Co, j.Ci, j.S, j.schunks, to * TILE,
                ti * TILE, bz / j.schunks, bz % j.schunks, sA, sB);
}

void launch_pw_wgrad(const void *dy, const void *x, float *dw, float *dbias,
                     int B, int Co, int Ci, long S, int schunks_opt,
                     hipStream_t stream) {
  const int to = (Co + TILE - 1) / TILE;
  const int ti = (Ci + TILE - 1) / TILE;
  long sc = schunks_opt;
  if (sc <= 0) {
    // default: fill ~512 workgroups, but keep >= 4 K-blocks per chunk
    sc = 512 / ((long)to * ti * B) + 1;
    long cap = S / (KB * 4);
    if (sc > cap) sc = cap;
    if (sc < 1) sc = 1;
  }
  int schunks = (int)sc;
  dim3 grid(to, ti, B * schunks);
  hipLaunchKernelGGL(pw_wgrad_kernel, grid, dim3(WG_THREADS), 0, stream,
                     (const __hip_bfloat16 *)dy, (const __hip_bfloat16 *)x, dw,
                     dbias, B, Co, Ci, S, schunks);
}

void launch_pw_wgrad_batched(const void *jobs_dev, const unsigned int *map_dev,
                             int n_blocks, hipStream_t stream) {
  hipLaunchKernelGGL(pw_wgrad_batched_kernel, dim3(n_blocks),
                     dim3(WG_THREADS), 0, stream,
                     (const PwWgradJob *)jobs_dev, map_dev);
}

// Capture-safe descriptor upload: the bytes ride in the kernel ARGUMENT
// (by value, < 4 KB kernarg limit), so filling the device buffer is a
// plain kernel launch that hipGraph capture records like any other --
// no H2D memcpy, no events, no pinned memory.  Each replay rewrites the
// same constants (descriptors are fixed for a captured graph).
struct DescChunk {
  unsigned char data[3584];
};

__global__ void desc_fill_kernel(unsigned char *__restrict__ dst, DescChunk c,
                                 int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[i] = c.data[i];
}

void launch_desc_fill(void *dst, const void *src, long bytes,
                      hipStream_t stream) {
  const int CH = (int)sizeof(DescChunk);
  for (long off = 0; off < bytes; off += CH) {
    DescChunk c;
    const int n = (int)((bytes - off) < CH ? (bytes - off) : CH);
    __builtin_memcpy(c.data, (const char *)src + off, n);
    hipLaunchKernelGGL(desc_fill_kernel, dim3((n + 255) / 256), dim3(256), 0,
                       stream, (unsigned char *)dst + off, c, n);
  }
}
