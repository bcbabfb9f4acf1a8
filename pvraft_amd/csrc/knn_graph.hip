// K1: fused kNN graph construction (capability of reference
// model/flot/graph.py:53-60, which materialises the full B x N x N distance
// matrix and argsorts it).  Here: tiled candidate streaming through LDS +
// per-query top-k maintained in LDS -- the N x N matrix never exists.
//
// Geometry: 128 threads (2 waves) per block, one query point per thread.
// Per-thread top-k rows are padded to K+1 floats so the rescan (all threads
// reading slot j of their own row) spreads across LDS banks.
#include <hip/hip_runtime.h>
#include "common.h"

#define KNN_THREADS 128
#define KNN_TILE 128
#define KNN_MAXK 48  // model uses 32 (reference extractor.py:10)

__global__ __launch_bounds__(KNN_THREADS) void knn_graph_kernel(
    const float *__restrict__ xyz,  // (B, N, 3)
    int *__restrict__ out_idx,      // (B, N, k)
    int B, int N, int k) {
  __shared__ float s_tile[KNN_TILE * 3];
  __shared__ float s_dist[KNN_THREADS * (KNN_MAXK + 1)];
  __shared__ int s_idx[KNN_THREADS * (KNN_MAXK + 1)];

  const int b = blockIdx.y;
  const int q = blockIdx.x * KNN_THREADS + threadIdx.x;
  const bool active = q < N;

  float qx = 0.f, qy = 0.f, qz = 0.f;
  if (active) {
    const float *p = xyz + ((long)b * N + q) * 3;
    qx = p[0];
    qy = p[1];
    qz = p[2];
  }

  float *my_dist = s_dist + threadIdx.x * (KNN_MAXK + 1);
  int *my_idx = s_idx + threadIdx.x * (KNN_MAXK + 1);

  int filled = 0;          // slots used so far (< k during warmup)
  float worst = -1.f;      // current k-th distance
  int worst_slot = 0;

  for (int tile = 0; tile < N; tile += KNN_TILE) {
    const int tile_n = min(KNN_TILE, N - tile);
    __syncthreads();
    // cooperative stage: thread t loads candidate t of the tile
    if (threadIdx.x < tile_n) {
      const float *p = xyz + ((long)b * N + tile + threadIdx.x) * 3;
      s_tile[threadIdx.x * 3 + 0] = p[0];
      s_tile[threadIdx.x * 3 + 1] = p[1];
      s_tile[threadIdx.x * 3 + 2] = p[2];
    }
    __syncthreads();
    if (!active) continue;

    for (int c = 0; c < tile_n; ++c) {
      const float dx = s_tile[c * 3 + 0] - qx;
      const float dy = s_tile[c * 3 + 1] - qy;
      const float dz = s_tile[c * 3 + 2] - qz;
      const float d = dx * dx + dy * dy + dz * dz;
      if (filled < k) {
        my_dist[filled] = d;
        my_idx[filled] = tile + c;
        ++filled;
        if (filled == k) {  // initial scan for the worst slot
          worst = my_dist[0];
          worst_slot = 0;
          for (int j = 1; j < k; ++j)
            if (my_dist[j] > worst) {
              worst = my_dist[j];
              worst_slot = j;
            }
        }
      } else if (d < worst) {
        my_dist[worst_slot] = d;
        my_idx[worst_slot] = tile + c;
        worst = my_dist[0];
        worst_slot = 0;
        for (int j = 1; j < k; ++j)
          if (my_dist[j] > worst) {
            worst = my_dist[j];
            worst_slot = j;
          }
      }
    }
  }

  if (active) {
    int *dst = out_idx + ((long)b * N + q) * k;
    for (int j = 0; j < k; ++j) dst[j] = my_idx[j];
  }
}

void launch_knn_graph(const float *xyz, int *out_idx, int B, int N, int k,
                      hipStream_t stream) {
  dim3 grid((N + KNN_THREADS - 1) / KNN_THREADS, B);
  hipLaunchKernelGGL(knn_graph_kernel, grid, dim3(KNN_THREADS), 0, stream,
                     xyz, out_idx, B, N, k);
}
