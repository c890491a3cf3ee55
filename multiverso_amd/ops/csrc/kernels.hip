// CDNA4 (gfx950 / MI355X) kernels for multiverso_amd.
//
// These are the rebuild's equivalents of the reference's CPU hot loops
// (SURVEY.md §2.9): K1 dense add (src/updater/updater.cpp:25-28), K2 SGD
// (sgd_updater.h:16-18), K3 momentum (momentum_updater.h:19-24), K4 adagrad
// (adagrad_updater.h:27-40, with the accumulator bug fixed), K5/K6 row
// scatter/gather (matrix_table.cpp:406-412 / :444-450).
//
// Design (per the CDNA4 HIP guide): every kernel here is memory-bound, so
// the shape is {block = 256 threads (4 waves of 64), float4 16B/lane
// vectorized loads/stores, grid-stride with the grid capped at 2048 blocks
// (256 CUs x 8 blocks)}. The stateful updaters (momentum/adagrad) fuse the
// state read-modify-write into the same pass as the weight update — one
// trip through HBM3E instead of the 3-4 separate passes a torch-op
// composition would make.
//
// No CUDA compatibility paths: this file is HIP-only, compiled by hipcc
// with --offload-arch=gfx950.

#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64
#define BLOCK 256
#define MAX_GRID 2048

static inline int grid_for(int64_t work_items) {
  int64_t blocks = (work_items + BLOCK - 1) / BLOCK;
  if (blocks > MAX_GRID) blocks = MAX_GRID;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// ---------------------------------------------------------------------------
// Elementwise updaters (K1-K4). Vectorized body processes n4 float4 groups;
// a scalar tail handles n % 4. All pointers are 16B-aligned (torch allocs).
// ---------------------------------------------------------------------------

__global__ void k_add_f4(float4* __restrict__ data,
                         const float4* __restrict__ delta, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 d = data[i], g = delta[i];
    d.x += g.x; d.y += g.y; d.z += g.z; d.w += g.w;
    data[i] = d;
  }
}

__global__ void k_add_tail(float* __restrict__ data,
                           const float* __restrict__ delta,
                           int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) data[i] += delta[i];
}

__global__ void k_sgd_f4(float4* __restrict__ data,
                         const float4* __restrict__ delta, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 d = data[i], g = delta[i];
    d.x -= g.x; d.y -= g.y; d.z -= g.z; d.w -= g.w;
    data[i] = d;
  }
}

__global__ void k_sgd_tail(float* __restrict__ data,
                           const float* __restrict__ delta,
                           int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) data[i] -= delta[i];
}

__global__ void k_momentum_f4(float4* __restrict__ data,
                              float4* __restrict__ m,
                              const float4* __restrict__ delta,
                              float mu, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float om = 1.0f - mu;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 d = data[i], mm = m[i], g = delta[i];
    mm.x = mu * mm.x + om * g.x; d.x -= mm.x;
    mm.y = mu * mm.y + om * g.y; d.y -= mm.y;
    mm.z = mu * mm.z + om * g.z; d.z -= mm.z;
    mm.w = mu * mm.w + om * g.w; d.w -= mm.w;
    m[i] = mm; data[i] = d;
  }
}

__global__ void k_momentum_tail(float* __restrict__ data, float* __restrict__ m,
                                const float* __restrict__ delta,
                                float mu, int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float mm = mu * m[i] + (1.0f - mu) * delta[i];
    m[i] = mm;
    data[i] -= mm;
  }
}

__global__ void k_adagrad_f4(float4* __restrict__ data,
                             float4* __restrict__ gsq,
                             const float4* __restrict__ delta,
                             float inv_lr, float rho, float eps, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 d = data[i], G = gsq[i], g4 = delta[i];
    float g;
    g = g4.x * inv_lr; G.x += g * g; d.x -= rho * g * __frsqrt_rn(G.x + eps);
    g = g4.y * inv_lr; G.y += g * g; d.y -= rho * g * __frsqrt_rn(G.y + eps);
    g = g4.z * inv_lr; G.z += g * g; d.z -= rho * g * __frsqrt_rn(G.z + eps);
    g = g4.w * inv_lr; G.w += g * g; d.w -= rho * g * __frsqrt_rn(G.w + eps);
    gsq[i] = G; data[i] = d;
  }
}

__global__ void k_adagrad_tail(float* __restrict__ data, float* __restrict__ gsq,
                               const float* __restrict__ delta,
                               float inv_lr, float rho, float eps,
                               int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float g = delta[i] * inv_lr;
    float G = gsq[i] + g * g;
    gsq[i] = G;
    data[i] -= rho * g * __frsqrt_rn(G + eps);
  }
}

// ---------------------------------------------------------------------------
// Row-keyed gather/scatter (K5/K6): shard is [local_rows, cols] row-major,
// rows[] are LOCAL row indices. Thread-per-element over (nrows * cols):
// adjacent lanes hit adjacent columns -> fully coalesced on both sides of
// the gather; the scatter uses device-scope atomicAdd so duplicate rows
// (the same row touched by several workers in one exchange) accumulate
// correctly.
// ---------------------------------------------------------------------------

__global__ void k_row_gather_f4(float4* __restrict__ out,
                                const float4* __restrict__ shard,
                                const int64_t* __restrict__ rows,
                                int64_t nrows, int64_t cols4) {
  int64_t total = nrows * cols4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int64_t r = i / cols4, c = i % cols4;
    out[i] = shard[rows[r] * cols4 + c];
  }
}

__global__ void k_row_gather(float* __restrict__ out,
                             const float* __restrict__ shard,
                             const int64_t* __restrict__ rows,
                             int64_t nrows, int64_t cols) {
  int64_t total = nrows * cols;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int64_t r = i / cols, c = i % cols;
    out[i] = shard[rows[r] * cols + c];
  }
}

__global__ void k_row_scatter_add(float* __restrict__ shard,
                                  const float* __restrict__ vals,
                                  const int64_t* __restrict__ rows,
                                  float sign, int64_t nrows, int64_t cols) {
  int64_t total = nrows * cols;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int64_t r = i / cols, c = i % cols;
    atomicAdd(&shard[rows[r] * cols + c], sign * vals[i]);
  }
}

// ---------------------------------------------------------------------------
// extern "C" launchers
// ---------------------------------------------------------------------------

extern "C" {

void mv_launch_add(float* data, const float* delta, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_add_f4<<<grid_for(n4), BLOCK, 0, s>>>(
      (float4*)data, (const float4*)delta, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_add_tail<<<1, 64, 0, s>>>(data, delta, n4 * 4, n);
}

void mv_launch_sgd(float* data, const float* delta, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_sgd_f4<<<grid_for(n4), BLOCK, 0, s>>>(
      (float4*)data, (const float4*)delta, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_sgd_tail<<<1, 64, 0, s>>>(data, delta, n4 * 4, n);
}

void mv_launch_momentum(float* data, float* m, const float* delta, float mu,
                        int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_momentum_f4<<<grid_for(n4), BLOCK, 0, s>>>(
      (float4*)data, (float4*)m, (const float4*)delta, mu, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_momentum_tail<<<1, 64, 0, s>>>(data, m, delta, mu, n4 * 4, n);
}

void mv_launch_adagrad(float* data, float* gsq, const float* delta,
                       float lr, float rho, float eps, int64_t n, hipStream_t s) {
  float inv_lr = 1.0f / lr;
  int64_t n4 = n / 4;
  if (n4) k_adagrad_f4<<<grid_for(n4), BLOCK, 0, s>>>(
      (float4*)data, (float4*)gsq, (const float4*)delta, inv_lr, rho, eps, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_adagrad_tail<<<1, 64, 0, s>>>(data, gsq, delta, inv_lr, rho, eps,
                                            n4 * 4, n);
}

void mv_launch_row_gather(float* out, const float* shard, const int64_t* rows,
                          int64_t nrows, int64_t cols, hipStream_t s) {
  if (!nrows || !cols) return;
  if (cols % 4 == 0) {
    int64_t cols4 = cols / 4;
    k_row_gather_f4<<<grid_for(nrows * cols4), BLOCK, 0, s>>>(
        (float4*)out, (const float4*)shard, rows, nrows, cols4);
  } else {
    k_row_gather<<<grid_for(nrows * cols), BLOCK, 0, s>>>(
        out, shard, rows, nrows, cols);
  }
}

void mv_launch_row_scatter_add(float* shard, const float* vals,
                               const int64_t* rows, float sign,
                               int64_t nrows, int64_t cols, hipStream_t s) {
  if (!nrows || !cols) return;
  k_row_scatter_add<<<grid_for(nrows * cols), BLOCK, 0, s>>>(
      shard, vals, rows, sign, nrows, cols);
}

}  // extern "C"
