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
#include <cstdio>
#include <cstdlib>

#define WAVE 64
#define BLOCK 256
#define MAX_GRID 2048
// elementwise streaming kernels saturate HBM at a smaller grid
// (tools/probe_sgd.hip sweep: 5.85 TB/s at 1024 blocks w/ nontemporal
// access vs 5.16 at 2048 plain). Kernels with >=4 concurrent HBM streams
// (fused add+get, momentum/adagrad/dcasgd state updates) peak lower
// still: 6.05/5.84 TB/s at 768 blocks vs 5.45/5.15 at 1024 (same sweep).
#define ELEM_GRID 1024
#define STATE_GRID 768
// 3-stream update kernels (read data + read delta + write data) peak at
// 768 blocks, not 1024: 6.05-6.07 vs 5.85-5.86 TB/s on two boxes
// (tools/probe_sgd2.hip round-2 sweep). The pure 2-stream copy keeps
// ELEM_GRID (6.15 at 1024).
#define UPD_GRID 768
// 4-stream fused Add+Get kernels prefer FAT blocks: 1024 threads x 768+
// blocks measured 6.14 TB/s vs 5.87 at 256x768 (tools/probe_sgd2.hip
// round-2 sweep; the 2/3-stream kernels regress at 1024 — they keep
// BLOCK=256). Grid counts are in BLOCKS of COPY_BLOCK threads.
#define COPY_BLOCK 1024
#define COPY_GRID 768

static inline int grid_copy(int64_t work_items) {
  int64_t blocks = (work_items + COPY_BLOCK - 1) / COPY_BLOCK;
  if (blocks > COPY_GRID) blocks = COPY_GRID;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

typedef float v4f __attribute__((ext_vector_type(4)));

static __device__ __forceinline__ v4f ntload(const v4f* p) {
  return __builtin_nontemporal_load(p);
}
static __device__ __forceinline__ void ntstore(v4f* p, v4f v) {
  __builtin_nontemporal_store(v, p);
}

static inline int grid_for_cap(int64_t work_items, int cap) {
  int64_t blocks = (work_items + BLOCK - 1) / BLOCK;
  if (blocks > cap) blocks = cap;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}
static inline int grid_for(int64_t work_items) {
  return grid_for_cap(work_items, MAX_GRID);
}

// ---------------------------------------------------------------------------
// Elementwise updaters (K1-K4). Vectorized body processes n4 float4 groups;
// a scalar tail handles n % 4. All pointers are 16B-aligned (torch allocs).
// ---------------------------------------------------------------------------

__global__ void k_add_f4(v4f* __restrict__ data,
                         const v4f* __restrict__ delta, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    ntstore(&data[i], ntload(&data[i]) + ntload(&delta[i]));
  }
}

__global__ void k_add_tail(float* __restrict__ data,
                           const float* __restrict__ delta,
                           int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) data[i] += delta[i];
}

__global__ void k_sgd_f4(v4f* __restrict__ data,
                         const v4f* __restrict__ delta, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    ntstore(&data[i], ntload(&data[i]) - ntload(&delta[i]));
  }
}

__global__ void k_sgd_tail(float* __restrict__ data,
                           const float* __restrict__ delta,
                           int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) data[i] -= delta[i];
}

__global__ void k_momentum_f4(v4f* __restrict__ data,
                              v4f* __restrict__ m,
                              const v4f* __restrict__ delta,
                              float mu, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float om = 1.0f - mu;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f mm = mu * ntload(&m[i]) + om * ntload(&delta[i]);
    ntstore(&m[i], mm);
    ntstore(&data[i], ntload(&data[i]) - mm);
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

__global__ void k_adagrad_f4(v4f* __restrict__ data,
                             v4f* __restrict__ gsq,
                             const v4f* __restrict__ delta,
                             float inv_lr, float rho, float eps, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f d = ntload(&data[i]), G = ntload(&gsq[i]);
    v4f g = ntload(&delta[i]) * inv_lr;
    G += g * g;
    d.x -= rho * g.x * __frsqrt_rn(G.x + eps);
    d.y -= rho * g.y * __frsqrt_rn(G.y + eps);
    d.z -= rho * g.z * __frsqrt_rn(G.z + eps);
    d.w -= rho * g.w * __frsqrt_rn(G.w + eps);
    ntstore(&gsq[i], G); ntstore(&data[i], d);
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

// Fused Add+Get (single-rank fast path): an Add immediately followed by
// a whole-table Get re-reads the shard the updater just wrote. Fusing
// the Get's copy-out into the updater pass saves that re-read — 2.0 GB
// instead of 2.5 GB per Add+Get step on the 1e6x128 headline config
// (observable semantics identical: shard updated AND out filled with the
// updated values).
__global__ void k_sgd_copy_f4(v4f* __restrict__ data,
                              const v4f* __restrict__ delta,
                              v4f* __restrict__ out, float sign, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f v = ntload(&data[i]) + sign * ntload(&delta[i]);
    ntstore(&data[i], v);
    ntstore(&out[i], v);
  }
}

__global__ void k_sgd_copy_tail(float* __restrict__ data,
                                const float* __restrict__ delta,
                                float* __restrict__ out, float sign,
                                int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float v = data[i] + sign * delta[i];
    data[i] = v;
    out[i] = v;
  }
}

// Fused stateful-updater + Get variants: same saving as k_sgd_copy for
// momentum/adagrad/dcasgd(a) — the updated weights stream to the Get
// buffer in the same pass instead of a separate shard re-read.
__global__ void k_momentum_copy_f4(v4f* __restrict__ data, v4f* __restrict__ m,
                                   const v4f* __restrict__ delta,
                                   v4f* __restrict__ out, float mu, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float om = 1.0f - mu;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f mm = mu * ntload(&m[i]) + om * ntload(&delta[i]);
    ntstore(&m[i], mm);
    v4f d = ntload(&data[i]) - mm;
    ntstore(&data[i], d);
    ntstore(&out[i], d);
  }
}

__global__ void k_momentum_copy_tail(float* __restrict__ data,
                                     float* __restrict__ m,
                                     const float* __restrict__ delta,
                                     float* __restrict__ out, float mu,
                                     int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float mm = mu * m[i] + (1.0f - mu) * delta[i];
    m[i] = mm;
    float d = data[i] - mm;
    data[i] = d;
    out[i] = d;
  }
}

__global__ void k_adagrad_copy_f4(v4f* __restrict__ data, v4f* __restrict__ gsq,
                                  const v4f* __restrict__ delta,
                                  v4f* __restrict__ out,
                                  float inv_lr, float rho, float eps,
                                  int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f d = ntload(&data[i]), G = ntload(&gsq[i]);
    v4f g = ntload(&delta[i]) * inv_lr;
    G += g * g;
    d.x -= rho * g.x * __frsqrt_rn(G.x + eps);
    d.y -= rho * g.y * __frsqrt_rn(G.y + eps);
    d.z -= rho * g.z * __frsqrt_rn(G.z + eps);
    d.w -= rho * g.w * __frsqrt_rn(G.w + eps);
    ntstore(&gsq[i], G);
    ntstore(&data[i], d);
    ntstore(&out[i], d);
  }
}

__global__ void k_adagrad_copy_tail(float* __restrict__ data,
                                    float* __restrict__ gsq,
                                    const float* __restrict__ delta,
                                    float* __restrict__ out,
                                    float inv_lr, float rho, float eps,
                                    int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float g = delta[i] * inv_lr;
    float G = gsq[i] + g * g;
    gsq[i] = G;
    float d = data[i] - rho * g * __frsqrt_rn(G + eps);
    data[i] = d;
    out[i] = d;
  }
}

// DC-ASGD updaters ("dcasgd"/"dcasgda", selected at reference
// updater.cpp:51-54 from a submodule ABSENT from the snapshot
// (.gitmodules:1-3, empty dir) — math reconstructed from the DC-ASGD
// paper (Zheng et al., "Asynchronous SGD with Delay Compensation",
// ICML 2017): the server compensates a delayed gradient g with a
// diagonal Hessian approximation g*g against the backup weights the
// worker pulled:
//   w    -= lr * (g + lambda * g*g * (w - bak))
//   bak   = w            (per-worker backup, updated after the step)
// Adaptive variant keeps a mean-square m = rho*m + (1-rho)*g*g and uses
// lambda / sqrt(m + eps) as the compensation coefficient.
__global__ void k_dcasgd_f4(v4f* __restrict__ data, v4f* __restrict__ bak,
                            const v4f* __restrict__ delta,
                            float lr, float lambda, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f w = ntload(&data[i]), b = ntload(&bak[i]);
    v4f g = ntload(&delta[i]);
    w -= lr * (g + lambda * g * g * (w - b));
    ntstore(&data[i], w);
    ntstore(&bak[i], w);
  }
}

__global__ void k_dcasgd_tail(float* __restrict__ data, float* __restrict__ bak,
                              const float* __restrict__ delta,
                              float lr, float lambda, int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float g = delta[i];
    float w = data[i] - lr * (g + lambda * g * g * (data[i] - bak[i]));
    data[i] = w;
    bak[i] = w;
  }
}

__global__ void k_dcasgda_f4(v4f* __restrict__ data, v4f* __restrict__ bak,
                             v4f* __restrict__ msq,
                             const v4f* __restrict__ delta,
                             float lr, float lambda, float rho, float eps,
                             int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float om = 1.0f - rho;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f w = ntload(&data[i]), b = ntload(&bak[i]);
    v4f g = ntload(&delta[i]);
    v4f m = rho * ntload(&msq[i]) + om * g * g;
    ntstore(&msq[i], m);
    v4f lam;
    lam.x = lambda * __frsqrt_rn(m.x + eps);
    lam.y = lambda * __frsqrt_rn(m.y + eps);
    lam.z = lambda * __frsqrt_rn(m.z + eps);
    lam.w = lambda * __frsqrt_rn(m.w + eps);
    w -= lr * (g + lam * g * g * (w - b));
    ntstore(&data[i], w);
    ntstore(&bak[i], w);
  }
}

__global__ void k_dcasgda_tail(float* __restrict__ data, float* __restrict__ bak,
                               float* __restrict__ msq,
                               const float* __restrict__ delta,
                               float lr, float lambda, float rho, float eps,
                               int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float g = delta[i];
    float m = rho * msq[i] + (1.0f - rho) * g * g;
    msq[i] = m;
    float w = data[i]
        - lr * (g + lambda * __frsqrt_rn(m + eps) * g * g * (data[i] - bak[i]));
    data[i] = w;
    bak[i] = w;
  }
}

__global__ void k_dcasgd_copy_f4(v4f* __restrict__ data, v4f* __restrict__ bak,
                                 const v4f* __restrict__ delta,
                                 v4f* __restrict__ out,
                                 float lr, float lambda, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f w = ntload(&data[i]), b = ntload(&bak[i]);
    v4f g = ntload(&delta[i]);
    w -= lr * (g + lambda * g * g * (w - b));
    ntstore(&data[i], w);
    ntstore(&bak[i], w);
    ntstore(&out[i], w);
  }
}

__global__ void k_dcasgd_copy_tail(float* __restrict__ data,
                                   float* __restrict__ bak,
                                   const float* __restrict__ delta,
                                   float* __restrict__ out,
                                   float lr, float lambda,
                                   int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float g = delta[i];
    float w = data[i] - lr * (g + lambda * g * g * (data[i] - bak[i]));
    data[i] = w;
    bak[i] = w;
    out[i] = w;
  }
}

__global__ void k_dcasgda_copy_f4(v4f* __restrict__ data, v4f* __restrict__ bak,
                                  v4f* __restrict__ msq,
                                  const v4f* __restrict__ delta,
                                  v4f* __restrict__ out,
                                  float lr, float lambda, float rho, float eps,
                                  int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float om = 1.0f - rho;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f w = ntload(&data[i]), b = ntload(&bak[i]);
    v4f g = ntload(&delta[i]);
    v4f m = rho * ntload(&msq[i]) + om * g * g;
    ntstore(&msq[i], m);
    v4f lam;
    lam.x = lambda * __frsqrt_rn(m.x + eps);
    lam.y = lambda * __frsqrt_rn(m.y + eps);
    lam.z = lambda * __frsqrt_rn(m.z + eps);
    lam.w = lambda * __frsqrt_rn(m.w + eps);
    w -= lr * (g + lam * g * g * (w - b));
    ntstore(&data[i], w);
    ntstore(&bak[i], w);
    ntstore(&out[i], w);
  }
}

__global__ void k_dcasgda_copy_tail(float* __restrict__ data,
                                    float* __restrict__ bak,
                                    float* __restrict__ msq,
                                    const float* __restrict__ delta,
                                    float* __restrict__ out,
                                    float lr, float lambda, float rho,
                                    float eps, int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    float g = delta[i];
    float m = rho * msq[i] + (1.0f - rho) * g * g;
    msq[i] = m;
    float w = data[i]
        - lr * (g + lambda * __frsqrt_rn(m + eps) * g * g * (data[i] - bak[i]));
    data[i] = w;
    bak[i] = w;
    out[i] = w;
  }
}

__global__ void k_copy_f4(v4f* __restrict__ dst,
                          const v4f* __restrict__ src, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    ntstore(&dst[i], ntload(&src[i]));
  }
}

__global__ void k_copy_tail(float* __restrict__ dst,
                            const float* __restrict__ src,
                            int64_t start, int64_t n) {
  int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[i] = src[i];
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

__global__ void k_row_gather_f4_u32(float4* __restrict__ out,
                                    const float4* __restrict__ shard,
                                    const int64_t* __restrict__ rows,
                                    uint32_t total4, uint32_t cols4) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total4;
       i += stride) {
    uint32_t r = i / cols4, c = i - r * cols4;
    out[i] = shard[rows[r] * (int64_t)cols4 + c];
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

// u32 loop-index variant: a 64-bit divide is ~40 cycles per element and
// these gathers run one divide per 4 B moved.  Only the (i -> r, c)
// math narrows — the shard offset stays 64-bit (1e9-row tables).
// Launcher picks this whenever total fits u32 (all real pulls do).
__global__ void k_row_gather_u32(float* __restrict__ out,
                                 const float* __restrict__ shard,
                                 const int64_t* __restrict__ rows,
                                 uint32_t total, uint32_t cols) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    uint32_t r = i / cols, c = i - r * cols;
    out[i] = shard[rows[r] * (int64_t)cols + c];
  }
}

__global__ void k_row_gather_f2_u32(float2* __restrict__ out,
                                    const float2* __restrict__ shard,
                                    const int64_t* __restrict__ rows,
                                    uint32_t total2, uint32_t cols2) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total2;
       i += stride) {
    uint32_t r = i / cols2, c = i - r * cols2;
    out[i] = shard[rows[r] * (int64_t)cols2 + c];
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

// unique-rows variants: when the caller guarantees no duplicate row ids
// (e.g. a sorted-unique union pushed from ONE rank) the atomics are
// unnecessary — plain read-modify-write is substantially faster for
// scattered few-column rows.
__global__ void k_row_scatter_add_u(float* __restrict__ shard,
                                    const float* __restrict__ vals,
                                    const int64_t* __restrict__ rows,
                                    float sign, int64_t nrows, int64_t cols) {
  int64_t total = nrows * cols;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int64_t r = i / cols, c = i % cols;
    int64_t k = rows[r] * cols + c;
    shard[k] += sign * vals[i];
  }
}

// u32 loop-index forms (see k_row_gather_u32: one 64-bit divide per 4 B
// is real VALU time on a gather/scatter; the shard offset stays 64-bit)
__global__ void k_row_scatter_add_u32(float* __restrict__ shard,
                                      const float* __restrict__ vals,
                                      const int64_t* __restrict__ rows,
                                      float sign, uint32_t total,
                                      uint32_t cols) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    uint32_t r = i / cols, c = i - r * cols;
    atomicAdd(&shard[rows[r] * (int64_t)cols + c], sign * vals[i]);
  }
}

__global__ void k_row_scatter_add_u_u32(float* __restrict__ shard,
                                        const float* __restrict__ vals,
                                        const int64_t* __restrict__ rows,
                                        float sign, uint32_t total,
                                        uint32_t cols) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    uint32_t r = i / cols, c = i - r * cols;
    int64_t k = rows[r] * (int64_t)cols + c;
    shard[k] += sign * vals[i];
  }
}

// ---------------------------------------------------------------------------
// extern "C" launchers
// ---------------------------------------------------------------------------

extern "C" {

// K7 Access / shard copy-out: non-temporal streaming copy.
void mv_launch_copy(float* dst, const float* src, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_copy_f4<<<grid_for_cap(n4, ELEM_GRID), BLOCK, 0, s>>>(
      (v4f*)dst, (const v4f*)src, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_copy_tail<<<1, 64, 0, s>>>(dst, src, n4 * 4, n);
}

void mv_launch_add(float* data, const float* delta, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_add_f4<<<grid_for_cap(n4, UPD_GRID), BLOCK, 0, s>>>(
      (v4f*)data, (const v4f*)delta, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_add_tail<<<1, 64, 0, s>>>(data, delta, n4 * 4, n);
}

void mv_launch_sgd(float* data, const float* delta, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_sgd_f4<<<grid_for_cap(n4, UPD_GRID), BLOCK, 0, s>>>(
      (v4f*)data, (const v4f*)delta, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_sgd_tail<<<1, 64, 0, s>>>(data, delta, n4 * 4, n);
}

void mv_launch_momentum(float* data, float* m, const float* delta, float mu,
                        int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_momentum_f4<<<grid_for_cap(n4, STATE_GRID), BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)m, (const v4f*)delta, mu, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_momentum_tail<<<1, 64, 0, s>>>(data, m, delta, mu, n4 * 4, n);
}

void mv_launch_adagrad(float* data, float* gsq, const float* delta,
                       float lr, float rho, float eps, int64_t n, hipStream_t s) {
  float inv_lr = 1.0f / lr;
  int64_t n4 = n / 4;
  if (n4) k_adagrad_f4<<<grid_for_cap(n4, STATE_GRID), BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)gsq, (const v4f*)delta, inv_lr, rho, eps, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_adagrad_tail<<<1, 64, 0, s>>>(data, gsq, delta, inv_lr, rho, eps,
                                            n4 * 4, n);
}

// sign = +1 for the default (add) updater, -1 for sgd.
// Fat-block geometry (COPY_BLOCK x COPY_GRID): see the round-2 sweep
// note at the COPY_BLOCK define.
void mv_launch_sgd_copy(float* data, const float* delta, float* out,
                        float sign, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_sgd_copy_f4<<<grid_copy(n4), COPY_BLOCK, 0, s>>>(
      (v4f*)data, (const v4f*)delta, (v4f*)out, sign, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_sgd_copy_tail<<<1, 64, 0, s>>>(data, delta, out, sign,
                                             n4 * 4, n);
}

void mv_launch_momentum_copy(float* data, float* m, const float* delta,
                             float* out, float mu, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_momentum_copy_f4<<<grid_copy(n4), COPY_BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)m, (const v4f*)delta, (v4f*)out, mu, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_momentum_copy_tail<<<1, 64, 0, s>>>(data, m, delta, out, mu,
                                                  n4 * 4, n);
}

void mv_launch_adagrad_copy(float* data, float* gsq, const float* delta,
                            float* out, float lr, float rho, float eps,
                            int64_t n, hipStream_t s) {
  float inv_lr = 1.0f / lr;
  int64_t n4 = n / 4;
  if (n4) k_adagrad_copy_f4<<<grid_copy(n4), COPY_BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)gsq, (const v4f*)delta, (v4f*)out, inv_lr, rho, eps,
      n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_adagrad_copy_tail<<<1, 64, 0, s>>>(data, gsq, delta, out,
                                                 inv_lr, rho, eps, n4 * 4, n);
}

void mv_launch_dcasgd_copy(float* data, float* bak, const float* delta,
                           float* out, float lr, float lambda, int64_t n,
                           hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_dcasgd_copy_f4<<<grid_copy(n4), COPY_BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)bak, (const v4f*)delta, (v4f*)out, lr, lambda, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_dcasgd_copy_tail<<<1, 64, 0, s>>>(data, bak, delta, out, lr,
                                                lambda, n4 * 4, n);
}

void mv_launch_dcasgda_copy(float* data, float* bak, float* msq,
                            const float* delta, float* out, float lr,
                            float lambda, float rho, float eps, int64_t n,
                            hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_dcasgda_copy_f4<<<grid_copy(n4), COPY_BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)bak, (v4f*)msq, (const v4f*)delta, (v4f*)out, lr,
      lambda, rho, eps, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_dcasgda_copy_tail<<<1, 64, 0, s>>>(data, bak, msq, delta, out,
                                                 lr, lambda, rho, eps,
                                                 n4 * 4, n);
}

void mv_launch_dcasgd(float* data, float* bak, const float* delta,
                      float lr, float lambda, int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_dcasgd_f4<<<grid_for_cap(n4, STATE_GRID), BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)bak, (const v4f*)delta, lr, lambda, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_dcasgd_tail<<<1, 64, 0, s>>>(data, bak, delta, lr, lambda,
                                           n4 * 4, n);
}

void mv_launch_dcasgda(float* data, float* bak, float* msq, const float* delta,
                       float lr, float lambda, float rho, float eps,
                       int64_t n, hipStream_t s) {
  int64_t n4 = n / 4;
  if (n4) k_dcasgda_f4<<<grid_for_cap(n4, STATE_GRID), BLOCK, 0, s>>>(
      (v4f*)data, (v4f*)bak, (v4f*)msq, (const v4f*)delta, lr, lambda, rho,
      eps, n4);
  int64_t tail = n - n4 * 4;
  if (tail) k_dcasgda_tail<<<1, 64, 0, s>>>(data, bak, msq, delta, lr, lambda,
                                            rho, eps, n4 * 4, n);
}

void mv_launch_row_gather(float* out, const float* shard, const int64_t* rows,
                          int64_t nrows, int64_t cols, hipStream_t s) {
  if (!nrows || !cols) return;
  int64_t total = nrows * cols;
  if (cols % 4 == 0 && total / 4 <= UINT32_MAX) {
    k_row_gather_f4_u32<<<grid_for(total / 4), BLOCK, 0, s>>>(
        (float4*)out, (const float4*)shard, rows, (uint32_t)(total / 4),
        (uint32_t)(cols / 4));
  } else if (cols % 4 == 0) {
    int64_t cols4 = cols / 4;
    k_row_gather_f4<<<grid_for(nrows * cols4), BLOCK, 0, s>>>(
        (float4*)out, (const float4*)shard, rows, nrows, cols4);
  } else if (cols % 2 == 0 && total / 2 <= UINT32_MAX) {
    k_row_gather_f2_u32<<<grid_for(total / 2), BLOCK, 0, s>>>(
        (float2*)out, (const float2*)shard, rows, (uint32_t)(total / 2),
        (uint32_t)(cols / 2));
  } else if (total <= UINT32_MAX) {
    k_row_gather_u32<<<grid_for(total), BLOCK, 0, s>>>(
        out, shard, rows, (uint32_t)total, (uint32_t)cols);
  } else {
    k_row_gather<<<grid_for(total), BLOCK, 0, s>>>(
        out, shard, rows, nrows, cols);
  }
}

void mv_launch_row_scatter_add(float* shard, const float* vals,
                               const int64_t* rows, float sign,
                               int64_t nrows, int64_t cols,
                               int assume_unique, hipStream_t s) {
  if (!nrows || !cols) return;
  int64_t total = nrows * cols;
  if (total <= UINT32_MAX) {
    if (assume_unique)
      k_row_scatter_add_u_u32<<<grid_for(total), BLOCK, 0, s>>>(
          shard, vals, rows, sign, (uint32_t)total, (uint32_t)cols);
    else
      k_row_scatter_add_u32<<<grid_for(total), BLOCK, 0, s>>>(
          shard, vals, rows, sign, (uint32_t)total, (uint32_t)cols);
  } else if (assume_unique) {
    k_row_scatter_add_u<<<grid_for(total), BLOCK, 0, s>>>(
        shard, vals, rows, sign, nrows, cols);
  } else {
    k_row_scatter_add<<<grid_for(total), BLOCK, 0, s>>>(
        shard, vals, rows, sign, nrows, cols);
  }
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Fused word2vec training kernel (K9+K10+K11):
// the reference's FeedForward / BPOutputLayer / context-embedding update
// (Applications/WordEmbedding/src/wordembedding.cpp:57-166) fused into ONE
// kernel over a block of training groups.
//
// A group = one (input set, output set) sample: skip-gram has 1 input and
// 1+neg outputs (labels 1,0,..); CBOW has window inputs; hierarchical
// softmax passes labels = 1-code over the Huffman path nodes. The caller
// maps global word ids to block-local rows of the gathered in_emb/out_emb
// buffers (the PS "requested parameters" of communicator.cpp:117-155).
//
// Geometry: one 64-lane wave per group, grid-stride over groups; each lane
// owns columns lane, lane+64, ... (DPL = ceil(dim/64) registers). The dot
// product reduces across the wave with __shfl_xor. Row updates use
// device-scope atomicAdd: waves share hot vocabulary rows (the reference
// used unsynchronized hogwild OpenMP updates; atomics keep the sum exact).
// ADAGRAD variant implements the app-side per-element accumulator math of
// wordembedding.cpp:101-165 (G += g^2; w += g*lr0/rsqrt(G) when G>1e-10).
// ---------------------------------------------------------------------------

template <int DPL, bool ADAGRAD, bool ATOMIC>
__global__ void k_w2v(float* __restrict__ in_emb, float* __restrict__ out_emb,
                      float* __restrict__ in_gsq, float* __restrict__ out_gsq,
                      const int64_t* __restrict__ in_idx,
                      const int* __restrict__ in_off,
                      const int64_t* __restrict__ out_idx,
                      const float* __restrict__ out_label,
                      const int* __restrict__ out_off,
                      float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    float h[DPL], err[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) { h[d] = 0.f; err[d] = 0.f; }
    int ib = in_off[g], ie = in_off[g + 1];
    for (int i = ib; i < ie; ++i) {
      const float* row = in_emb + in_idx[i] * dim;
      // clamped UNCONDITIONAL loads: a divergent `if (c < dim)` around
      // the load compiled to a branch + s_waitcnt vmcnt(0) PER DPL
      // GROUP (full-latency serial, the round-2 ladder disease).
      // min() keeps the read in-bounds; inactive lanes accumulate a
      // duplicate element that nothing reads (wv[d]=0 gates every use).
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        h[d] += row[c < dim ? c : dim - 1];
      }
    }
    if (ie - ib > 1) {
      float inv = 1.f / (float)(ie - ib);
#pragma unroll
      for (int d = 0; d < DPL; ++d) h[d] *= inv;
    }
    int ob = out_off[g], oe = out_off[g + 1];
    for (int o = ob; o < oe; ++o) {
      float* w = out_emb + out_idx[o] * dim;
      float wv[DPL];
      float f = 0.f;
      // unconditional clamped load + select (see the h loop above):
      // wv[d] MUST be 0 for inactive lanes (it feeds the wave-reduced
      // dot), so the select stays — but on the value, not the load
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        float x = w[c < dim ? c : dim - 1];
        wv[d] = (c < dim) ? x : 0.f;
        f += h[d] * wv[d];
      }
#pragma unroll
      for (int s = 32; s; s >>= 1) f += __shfl_xor(f, s, 64);
      f = 1.f / (1.f + expf(-f));
      float e = out_label[o] - f;  // NS: label-f; HS: caller passes 1-code
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          err[d] += e * wv[d];  // hidden_err uses the PRE-update classifier
          if (ADAGRAD) {
            float gg = e * h[d];
            float* gq = out_gsq + out_idx[o] * dim + c;
            float G2;
            if (ATOMIC) G2 = atomicAdd(gq, gg * gg) + gg * gg;
            else { G2 = *gq + gg * gg; *gq = G2; }
            float dw = (G2 > 1e-10f) ? gg * lr * __frsqrt_rn(G2) : 0.f;
            if (ATOMIC) atomicAdd(&w[c], dw);
            else w[c] = wv[d] + dw;
          } else {
            if (ATOMIC) atomicAdd(&w[c], e * lr * h[d]);
            else w[c] = wv[d] + e * lr * h[d];
          }
        }
      }
    }
    for (int i = ib; i < ie; ++i) {
      float* row = in_emb + in_idx[i] * dim;
      if (!ADAGRAD && !ATOMIC) {
        // hogwild-default path: branch-free rmw — unconditional clamped
        // loads (issue back-to-back under one wait), guarded stores
        float cur[DPL];
#pragma unroll
        for (int d = 0; d < DPL; ++d) {
          int c = lane + 64 * d;
          cur[d] = row[c < dim ? c : dim - 1];
        }
#pragma unroll
        for (int d = 0; d < DPL; ++d) {
          int c = lane + 64 * d;
          if (c < dim) row[c] = cur[d] + lr * err[d];
        }
        continue;
      }
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          if (ADAGRAD) {
            float* gq = in_gsq + in_idx[i] * dim + c;
            float G2;
            if (ATOMIC) G2 = atomicAdd(gq, err[d] * err[d]) + err[d] * err[d];
            else { G2 = *gq + err[d] * err[d]; *gq = G2; }
            float dw = (G2 > 1e-10f) ? err[d] * lr * __frsqrt_rn(G2) : 0.f;
            if (ATOMIC) atomicAdd(&row[c], dw);
            else row[c] += dw;
          } else {
            if (ATOMIC) atomicAdd(&row[c], lr * err[d]);
            else row[c] += lr * err[d];
          }
        }
      }
    }
  }
}

// Negative-sampling variant with IN-KERNEL negative generation: for
// NS-only training (no hierarchical softmax) the output list of group g
// is implicit — 1 positive (centers[g], label 1) + `neg` draws from the
// per-block pool with the reference's own LCG scheme
// (wordembedding.cpp:276-279: next_random = next_random*25214903917+11;
// index = (next_random >> 8) % pool_size; skip if == positive). This
// removes the host-side randint/gather/label/ragged-offset build and
// shrinks the per-group index traffic from (1+neg) ids to 1.
template <int DPL, bool ADAGRAD, bool ATOMIC>
__global__ void k_w2v_ns(float* __restrict__ in_emb, float* __restrict__ out_emb,
                         float* __restrict__ in_gsq, float* __restrict__ out_gsq,
                         const int64_t* __restrict__ in_idx,
                         const int* __restrict__ in_off,
                         const int64_t* __restrict__ centers,
                         const int64_t* __restrict__ pool, int64_t pool_n,
                         int neg, uint64_t seed,
                         float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    float h[DPL], err[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) { h[d] = 0.f; err[d] = 0.f; }
    // in_off == nullptr: skip-gram (exactly one input per group) — the
    // offset loads vanish (probe V7: 5% over the ragged form)
    int ib = in_off ? in_off[g] : g, ie = in_off ? in_off[g + 1] : g + 1;
    for (int i = ib; i < ie; ++i) {
      const float* row = in_emb + in_idx[i] * dim;
      // clamped UNCONDITIONAL loads: a divergent `if (c < dim)` around
      // the load compiled to a branch + s_waitcnt vmcnt(0) PER DPL
      // GROUP (full-latency serial, the round-2 ladder disease).
      // min() keeps the read in-bounds; inactive lanes accumulate a
      // duplicate element that nothing reads (wv[d]=0 gates every use).
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        h[d] += row[c < dim ? c : dim - 1];
      }
    }
    if (ie - ib > 1) {
      float inv = 1.f / (float)(ie - ib);
#pragma unroll
      for (int d = 0; d < DPL; ++d) h[d] *= inv;
    }
    int64_t pos = centers[g];
    // per-group random stream (same recurrence as util.cpp:144)
    uint64_t next_random = seed + (uint64_t)g * 25214903917ull + 11ull;
    for (int o = 0; o <= neg; ++o) {
      int64_t node;
      float label;
      if (o == 0) {
        node = pos; label = 1.f;
      } else {
        next_random = next_random * 25214903917ull + 11ull;
        node = pool[(int64_t)((next_random >> 8) % (uint64_t)pool_n)];
        label = 0.f;
        if (node == pos) continue;  // wordembedding.cpp:279
      }
      float* w = out_emb + node * dim;
      float wv[DPL];
      float f = 0.f;
      // unconditional clamped load + select (see the h loop above):
      // wv[d] MUST be 0 for inactive lanes (it feeds the wave-reduced
      // dot), so the select stays — but on the value, not the load
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        float x = w[c < dim ? c : dim - 1];
        wv[d] = (c < dim) ? x : 0.f;
        f += h[d] * wv[d];
      }
#pragma unroll
      for (int s = 32; s; s >>= 1) f += __shfl_xor(f, s, 64);
      f = 1.f / (1.f + expf(-f));
      float e = label - f;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          err[d] += e * wv[d];
          if (ADAGRAD) {
            float gg = e * h[d];
            float* gq = out_gsq + node * dim + c;
            float G2;
            if (ATOMIC) G2 = atomicAdd(gq, gg * gg) + gg * gg;
            else { G2 = *gq + gg * gg; *gq = G2; }
            float dw = (G2 > 1e-10f) ? gg * lr * __frsqrt_rn(G2) : 0.f;
            if (ATOMIC) atomicAdd(&w[c], dw);
            else w[c] = wv[d] + dw;
          } else {
            if (ATOMIC) atomicAdd(&w[c], e * lr * h[d]);
            else w[c] = wv[d] + e * lr * h[d];
          }
        }
      }
    }
    for (int i = ib; i < ie; ++i) {
      float* row = in_emb + in_idx[i] * dim;
      if (!ADAGRAD && !ATOMIC) {
        // hogwild-default path: branch-free rmw — unconditional clamped
        // loads (issue back-to-back under one wait), guarded stores
        float cur[DPL];
#pragma unroll
        for (int d = 0; d < DPL; ++d) {
          int c = lane + 64 * d;
          cur[d] = row[c < dim ? c : dim - 1];
        }
#pragma unroll
        for (int d = 0; d < DPL; ++d) {
          int c = lane + 64 * d;
          if (c < dim) row[c] = cur[d] + lr * err[d];
        }
        continue;
      }
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          if (ADAGRAD) {
            float* gq = in_gsq + in_idx[i] * dim + c;
            float G2;
            if (ATOMIC) G2 = atomicAdd(gq, err[d] * err[d]) + err[d] * err[d];
            else { G2 = *gq + err[d] * err[d]; *gq = G2; }
            float dw = (G2 > 1e-10f) ? err[d] * lr * __frsqrt_rn(G2) : 0.f;
            if (ATOMIC) atomicAdd(&row[c], dw);
            else row[c] += dw;
          } else {
            if (ATOMIC) atomicAdd(&row[c], lr * err[d]);
            else row[c] += lr * err[d];
          }
        }
      }
    }
  }
}

// Hogwild (plain-store) row updates by default — the reference's own
// unsynchronized OpenMP trainers race identically (wordembedding.cpp
// trainers share rows with no locks); on MI355X plain stores measured
// 11x faster than per-element atomics (tools/probe_w2v.hip: 4.6 ms vs
// 52.8 ms per 2.1M-group launch). use_atomic=1 restores exact
// accumulation for strict tests.
extern "C" void mv_launch_w2v(float* in_emb, float* out_emb,
                              float* in_gsq, float* out_gsq,
                              const int64_t* in_idx, const int* in_off,
                              const int64_t* out_idx, const float* out_label,
                              const int* out_off, float lr, int64_t G,
                              int64_t dim, int use_adagrad, int use_atomic,
                              hipStream_t s) {
  if (!G) return;
  int grid = grid_for(G * 64);  // 4 waves per 256-thread block
  int dpl = (int)((dim + 63) / 64);
#define W2V_LAUNCH(D, A, AT)                                                 \
  k_w2v<D, A, AT><<<grid, BLOCK, 0, s>>>(in_emb, out_emb, in_gsq, out_gsq,   \
      in_idx, in_off, out_idx, out_label, out_off, lr, (int)G, (int)dim)
#define W2V_CASE(D)                                                          \
  case D:                                                                    \
    if (use_adagrad) {                                                       \
      if (use_atomic) W2V_LAUNCH(D, true, true);                             \
      else W2V_LAUNCH(D, true, false);                                       \
    } else {                                                                 \
      if (use_atomic) W2V_LAUNCH(D, false, true);                            \
      else W2V_LAUNCH(D, false, false);                                      \
    }                                                                        \
    break;
  // dpl > 8 rounds up to the next instantiated register bucket (the
  // kernel guards c < dim, so oversize DPL only costs idle iterations);
  // beyond 2048 columns the per-lane register arrays (3*DPL VGPRs)
  // would spill — refuse LOUDLY instead of silently doing nothing
  // (VERDICT r1 weak #6).
  if (dpl > 8) dpl = (dpl <= 12) ? 12 : (dpl <= 16) ? 16
                   : (dpl <= 24) ? 24 : (dpl <= 32) ? 32 : dpl;
  switch (dpl) {
    W2V_CASE(1) W2V_CASE(2) W2V_CASE(3) W2V_CASE(4)
    W2V_CASE(5) W2V_CASE(6) W2V_CASE(7) W2V_CASE(8)
    W2V_CASE(12) W2V_CASE(16) W2V_CASE(24) W2V_CASE(32)
    default:
      fprintf(stderr,
              "mv_launch_w2v: dim=%lld exceeds the 2048-column kernel "
              "limit\n", (long long)dim);
      abort();
  }
#undef W2V_CASE
#undef W2V_LAUNCH
}

extern "C" void mv_launch_w2v_ns(float* in_emb, float* out_emb,
                                 float* in_gsq, float* out_gsq,
                                 const int64_t* in_idx, const int* in_off,
                                 const int64_t* centers,
                                 const int64_t* pool, int64_t pool_n,
                                 int neg, uint64_t seed, float lr, int64_t G,
                                 int64_t dim, int use_adagrad, int use_atomic,
                                 hipStream_t s) {
  if (!G) return;
  int grid = grid_for(G * 64);
  int dpl = (int)((dim + 63) / 64);
#define W2VNS_LAUNCH(D, A, AT)                                               \
  k_w2v_ns<D, A, AT><<<grid, BLOCK, 0, s>>>(in_emb, out_emb, in_gsq,         \
      out_gsq, in_idx, in_off, centers, pool, pool_n, neg, seed, lr,         \
      (int)G, (int)dim)
#define W2VNS_CASE(D)                                                        \
  case D:                                                                    \
    if (use_adagrad) {                                                       \
      if (use_atomic) W2VNS_LAUNCH(D, true, true);                           \
      else W2VNS_LAUNCH(D, true, false);                                     \
    } else {                                                                 \
      if (use_atomic) W2VNS_LAUNCH(D, false, true);                          \
      else W2VNS_LAUNCH(D, false, false);                                    \
    }                                                                        \
    break;
  if (dpl > 8) dpl = (dpl <= 12) ? 12 : (dpl <= 16) ? 16
                   : (dpl <= 24) ? 24 : (dpl <= 32) ? 32 : dpl;
  switch (dpl) {
    W2VNS_CASE(1) W2VNS_CASE(2) W2VNS_CASE(3) W2VNS_CASE(4)
    W2VNS_CASE(5) W2VNS_CASE(6) W2VNS_CASE(7) W2VNS_CASE(8)
    W2VNS_CASE(12) W2VNS_CASE(16) W2VNS_CASE(24) W2VNS_CASE(32)
    default:
      fprintf(stderr,
              "mv_launch_w2v_ns: dim=%lld exceeds the 2048-column kernel "
              "limit\n", (long long)dim);
      abort();
  }
#undef W2VNS_CASE
#undef W2VNS_LAUNCH
}

// ---------------------------------------------------------------------------
// Fused sparse logistic regression minibatch (K13/K14 fused): the
// reference's per-sample scalar loops (LogisticRegression
// objective/objective.cpp:63-188 sigmoid path + updater apply) as TWO
// kernels per minibatch instead of ~20 torch ops.
//
// Forward: one 64-lane wave per sample walks its CSR segment
// [ptr[i], ptr[i+1]), accumulates s = sum(vals_j * w[keys_j]), wave-
// reduces, then err[i] = (sigmoid(s) - y_i) * wt_i and the log loss.
// Scatter: same geometry; lanes apply w[keys_j] -= lr * (vals_j * err_i
// + reg(w)) with device atomics (duplicate keys accumulate, matching
// index_add semantics). reg: 0 none, 1 = l1 coef*sign(w), 2 = l2 coef*w.
// ---------------------------------------------------------------------------

__global__ void k_lr_sigmoid_fwd(const float* __restrict__ w,
                                 const int64_t* __restrict__ keys,
                                 const float* __restrict__ vals,
                                 const int* __restrict__ ptr,
                                 const float* __restrict__ labels,
                                 const float* __restrict__ wts,
                                 float* __restrict__ err,
                                 float* __restrict__ loss, int B) {
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int i = wid; i < B; i += nwaves) {
    int jb = ptr[i], je = ptr[i + 1];
    float s = 0.f;
    for (int j = jb + lane; j < je; j += 64) s += vals[j] * w[keys[j]];
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) s += __shfl_xor(s, sh, 64);
    if (lane == 0) {
      float p = 1.f / (1.f + expf(-s));
      float y = labels[i];
      float e = p - y;
      if (wts) e *= wts[i];
      err[i] = e;
      const float eps = 1e-12f;
      loss[i] = -(y * logf(p + eps) + (1.f - y) * logf(1.f - p + eps));
    }
  }
}

__global__ void k_lr_sigmoid_scatter(float* __restrict__ w,
                                     const int64_t* __restrict__ keys,
                                     const float* __restrict__ vals,
                                     const int* __restrict__ ptr,
                                     const float* __restrict__ err,
                                     float lr, int reg_type, float reg_coef,
                                     int B) {
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int i = wid; i < B; i += nwaves) {
    int jb = ptr[i], je = ptr[i + 1];
    float e = err[i];
    for (int j = jb + lane; j < je; j += 64) {
      int64_t k = keys[j];
      float g = vals[j] * e;
      if (reg_type == 1) {
        float wv = w[k];
        g += reg_coef * ((wv > 0.f) - (wv < 0.f));
      } else if (reg_type == 2) {
        g += reg_coef * w[k];
      }
      atomicAdd(&w[k], -lr * g);
    }
  }
}

extern "C" void mv_launch_lr_sigmoid_fwd(
    const float* w, const int64_t* keys, const float* vals, const int* ptr,
    const float* labels, const float* wts, float* err, float* loss,
    int64_t B, hipStream_t s) {
  if (!B) return;
  k_lr_sigmoid_fwd<<<grid_for(B * 64), BLOCK, 0, s>>>(w, keys, vals, ptr,
                                                      labels, wts, err, loss,
                                                      (int)B);
}

extern "C" void mv_launch_lr_sigmoid_scatter(
    float* w, const int64_t* keys, const float* vals, const int* ptr,
    const float* err, float lr, int reg_type, float reg_coef, int64_t B,
    hipStream_t s) {
  if (!B) return;
  k_lr_sigmoid_scatter<<<grid_for(B * 64), BLOCK, 0, s>>>(
      w, keys, vals, ptr, err, lr, reg_type, reg_coef, (int)B);
}

// ---------------------------------------------------------------------------
// Keyed AdaGrad scatter-update (K15 + K4 keyed form): for each incoming
// (row, value) pair apply the adagrad step to the owned shard row. Used by
// the LogisticRegression sparse path (BASELINE config: 1e9 sparse
// features, AdaGrad updater). Duplicate rows race benignly (hogwild):
// G accumulates via atomicAdd; the weight step uses the post-add G.
// ---------------------------------------------------------------------------

__global__ void k_row_scatter_adagrad(float* __restrict__ shard,
                                      float* __restrict__ gsq,
                                      const float* __restrict__ vals,
                                      const int64_t* __restrict__ rows,
                                      float inv_lr, float rho, float eps,
                                      int64_t nrows, int64_t cols) {
  int64_t total = nrows * cols;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int64_t r = i / cols, c = i % cols;
    int64_t k = rows[r] * cols + c;
    float g = vals[i] * inv_lr;
    if (g != 0.0f) {
      float G = atomicAdd(&gsq[k], g * g) + g * g;
      atomicAdd(&shard[k], -rho * g * __frsqrt_rn(G + eps));
    }
  }
}

__global__ void k_row_scatter_adagrad_u(float* __restrict__ shard,
                                        float* __restrict__ gsq,
                                        const float* __restrict__ vals,
                                        const int64_t* __restrict__ rows,
                                        float inv_lr, float rho, float eps,
                                        int64_t nrows, int64_t cols) {
  int64_t total = nrows * cols;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int64_t r = i / cols, c = i % cols;
    int64_t k = rows[r] * cols + c;
    float g = vals[i] * inv_lr;
    if (g != 0.0f) {
      float G = gsq[k] + g * g;
      gsq[k] = G;
      shard[k] -= rho * g * __frsqrt_rn(G + eps);
    }
  }
}

// u32 loop-index forms (one 64-bit divide per element is real VALU
// time; shard/gsq offsets stay 64-bit for 1e9-row tables)
__global__ void k_row_scatter_adagrad_u32(float* __restrict__ shard,
                                          float* __restrict__ gsq,
                                          const float* __restrict__ vals,
                                          const int64_t* __restrict__ rows,
                                          float inv_lr, float rho, float eps,
                                          uint32_t total, uint32_t cols) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    uint32_t r = i / cols, c = i - r * cols;
    int64_t k = rows[r] * (int64_t)cols + c;
    float g = vals[i] * inv_lr;
    if (g != 0.0f) {
      float G = atomicAdd(&gsq[k], g * g) + g * g;
      atomicAdd(&shard[k], -rho * g * __frsqrt_rn(G + eps));
    }
  }
}

__global__ void k_row_scatter_adagrad_u_u32(float* __restrict__ shard,
                                            float* __restrict__ gsq,
                                            const float* __restrict__ vals,
                                            const int64_t* __restrict__ rows,
                                            float inv_lr, float rho,
                                            float eps, uint32_t total,
                                            uint32_t cols) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    uint32_t r = i / cols, c = i - r * cols;
    int64_t k = rows[r] * (int64_t)cols + c;
    float g = vals[i] * inv_lr;
    if (g != 0.0f) {
      float G = gsq[k] + g * g;
      gsq[k] = G;
      shard[k] -= rho * g * __frsqrt_rn(G + eps);
    }
  }
}

extern "C" void mv_launch_row_scatter_adagrad(
    float* shard, float* gsq, const float* vals, const int64_t* rows,
    float lr, float rho, float eps, int64_t nrows, int64_t cols,
    int assume_unique, hipStream_t s) {
  if (!nrows || !cols) return;
  int64_t total = nrows * cols;
  if (total <= UINT32_MAX) {
    if (assume_unique)
      k_row_scatter_adagrad_u_u32<<<grid_for(total), BLOCK, 0, s>>>(
          shard, gsq, vals, rows, 1.0f / lr, rho, eps, (uint32_t)total,
          (uint32_t)cols);
    else
      k_row_scatter_adagrad_u32<<<grid_for(total), BLOCK, 0, s>>>(
          shard, gsq, vals, rows, 1.0f / lr, rho, eps, (uint32_t)total,
          (uint32_t)cols);
  } else if (assume_unique) {
    k_row_scatter_adagrad_u<<<grid_for(total), BLOCK, 0, s>>>(
        shard, gsq, vals, rows, 1.0f / lr, rho, eps, nrows, cols);
  } else {
    k_row_scatter_adagrad<<<grid_for(total), BLOCK, 0, s>>>(
        shard, gsq, vals, rows, 1.0f / lr, rho, eps, nrows, cols);
  }
}

// ---------------------------------------------------------------------------
// Fused multiclass softmax minibatch (K13 softmax path,
// objective.cpp:193-230): one 64-lane wave per sample. Forward: each lane
// walks the sample's CSR features strided by 64, accumulating all K class
// logits in registers (K <= 64); per-class wave reduction via shfl_xor;
// lane k then owns class k for the softmax (max/sum reduced the same way)
// and writes err[i*K+k] = (p_k - [k==y]) * wt and lane 0 the NLL loss.
//
// MFMA verdict: this op is a gather-dominated GEMV batch —
// arithmetic intensity 2K FLOP per (4B value + K*4B gathered row) = 0.5
// FLOP/B at any K, ~150x below the f32-MFMA ridge (157 TF / 8 TB/s = 20
// FLOP/B); staging gathered rows through LDS to feed v_mfma_f32_16x16x4_f32
// adds LDS round-trips without reducing HBM bytes, so matrix cores cannot
// help. Measured confirmation in profiles/ (MemUnitBusy ~ 100%, VALU low).
// docs/ENGINEERING_NOTES.md carries the full argument.
// ---------------------------------------------------------------------------

// KKC != 0 pins the runtime class count at compile time: the `k < KK`
// guards in the unrolled row loop then constant-fold away.  With a
// runtime KK the compiler emits a PER-CLASS BRANCH LADDER with a full
// `s_waitcnt vmcnt(0)` after every gathered-row load — each feature's
// row read serializes at full memory latency (the same disease measured
// at 5x on k_lr_dense_fwd; see profiles/round2_dense_fused.md).  The
// launcher dispatches exact K for K <= 16, guarded buckets beyond.
template <int K, int KKC = 0>
__global__ void k_lr_softmax_fwd(const float* __restrict__ w,
                                 const int64_t* __restrict__ keys,
                                 const float* __restrict__ vals,
                                 const int* __restrict__ ptr,
                                 const float* __restrict__ labels,
                                 const float* __restrict__ wts,
                                 float* __restrict__ err,
                                 float* __restrict__ loss, int B, int KK) {
  if (KKC != 0) KK = KKC;
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int i = wid; i < B; i += nwaves) {
    int jb = ptr[i], je = ptr[i + 1];
    float l[K];
#pragma unroll
    for (int k = 0; k < K; ++k) l[k] = 0.f;
    for (int j = jb + lane; j < je; j += 64) {
      float v = vals[j];
      const float* row = w + keys[j] * KK;
#pragma unroll
      for (int k = 0; k < K; ++k)
        if (k < KK) l[k] += v * row[k];
    }
#pragma unroll
    for (int k = 0; k < K; ++k)
#pragma unroll
      for (int sh = 32; sh; sh >>= 1) l[k] += __shfl_xor(l[k], sh, 64);
    // lane k owns class k from here (logits identical on all lanes);
    // unrolled select keeps l[] in registers (no dynamic indexing)
    float mine = -1e30f;
#pragma unroll
    for (int k = 0; k < K; ++k)
      if (k < KK && lane == k) mine = l[k];
    float mx = mine;
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) mx = fmaxf(mx, __shfl_xor(mx, sh, 64));
    float e = (lane < KK) ? expf(mine - mx) : 0.f;
    float se = e;
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) se += __shfl_xor(se, sh, 64);
    float p = e / se;
    int y = (int)labels[i];
    float wt = wts ? wts[i] : 1.f;
    if (lane < KK) err[(int64_t)i * KK + lane] = (p - (lane == y)) * wt;
    float py = __shfl(p, y, 64);
    if (lane == 0) loss[i] = -logf(py + 1e-12f);
  }
}

__global__ void k_lr_softmax_scatter(float* __restrict__ w,
                                     const int64_t* __restrict__ keys,
                                     const float* __restrict__ vals,
                                     const int* __restrict__ ptr,
                                     const float* __restrict__ err,
                                     float lr, int reg_type, float reg_coef,
                                     int B, int K) {
  // flattened (feature, class) pairs per sample: lane t handles pair
  // t, t+64, ... -> consecutive lanes hit consecutive classes of the
  // same w row (coalesced atomics)
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int i = wid; i < B; i += nwaves) {
    int jb = ptr[i], je = ptr[i + 1];
    // u32 pair math: nnz*K < 2^32 always (B*nnz is int32 CSR already);
    // a 64-bit div per pair was ~40 VALU cycles each
    uint32_t pairs = (uint32_t)(je - jb) * (uint32_t)K;
    const float* e = err + (int64_t)i * K;
    for (uint32_t t = lane; t < pairs; t += 64) {
      uint32_t q = t / (uint32_t)K;
      int j = jb + (int)q;
      int k = (int)(t - q * (uint32_t)K);
      int64_t idx = keys[j] * K + k;
      float g = vals[j] * e[k];
      if (reg_type == 1) {
        float wv = w[idx];
        g += reg_coef * ((wv > 0.f) - (wv < 0.f));
      } else if (reg_type == 2) {
        g += reg_coef * w[idx];
      }
      atomicAdd(&w[idx], -lr * g);
    }
  }
}

extern "C" void mv_launch_lr_softmax_fwd(
    const float* w, const int64_t* keys, const float* vals, const int* ptr,
    const float* labels, const float* wts, float* err, float* loss,
    int64_t B, int64_t K, hipStream_t s) {
  if (!B) return;
  int grid = grid_for(B * 64);
#define SMAX_ARGS w, keys, vals, ptr, labels, wts, err, loss, (int)B, (int)K
#define SMAX_EXACT(KC)                                                       \
  case KC:                                                                   \
    k_lr_softmax_fwd<KC, KC><<<grid, BLOCK, 0, s>>>(SMAX_ARGS);              \
    return
  switch (K) {
    SMAX_EXACT(2); SMAX_EXACT(3); SMAX_EXACT(4); SMAX_EXACT(5);
    SMAX_EXACT(6); SMAX_EXACT(7); SMAX_EXACT(8); SMAX_EXACT(9);
    SMAX_EXACT(10); SMAX_EXACT(11); SMAX_EXACT(12); SMAX_EXACT(13);
    SMAX_EXACT(14); SMAX_EXACT(15); SMAX_EXACT(16);
    default: break;
  }
  if (K <= 32) k_lr_softmax_fwd<32><<<grid, BLOCK, 0, s>>>(SMAX_ARGS);
  else if (K <= 64) k_lr_softmax_fwd<64><<<grid, BLOCK, 0, s>>>(SMAX_ARGS);
  else __builtin_trap();  // K > 64: host refuses before launch
#undef SMAX_EXACT
#undef SMAX_ARGS
}

extern "C" void mv_launch_lr_softmax_scatter(
    float* w, const int64_t* keys, const float* vals, const int* ptr,
    const float* err, float lr, int reg_type, float reg_coef, int64_t B,
    int64_t K, hipStream_t s) {
  if (!B) return;
  k_lr_softmax_scatter<<<grid_for(B * 64), BLOCK, 0, s>>>(
      w, keys, vals, ptr, err, lr, reg_type, reg_coef, (int)B, (int)K);
}

// ---------------------------------------------------------------------------
// Fused FTRL minibatch (objective.cpp:250-345 + updater.cpp:79-101):
// state rows carry (z | n) interleaved as [row*2K + k] = z_k and
// [row*2K + K + k] = n_k, exactly the app's (z|n) column layout.
//
// Forward: wave per sample; each lane walks features strided, per class
// reconstructs w from (z, n) — w = (sign(z)*l1 - z) / ((beta+sqrt(n))/
// alpha + l2) when |z| > l1 else 0 — accumulates the K logits, wave-
// reduces, sigmoid -> err (FTRL is a sigmoid objective in the reference).
// Scatter: per (feature, class) pair, g = val*err; dz = (sqrt(n+g^2) -
// sqrt(n))/alpha * w - g; dn = -g^2; state -= (dz | dn) applied in place
// (the local chunk buffer; the push is pulled-local, server 'sgd'
// subtracts). Duplicate keys inside one minibatch race hogwild-style —
// the reference's async workers raced these identically across workers.
// ---------------------------------------------------------------------------

static __device__ __forceinline__ float ftrl_w(float z, float n,
                                               float alpha_inv, float beta,
                                               float l1, float l2) {
  float nz = fmaxf(n, 0.f);
  float sgn = (z > 0.f) - (z < 0.f);
  float w = (sgn * l1 - z) / ((beta + sqrtf(nz)) * alpha_inv + l2);
  return (fabsf(z) > l1) ? w : 0.f;
}

template <int K, int KKC = 0>
__global__ void k_lr_ftrl_fwd(const float* __restrict__ zn,
                              const int64_t* __restrict__ keys,
                              const float* __restrict__ vals,
                              const int* __restrict__ ptr,
                              const float* __restrict__ labels,
                              const float* __restrict__ wts,
                              float* __restrict__ err,
                              float* __restrict__ loss,
                              float alpha_inv, float beta, float l1,
                              float l2, int B, int KK) {
  if (KKC != 0) KK = KKC;  // see k_lr_softmax_fwd: exact-K folds guards
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int i = wid; i < B; i += nwaves) {
    int jb = ptr[i], je = ptr[i + 1];
    float l[K];
#pragma unroll
    for (int k = 0; k < K; ++k) l[k] = 0.f;
    for (int j = jb + lane; j < je; j += 64) {
      float v = vals[j];
      const float* row = zn + keys[j] * (2 * KK);
#pragma unroll
      for (int k = 0; k < K; ++k)
        if (k < KK)
          l[k] += v * ftrl_w(row[k], row[KK + k], alpha_inv, beta, l1, l2);
    }
#pragma unroll
    for (int k = 0; k < K; ++k)
#pragma unroll
      for (int sh = 32; sh; sh >>= 1) l[k] += __shfl_xor(l[k], sh, 64);
    if (lane < KK) {
      float mine = 0.f;
#pragma unroll
      for (int k = 0; k < K; ++k)
        if (lane == k) mine = l[k];
      float p = 1.f / (1.f + expf(-mine));
      int y = (int)labels[i];
      float yk = (KK == 1) ? labels[i] : (float)(lane == y);
      float wt = wts ? wts[i] : 1.f;
      err[(int64_t)i * KK + lane] = (p - yk) * wt;
      if (lane == 0) {
        // binary NLL on class-0 probability (torch path parity)
        float pp = p;
        float yy = yk;
        loss[i] = -(yy * logf(pp + 1e-12f)
                    + (1.f - yy) * logf(1.f - pp + 1e-12f));
      }
    }
  }
}

__global__ void k_lr_ftrl_scatter(float* __restrict__ zn,
                                  const int64_t* __restrict__ keys,
                                  const float* __restrict__ vals,
                                  const int* __restrict__ ptr,
                                  const float* __restrict__ err,
                                  float alpha_inv, float beta, float l1,
                                  float l2, int B, int K) {
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  for (int i = wid; i < B; i += nwaves) {
    int jb = ptr[i], je = ptr[i + 1];
    uint32_t pairs = (uint32_t)(je - jb) * (uint32_t)K;  // see softmax scatter
    const float* e = err + (int64_t)i * K;
    for (uint32_t t = lane; t < pairs; t += 64) {
      uint32_t q = t / (uint32_t)K;
      int j = jb + (int)q;
      int k = (int)(t - q * (uint32_t)K);
      int64_t base = keys[j] * (2 * (int64_t)K);
      float z = zn[base + k];
      float n = zn[base + K + k];
      float w = ftrl_w(z, n, alpha_inv, beta, l1, l2);
      float g = vals[j] * e[k];
      float g2 = g * g;
      float nc = fmaxf(n, 0.f);
      float dz = alpha_inv * (sqrtf(nc + g2) - sqrtf(nc)) * w - g;
      // local state -= (dz | dn): z -= dz, n -= -g2
      zn[base + k] = z - dz;
      zn[base + K + k] = n + g2;
    }
  }
}

extern "C" void mv_launch_lr_ftrl_fwd(
    const float* zn, const int64_t* keys, const float* vals, const int* ptr,
    const float* labels, const float* wts, float* err, float* loss,
    float alpha_inv, float beta, float l1, float l2, int64_t B, int64_t K,
    hipStream_t s) {
  if (!B) return;
  int grid = grid_for(B * 64);
#define FTRL_ARGS zn, keys, vals, ptr, labels, wts, err, loss, alpha_inv, \
                  beta, l1, l2, (int)B, (int)K
#define FTRL_EXACT(KC)                                                       \
  case KC:                                                                   \
    k_lr_ftrl_fwd<KC, KC><<<grid, BLOCK, 0, s>>>(FTRL_ARGS);                 \
    return
  switch (K) {
    FTRL_EXACT(1); FTRL_EXACT(2); FTRL_EXACT(3); FTRL_EXACT(4);
    FTRL_EXACT(5); FTRL_EXACT(6); FTRL_EXACT(7); FTRL_EXACT(8);
    FTRL_EXACT(9); FTRL_EXACT(10); FTRL_EXACT(11); FTRL_EXACT(12);
    FTRL_EXACT(13); FTRL_EXACT(14); FTRL_EXACT(15); FTRL_EXACT(16);
    default: break;
  }
  if (K <= 32) k_lr_ftrl_fwd<32><<<grid, BLOCK, 0, s>>>(FTRL_ARGS);
  else __builtin_trap();  // K > 32: host refuses before launch
#undef FTRL_EXACT
#undef FTRL_ARGS
}

extern "C" void mv_launch_lr_ftrl_scatter(
    float* zn, const int64_t* keys, const float* vals, const int* ptr,
    const float* err, float alpha_inv, float beta, float l1, float l2,
    int64_t B, int64_t K, hipStream_t s) {
  if (!B) return;
  k_lr_ftrl_scatter<<<grid_for(B * 64), BLOCK, 0, s>>>(
      zn, keys, vals, ptr, err, alpha_inv, beta, l1, l2, (int)B, (int)K);
}

// ---------------------------------------------------------------------------
// dtype-generic elementwise updaters. The reference instantiates its
// tables for int/float/double (array_table.cpp:153-154); fp32 keeps the
// tuned float4/nontemporal path above, and these scalar grid-stride
// forms carry double (all updaters) and int32/int64 (plain add — the
// reference's int specialization is add-only, updater.cpp:40-43).
// Still memory-bound: 8B scalar accesses per lane stream HBM fine for
// these parity paths.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void k_add_g(T* __restrict__ data, const T* __restrict__ delta,
                        int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    data[i] += delta[i];
}

template <typename T>
__global__ void k_sgd_g(T* __restrict__ data, const T* __restrict__ delta,
                        int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    data[i] -= delta[i];
}

template <typename T>
__global__ void k_momentum_g(T* __restrict__ data, T* __restrict__ m,
                             const T* __restrict__ delta, T mu, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    T mm = mu * m[i] + ((T)1 - mu) * delta[i];
    m[i] = mm;
    data[i] -= mm;
  }
}

template <typename T>
__global__ void k_adagrad_g(T* __restrict__ data, T* __restrict__ gsq,
                            const T* __restrict__ delta, T inv_lr, T rho,
                            T eps, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    T g = delta[i] * inv_lr;
    T G = gsq[i] + g * g;
    gsq[i] = G;
    data[i] -= rho * g / sqrt(G + eps);
  }
}

template <typename T>
__global__ void k_sgd_copy_g(T* __restrict__ data,
                             const T* __restrict__ delta,
                             T* __restrict__ out, T sign, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    T v = data[i] + sign * delta[i];
    data[i] = v;
    out[i] = v;
  }
}

extern "C" {
void mv_launch_add_f64(double* d, const double* x, int64_t n, hipStream_t s) {
  if (n > 0) k_add_g<double><<<grid_for_cap(n, ELEM_GRID), BLOCK, 0, s>>>(d, x, n);
}
void mv_launch_sgd_f64(double* d, const double* x, int64_t n, hipStream_t s) {
  if (n > 0) k_sgd_g<double><<<grid_for_cap(n, ELEM_GRID), BLOCK, 0, s>>>(d, x, n);
}
void mv_launch_momentum_f64(double* d, double* m, const double* x, double mu,
                            int64_t n, hipStream_t s) {
  if (n > 0) k_momentum_g<double><<<grid_for_cap(n, STATE_GRID), BLOCK, 0, s>>>(
      d, m, x, mu, n);
}
void mv_launch_adagrad_f64(double* d, double* g, const double* x, double lr,
                           double rho, double eps, int64_t n, hipStream_t s) {
  if (n > 0) k_adagrad_g<double><<<grid_for_cap(n, STATE_GRID), BLOCK, 0, s>>>(
      d, g, x, 1.0 / lr, rho, eps, n);
}
void mv_launch_sgd_copy_f64(double* d, const double* x, double* out,
                            double sign, int64_t n, hipStream_t s) {
  if (n > 0) k_sgd_copy_g<double><<<grid_for_cap(n, STATE_GRID), BLOCK, 0, s>>>(
      d, x, out, sign, n);
}
void mv_launch_add_i32(int32_t* d, const int32_t* x, int64_t n, hipStream_t s) {
  if (n > 0) k_add_g<int32_t><<<grid_for_cap(n, ELEM_GRID), BLOCK, 0, s>>>(d, x, n);
}
void mv_launch_add_i64(int64_t* d, const int64_t* x, int64_t n, hipStream_t s) {
  if (n > 0) k_add_g<int64_t><<<grid_for_cap(n, ELEM_GRID), BLOCK, 0, s>>>(d, x, n);
}
}  // extern "C"

// ---------------------------------------------------------------------------
// Dense-mode minibatch post-GEMM fusion (reference sparse=false — its
// mnist.config deployment, objective.cpp:193-230 dense branch).  The two
// GEMMs (scores = X@W, grad = X^T@diff) stay on rocBLAS/MFMA; this kernel
// replaces the ~10 elementwise torch ops BETWEEN them (softmax/sigmoid,
// one_hot, diff, weighting, per-sample loss, loss mean-accumulate) with
// ONE launch that turns the logits buffer into the diff buffer in place
// and atomically accumulates mean loss into a chunk-wide scalar (one
// host sync per chunk).
//
// K >= 2: softmax, one wave per sample, lane k owns class k (K <= 64);
// loss = -log(p_y + eps) (SoftmaxObjective.loss).  K == 1: sigmoid;
// loss = -(y log p + (1-y) log(1-p)) (base Objective.loss at O=1).
// ---------------------------------------------------------------------------

__global__ void k_lr_dense_post(float* __restrict__ logits,
                                const float* __restrict__ labels,
                                const float* __restrict__ wts,
                                float* __restrict__ loss_acc,
                                float inv_b, int B, int K) {
  // Loss goes lane-private -> one LDS atomic per wave -> ONE global
  // atomic per block: a naive per-sample atomicAdd on the single
  // loss_acc cache line serialized 4096 waves and cost 54 us (measured,
  // profiles/round2_dense_fused.md) — 13x the kernel's real work.
  __shared__ float block_loss;
  if (threadIdx.x == 0) block_loss = 0.f;
  __syncthreads();
  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  const float eps = 1e-12f;
  float ll_sum = 0.f;
  for (int i = wid; i < B; i += nwaves) {
    float wt = wts ? wts[i] : 1.f;
    if (K == 1) {
      if (lane == 0) {
        float s = logits[i];
        float p = 1.f / (1.f + expf(-s));
        float y = labels[i];
        logits[i] = (p - y) * wt;
        ll_sum -= y * logf(p + eps) + (1.f - y) * logf(1.f - p + eps);
      }
      continue;
    }
    float l = (lane < K) ? logits[(int64_t)i * K + lane] : -1e30f;
    float mx = l;
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) mx = fmaxf(mx, __shfl_xor(mx, sh, 64));
    float e = (lane < K) ? expf(l - mx) : 0.f;
    float se = e;
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) se += __shfl_xor(se, sh, 64);
    float p = e / se;
    int y = (int)labels[i];
    if (lane < K) logits[(int64_t)i * K + lane] = (p - (lane == y)) * wt;
    float py = __shfl(p, y, 64);
    if (lane == 0) ll_sum -= logf(py + eps);
  }
  if (lane == 0 && ll_sum != 0.f) atomicAdd(&block_loss, ll_sum);
  __syncthreads();
  if (threadIdx.x == 0 && block_loss != 0.f)
    atomicAdd(loss_acc, block_loss * inv_b);
}

extern "C" void mv_launch_lr_dense_post(float* logits, const float* labels,
                                        const float* wts, float* loss_acc,
                                        float inv_b, int64_t B, int64_t K,
                                        hipStream_t s) {
  if (!B) return;
  k_lr_dense_post<<<grid_for_cap(B * 64, 512), BLOCK, 0, s>>>(
      logits, labels, wts, loss_acc, inv_b, (int)B, (int)K);
}

// ---------------------------------------------------------------------------
// Dense-mode fused FORWARD (scores = X@W -> softmax/sigmoid -> diff +
// loss) in ONE kernel: W staged to LDS once per workgroup (d*K*4 B,
// odd-padded row stride so the per-element class reads are LDS
// bank-conflict-free), one 64-lane wave per sample, lane e-strided over
// the X row (coalesced 256 B/instr stream — X is read EXACTLY ONCE, the
// memory floor for this op).  Replaces the rocBLAS skinny GEMM
// [B,d]x[d,K<=16] whose 16x16 macro-tiles ran at 1.3 TB/s.
//
// K is an EXACT compile-time parameter (one instantiation per class
// count 1..16).  The first version carried a runtime `if (k < K)` guard
// inside the unrolled class loop; the compiler turned that into a
// per-class branch ladder with a full `s_waitcnt vmcnt(0) lgkmcnt(0)`
// after EVERY ds_read_b32 — 102 us measured (5x SLOWER than the rocBLAS
// pair it meant to replace).  Branch-free exact-K bodies batch the K
// LDS reads under one waitcnt.  Falls back to rocBLAS +
// k_lr_dense_post when K > 16 or d*(K|1)*4 exceeds the LDS budget.
// ---------------------------------------------------------------------------

template <int K, bool NT>
__global__ void k_lr_dense_fwd(const float* __restrict__ X,
                               const float* __restrict__ W,
                               const float* __restrict__ labels,
                               const float* __restrict__ wts,
                               float* __restrict__ diff,
                               float* __restrict__ loss_acc,
                               float inv_b, int B, int d) {
  extern __shared__ float wlds[];  // [d][K|1]
  __shared__ float block_loss;
  constexpr int S = K | 1;  // odd stride: (S*lane) % 64 banks all distinct
  for (int t = threadIdx.x; t < d * K; t += blockDim.x)
    wlds[(t / K) * S + (t % K)] = W[t];
  if (threadIdx.x == 0) block_loss = 0.f;
  __syncthreads();

  int wid = (int)((blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (int64_t)blockDim.x) >> 6);
  const float eps = 1e-12f;
  float ll_sum = 0.f;
  for (int i = wid; i < B; i += nwaves) {
    const float* __restrict__ xrow = X + (int64_t)i * d;
    float acc[K];
#pragma unroll
    for (int k = 0; k < K; ++k) acc[k] = 0.f;
    for (int e = lane; e < d; e += 64) {
      float x = NT ? __builtin_nontemporal_load(xrow + e) : xrow[e];
      const float* wr = wlds + e * S;
#pragma unroll
      for (int k = 0; k < K; ++k) acc[k] += x * wr[k];
    }
#pragma unroll
    for (int k = 0; k < K; ++k)
#pragma unroll
      for (int sh = 32; sh; sh >>= 1) acc[k] += __shfl_xor(acc[k], sh, 64);
    float wt = wts ? wts[i] : 1.f;
    if (K == 1) {
      if (lane == 0) {
        float p = 1.f / (1.f + expf(-acc[0]));
        float y = labels[i];
        diff[i] = (p - y) * wt;
        ll_sum -= y * logf(p + eps) + (1.f - y) * logf(1.f - p + eps);
      }
      continue;
    }
    // lane k owns class k (logits identical on all lanes after the
    // reductions); unrolled select keeps acc[] in registers
    float mine = -1e30f;
#pragma unroll
    for (int k = 0; k < K; ++k)
      if (lane == k) mine = acc[k];
    float mx = mine;
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) mx = fmaxf(mx, __shfl_xor(mx, sh, 64));
    float e = (lane < K) ? expf(mine - mx) : 0.f;
    float se = e;
#pragma unroll
    for (int sh = 32; sh; sh >>= 1) se += __shfl_xor(se, sh, 64);
    float p = e / se;
    int y = (int)labels[i];
    if (lane < K) diff[(int64_t)i * K + lane] = (p - (lane == y)) * wt;
    float py = __shfl(p, y, 64);
    if (lane == 0) ll_sum -= logf(py + eps);
  }
  if (lane == 0 && ll_sum != 0.f) atomicAdd(&block_loss, ll_sum);
  __syncthreads();
  if (threadIdx.x == 0 && block_loss != 0.f)
    atomicAdd(loss_acc, block_loss * inv_b);
}

extern "C" int mv_launch_lr_dense_fwd(const float* X, const float* W,
                                      const float* labels, const float* wts,
                                      float* diff, float* loss_acc,
                                      float inv_b, int64_t B, int64_t d,
                                      int64_t K, int nt, hipStream_t s) {
  if (!B) return 1;
  size_t lds = (size_t)d * (K | 1) * sizeof(float);
  if (K > 16 || lds > 144 * 1024) return 0;  // rocBLAS + post fallback
  // The W staging leaves LDS room for only ~1 WG/CU, so the workgroup
  // carries the CU's whole latency-hiding budget itself: 1024 threads
  // = 16 waves, one workgroup per CU, one staging pass per CU.
  int64_t blocks = (B + 15) / 16;  // 16 waves/block, one sample per wave
  if (blocks > 256) blocks = 256;  // 1 LDS-heavy WG per CU
  int grid = (int)blocks;
  switch (K) {
#define LAUNCH_KT(KT)                                                      \
  case KT:                                                                 \
    if (nt)                                                                \
      k_lr_dense_fwd<KT, true><<<grid, COPY_BLOCK, lds, s>>>(              \
          X, W, labels, wts, diff, loss_acc, inv_b, (int)B, (int)d);       \
    else                                                                   \
      k_lr_dense_fwd<KT, false><<<grid, COPY_BLOCK, lds, s>>>(             \
          X, W, labels, wts, diff, loss_acc, inv_b, (int)B, (int)d);       \
    break
    LAUNCH_KT(1); LAUNCH_KT(2); LAUNCH_KT(3); LAUNCH_KT(4);
    LAUNCH_KT(5); LAUNCH_KT(6); LAUNCH_KT(7); LAUNCH_KT(8);
    LAUNCH_KT(9); LAUNCH_KT(10); LAUNCH_KT(11); LAUNCH_KT(12);
    LAUNCH_KT(13); LAUNCH_KT(14); LAUNCH_KT(15); LAUNCH_KT(16);
#undef LAUNCH_KT
    default: return 0;
  }
  return 1;
}

