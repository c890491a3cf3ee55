// Round-2 probe: the fused Add+Get kernel (k_sgd_copy_f4 shape: read
// data + read delta + write data + write out = 2.0 GB/step on the
// headline config) swept over {block size} x {per-lane unroll} x
// {grid} x {nontemporal}. Round 1 fixed BLOCK=256, 1x float4/lane,
// STATE_GRID=768; this asks whether wider per-wave requests (2x float4
// = 32 B/lane) or fatter blocks buy anything at 4 concurrent HBM
// streams.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef float v4f __attribute__((ext_vector_type(4)));
#define NT_L(p) __builtin_nontemporal_load(p)
#define NT_S(p, v) __builtin_nontemporal_store(v, p)

template <int U, bool NT>
__global__ void sgd_copy_u(v4f* __restrict__ d, const v4f* __restrict__ g,
                           v4f* __restrict__ o, long n4) {
  long stride = (long)gridDim.x * blockDim.x * U;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * U;
       i < n4; i += stride) {
#pragma unroll
    for (int u = 0; u < U; ++u) {
      long j = i + u;
      if (j < n4) {
        v4f dv = NT ? NT_L(&d[j]) : d[j];
        v4f gv = NT ? NT_L(&g[j]) : g[j];
        v4f r = dv - gv;
        if (NT) { NT_S(&d[j], r); NT_S(&o[j], r); }
        else { d[j] = r; o[j] = r; }
      }
    }
  }
}

int probe2_main();

int main() {
  long n = 128L * 1000000;  // 1e6 x 128 fp32
  long n4 = n / 4;
  float *d, *g, *o;
  hipMalloc(&d, n * 4); hipMalloc(&g, n * 4); hipMalloc(&o, n * 4);
  hipMemset(d, 0, n * 4); hipMemset(g, 0, n * 4);
  auto run = [&](auto kern, int grid, int block, const char* name) {
    kern<<<grid, block>>>((v4f*)d, (const v4f*)g, (v4f*)o, n4);
    hipDeviceSynchronize();
    hipEvent_t a, b; hipEventCreate(&a); hipEventCreate(&b);
    hipEventRecord(a);
    for (int i = 0; i < 20; ++i)
      kern<<<grid, block>>>((v4f*)d, (const v4f*)g, (v4f*)o, n4);
    hipEventRecord(b); hipEventSynchronize(b);
    float ms; hipEventElapsedTime(&ms, a, b); ms /= 20;
    printf("%s block=%4d grid=%5d: %.3f ms  %.2f TB/s\n", name, block,
           grid, ms, 4.0 * n * 4 / ms / 1e9);
  };
  for (int block : {256, 512, 1024}) {
    for (int grid : {512, 768, 1024, 1536, 2048}) {
      run(sgd_copy_u<1, true>, grid, block, "U1 nt");
      run(sgd_copy_u<2, true>, grid, block, "U2 nt");
      run(sgd_copy_u<1, false>, grid, block, "U1   ");
    }
  }
  probe2_main();
  return 0;
}

// -- appended: 2-stream (sgd: r data, r delta, w data) and copy
// (r src, w dst) shapes at the block sizes the fused probe favored --
template <bool NT>
__global__ void sgd2(v4f* __restrict__ d, const v4f* __restrict__ g,
                     long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    v4f dv = NT ? NT_L(&d[i]) : d[i];
    v4f gv = NT ? NT_L(&g[i]) : g[i];
    v4f r = dv - gv;
    if (NT) NT_S(&d[i], r); else d[i] = r;
  }
}
template <bool NT>
__global__ void cpy(v4f* __restrict__ d, const v4f* __restrict__ g,
                    long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    v4f gv = NT ? NT_L(&g[i]) : g[i];
    if (NT) NT_S(&d[i], gv); else d[i] = gv;
  }
}

int probe2_main() {
  long n = 128L * 1000000, n4 = n / 4;
  float *d, *g;
  (void)hipMalloc(&d, n * 4); (void)hipMalloc(&g, n * 4);
  (void)hipMemset(d, 0, n * 4); (void)hipMemset(g, 0, n * 4);
  auto run2 = [&](auto kern, int grid, int block, double bytes,
                  const char* name) {
    kern<<<grid, block>>>((v4f*)d, (const v4f*)g, n4);
    (void)hipDeviceSynchronize();
    hipEvent_t a, b; (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    (void)hipEventRecord(a);
    for (int i = 0; i < 20; ++i)
      kern<<<grid, block>>>((v4f*)d, (const v4f*)g, n4);
    (void)hipEventRecord(b); (void)hipEventSynchronize(b);
    float ms; (void)hipEventElapsedTime(&ms, a, b); ms /= 20;
    printf("%s block=%4d grid=%5d: %.3f ms  %.2f TB/s\n", name, block,
           grid, ms, bytes / ms / 1e9);
  };
  for (int block : {256, 1024}) {
    for (int grid : {768, 1024, 1536, 2048}) {
      run2(sgd2<true>, grid, block, 3.0 * n * 4, "sgd2 nt");
      run2(cpy<true>, grid, block, 2.0 * n * 4, "copy nt");
    }
  }
  return 0;
}
