// A/B probe for the elementwise SGD updater (K2) roofline on gfx950:
// variants x grid sizes; traffic = read delta + read/write data = 12 B
// per element. Target: ~6.3 TB/s achievable HBM.
// build: hipcc --offload-arch=gfx950 -O3 tools/probe_sgd.hip -o probe_sgd
#include <hip/hip_runtime.h>
#include <cstdio>

#define BLOCK 256
typedef float v4f __attribute__((ext_vector_type(4)));

__global__ void sgd_f4(float4* __restrict__ d, const float4* __restrict__ g, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 a = d[i], b = g[i];
    a.x -= b.x; a.y -= b.y; a.z -= b.z; a.w -= b.w;
    d[i] = a;
  }
}

__global__ void sgd_f4_nt(float4* __restrict__ d, const float4* __restrict__ g, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 a = d[i];
    v4f b = __builtin_nontemporal_load((const v4f*)&g[i]);
    a.x -= b[0]; a.y -= b[1]; a.z -= b[2]; a.w -= b[3];
    d[i] = a;
  }
}

__global__ void sgd_f4_ntboth(float4* __restrict__ d, const float4* __restrict__ g, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f a = __builtin_nontemporal_load((v4f*)&d[i]);
    v4f b = __builtin_nontemporal_load((const v4f*)&g[i]);
    a -= b;
    __builtin_nontemporal_store(a, (v4f*)&d[i]);
  }
}

// fused Add+Get: read d, read g, write d, write o = 16 B/element
__global__ void sgd_copy_nt(float4* __restrict__ d, const float4* __restrict__ g,
                            float4* __restrict__ o, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f a = __builtin_nontemporal_load((v4f*)&d[i]);
    v4f b = __builtin_nontemporal_load((const v4f*)&g[i]);
    a -= b;
    __builtin_nontemporal_store(a, (v4f*)&d[i]);
    __builtin_nontemporal_store(a, (v4f*)&o[i]);
  }
}

__global__ void sgd_copy_plain(float4* __restrict__ d, const float4* __restrict__ g,
                               float4* __restrict__ o, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 a = d[i], b = g[i];
    a.x -= b.x; a.y -= b.y; a.z -= b.z; a.w -= b.w;
    d[i] = a;
    o[i] = a;
  }
}

// momentum/adagrad shape: 3 reads (data, state, delta) + 2 writes
// (data, state) = 20 B/element
__global__ void mom_nt(float4* __restrict__ d, float4* __restrict__ m,
                       const float4* __restrict__ g, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    v4f a = __builtin_nontemporal_load((v4f*)&d[i]);
    v4f mm = __builtin_nontemporal_load((v4f*)&m[i]);
    v4f b = __builtin_nontemporal_load((const v4f*)&g[i]);
    mm = 0.9f * mm + 0.1f * b;
    a -= mm;
    __builtin_nontemporal_store(mm, (v4f*)&m[i]);
    __builtin_nontemporal_store(a, (v4f*)&d[i]);
  }
}

int main() {
  long n = 128L * 1000 * 1000;  // 1e6x128
  long n4 = n / 4;
  float *d, *g;
  (void)hipMalloc(&d, n * 4);
  (void)hipMalloc(&g, n * 4);
  (void)hipMemset(d, 0, n * 4);
  (void)hipMemset(g, 0, n * 4);
  auto bench = [&](auto kern, int grid, const char* name) {
    kern<<<grid, BLOCK>>>((float4*)d, (const float4*)g, n4);
    (void)hipDeviceSynchronize();
    hipEvent_t a, b;
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    (void)hipEventRecord(a);
    for (int r = 0; r < 10; ++r)
      kern<<<grid, BLOCK>>>((float4*)d, (const float4*)g, n4);
    (void)hipEventRecord(b);
    (void)hipEventSynchronize(b);
    float ms;
    (void)hipEventElapsedTime(&ms, a, b);
    ms /= 10;
    printf("%s grid=%5d: %.3f ms  %.2f TB/s\n", name, grid, ms,
           n * 12.0 / (ms * 1e-3) / 1e12);
  };
  for (int grid : {1024, 2048, 4096, 8192, 16384}) {
    bench(sgd_f4, grid, "plain  ");
    bench(sgd_f4_nt, grid, "nt-load");
    bench(sgd_f4_ntboth, grid, "nt-both");
  }
  float* o;
  (void)hipMalloc(&o, n * 4);
  auto bench3 = [&](auto kern, int grid, const char* name) {
    kern<<<grid, BLOCK>>>((float4*)d, (const float4*)g, (float4*)o, n4);
    (void)hipDeviceSynchronize();
    hipEvent_t a, b;
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    (void)hipEventRecord(a);
    for (int r = 0; r < 5; ++r)
      kern<<<grid, BLOCK>>>((float4*)d, (const float4*)g, (float4*)o, n4);
    (void)hipEventRecord(b);
    (void)hipEventSynchronize(b);
    float ms;
    (void)hipEventElapsedTime(&ms, a, b);
    ms /= 5;
    printf("%s grid=%5d: %.3f ms  %.2f TB/s\n", name, grid, ms,
           n * 16.0 / (ms * 1e-3) / 1e12);
  };
  for (int grid : {256, 384, 512, 768, 1024, 2048}) {
    bench3(sgd_copy_nt, grid, "fused-nt   ");
  }
  // momentum shape (m reuses o as state; 20 B/element)
  auto bench_m = [&](int grid) {
    mom_nt<<<grid, BLOCK>>>((float4*)d, (float4*)o, (const float4*)g, n4);
    (void)hipDeviceSynchronize();
    hipEvent_t a, b;
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    (void)hipEventRecord(a);
    for (int r = 0; r < 5; ++r)
      mom_nt<<<grid, BLOCK>>>((float4*)d, (float4*)o, (const float4*)g, n4);
    (void)hipEventRecord(b);
    (void)hipEventSynchronize(b);
    float ms;
    (void)hipEventElapsedTime(&ms, a, b);
    ms /= 5;
    printf("mom-nt grid=%5d: %.3f ms  %.2f TB/s\n", grid, ms,
           n * 20.0 / (ms * 1e-3) / 1e12);
  };
  for (int grid : {256, 384, 512, 768, 1024, 2048}) bench_m(grid);
  return 0;
}
