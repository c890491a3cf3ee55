// A/B probe for the k_w2v kernel bottleneck on gfx950.
// Variants: 0 = current (scalar f32 atomics), 1 = plain stores (racy
// ceiling — perf bound only), 2 = packed v2f32 atomics (half the atomic
// transactions), 3 = no in/out updates at all (pure read+dot ceiling).
// Prints ms per launch for each variant on a Zipf-like group mix.
//
// build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/probe_w2v.hip -o gpurun_out/probe_w2v
// run:   ./gpurun_out/probe_w2v [G] [V] [dim] [neg]

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

#define BLOCK 256
#define MAX_GRID 2048

typedef float v2f __attribute__((ext_vector_type(2)));

// VARIANT 2: float2 loads/atomics — lane owns 2 consecutive floats
// (8B/lane, 512B/wave-instruction); separate body below.
template <int DPL2>
__global__ void k_w2v_probe_f2(float* __restrict__ in_emb,
                               float* __restrict__ out_emb,
                               const long* __restrict__ in_idx,
                               const int* __restrict__ in_off,
                               const long* __restrict__ out_idx,
                               const float* __restrict__ out_label,
                               const int* __restrict__ out_off,
                               float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (long)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (long)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    v2f h[DPL2], err[DPL2];
#pragma unroll
    for (int d = 0; d < DPL2; ++d) { h[d] = {0.f, 0.f}; err[d] = {0.f, 0.f}; }
    int ib = in_off[g], ie = in_off[g + 1];
    for (int i = ib; i < ie; ++i) {
      const float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL2; ++d) {
        int c = lane * 2 + 128 * d;
        if (c + 1 < dim) { v2f v = *(const v2f*)(row + c); h[d] += v; }
        else if (c < dim) { h[d].x += row[c]; }
      }
    }
    int ob = out_off[g], oe = out_off[g + 1];
    for (int o = ob; o < oe; ++o) {
      float* w = out_emb + out_idx[o] * dim;
      v2f wv[DPL2];
      float f = 0.f;
#pragma unroll
      for (int d = 0; d < DPL2; ++d) {
        int c = lane * 2 + 128 * d;
        wv[d] = {0.f, 0.f};
        if (c + 1 < dim) wv[d] = *(const v2f*)(w + c);
        else if (c < dim) wv[d].x = w[c];
        f += h[d].x * wv[d].x + h[d].y * wv[d].y;
      }
#pragma unroll
      for (int s = 32; s; s >>= 1) f += __shfl_xor(f, s, 64);
      f = 1.f / (1.f + expf(-f));
      float e = out_label[o] - f;
#pragma unroll
      for (int d = 0; d < DPL2; ++d) {
        int c = lane * 2 + 128 * d;
        err[d] += e * wv[d];
        if (c + 1 < dim) {
          atomicAdd(&w[c], e * lr * h[d].x);
          atomicAdd(&w[c + 1], e * lr * h[d].y);
        } else if (c < dim) {
          atomicAdd(&w[c], e * lr * h[d].x);
        }
      }
    }
    for (int i = ib; i < ie; ++i) {
      float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL2; ++d) {
        int c = lane * 2 + 128 * d;
        if (c + 1 < dim) {
          atomicAdd(&row[c], lr * err[d].x);
          atomicAdd(&row[c + 1], lr * err[d].y);
        } else if (c < dim) {
          atomicAdd(&row[c], lr * err[d].x);
        }
      }
    }
  }
}

template <int DPL, int VARIANT>
__global__ void k_w2v_probe(float* __restrict__ in_emb,
                            float* __restrict__ out_emb,
                            const long* __restrict__ in_idx,
                            const int* __restrict__ in_off,
                            const long* __restrict__ out_idx,
                            const float* __restrict__ out_label,
                            const int* __restrict__ out_off,
                            float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (long)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (long)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    float h[DPL], err[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) { h[d] = 0.f; err[d] = 0.f; }
    int ib = in_off[g], ie = in_off[g + 1];
    for (int i = ib; i < ie; ++i) {
      const float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) h[d] += row[c];
      }
    }
    int ob = out_off[g], oe = out_off[g + 1];
    for (int o = ob; o < oe; ++o) {
      float* w = out_emb + out_idx[o] * dim;
      float wv[DPL];
      float f = 0.f;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        wv[d] = (c < dim) ? w[c] : 0.f;
        f += h[d] * wv[d];
      }
#pragma unroll
      for (int s = 32; s; s >>= 1) f += __shfl_xor(f, s, 64);
      f = 1.f / (1.f + expf(-f));
      float e = out_label[o] - f;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          err[d] += e * wv[d];
          if (VARIANT == 0) {
            atomicAdd(&w[c], e * lr * h[d]);
          } else if (VARIANT == 1) {
            w[c] = wv[d] + e * lr * h[d];
          } else if (VARIANT == 3) {
            // no update
          }
        }
      }
    }
    for (int i = ib; i < ie; ++i) {
      float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          if (VARIANT == 0) atomicAdd(&row[c], lr * err[d]);
          else if (VARIANT == 1) row[c] += lr * err[d];
        }
      }
    }
  }
}

// VARIANT 4: plain stores + tile-of-8 output-row prefetch. All output rows
// of a tile are loaded before any dot/update, so the per-group row fetches
// overlap instead of serializing (6 dependent HBM/L2 round trips -> 1-2).
template <int DPL, int KT>
__global__ void k_w2v_probe_pf(float* __restrict__ in_emb,
                               float* __restrict__ out_emb,
                               const long* __restrict__ in_idx,
                               const int* __restrict__ in_off,
                               const long* __restrict__ out_idx,
                               const float* __restrict__ out_label,
                               const int* __restrict__ out_off,
                               float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (long)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (long)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    float h[DPL], err[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) { h[d] = 0.f; err[d] = 0.f; }
    int ib = in_off[g], ie = in_off[g + 1];
    for (int i = ib; i < ie; ++i) {
      const float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) h[d] += row[c];
      }
    }
    int ob = out_off[g], oe = out_off[g + 1];
    for (int t = ob; t < oe; t += KT) {
      int kn = min(KT, oe - t);
      float* wp[KT];
      float wv[KT][DPL];
      float f[KT];
      // phase 1: issue every row load in the tile (no dependent use between)
#pragma unroll
      for (int k = 0; k < KT; ++k) {
        if (k < kn) {
          wp[k] = out_emb + out_idx[t + k] * dim;
#pragma unroll
          for (int d = 0; d < DPL; ++d) {
            int c = lane + 64 * d;
            wv[k][d] = (c < dim) ? wp[k][c] : 0.f;
          }
        }
      }
      // phase 2: dots + wave reductions
#pragma unroll
      for (int k = 0; k < KT; ++k) {
        f[k] = 0.f;
        if (k < kn) {
#pragma unroll
          for (int d = 0; d < DPL; ++d) f[k] += h[d] * wv[k][d];
        }
      }
#pragma unroll
      for (int s = 32; s; s >>= 1)
#pragma unroll
        for (int k = 0; k < KT; ++k) f[k] += __shfl_xor(f[k], s, 64);
      // phase 3: errors + batched row writes
#pragma unroll
      for (int k = 0; k < KT; ++k) {
        if (k < kn) {
          float e = out_label[t + k] - 1.f / (1.f + expf(-f[k]));
          float el = e * lr;
#pragma unroll
          for (int d = 0; d < DPL; ++d) {
            int c = lane + 64 * d;
            if (c < dim) {
              err[d] += e * wv[k][d];
              wp[k][c] = wv[k][d] + el * h[d];
            }
          }
        }
      }
    }
    for (int i = ib; i < ie; ++i) {
      float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) row[c] += lr * err[d];
      }
    }
  }
}

// VARIANT 6: mirror of the production k_w2v_ns — in-kernel LCG negatives
// from a pool, plain stores; isolates the pool-indirection cost vs V1.
// VARIANT 7: V6 + skip-gram specialization (1 input/group, no in_off).
template <int DPL, bool NIN1>
__global__ void k_w2v_probe_ns(float* __restrict__ in_emb,
                               float* __restrict__ out_emb,
                               const long* __restrict__ in_idx,
                               const int* __restrict__ in_off,
                               const long* __restrict__ centers,
                               const long* __restrict__ pool, long pool_n,
                               int neg, unsigned long long seed,
                               float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (long)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (long)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    float h[DPL], err[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) { h[d] = 0.f; err[d] = 0.f; }
    int ib = NIN1 ? g : in_off[g], ie = NIN1 ? g + 1 : in_off[g + 1];
    for (int i = ib; i < ie; ++i) {
      const float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) h[d] += row[c];
      }
    }
    long pos = centers[g];
    unsigned long long next_random = seed + (unsigned long long)g * 25214903917ull + 11ull;
    for (int o = 0; o <= neg; ++o) {
      long node;
      float label;
      if (o == 0) { node = pos; label = 1.f; }
      else {
        next_random = next_random * 25214903917ull + 11ull;
        node = pool[(long)((next_random >> 8) % (unsigned long long)pool_n)];
        label = 0.f;
        if (node == pos) continue;
      }
      float* w = out_emb + node * dim;
      float wv[DPL];
      float f = 0.f;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        wv[d] = (c < dim) ? w[c] : 0.f;
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
          w[c] = wv[d] + e * lr * h[d];
        }
      }
    }
    for (int i = ib; i < ie; ++i) {
      float* row = in_emb + in_idx[i] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) row[c] += lr * err[d];
      }
    }
  }
}

// VARIANT 8: V7 + depth-1 software pipeline over the output rows: the
// next row's load issues before the current row's dot/update, hiding one
// row-fetch latency per output while keeping VGPRs low (8 waves/SIMD).
template <int DPL>
__global__ void k_w2v_probe_ns_pf(float* __restrict__ in_emb,
                                  float* __restrict__ out_emb,
                                  const long* __restrict__ in_idx,
                                  const int* __restrict__ in_off,
                                  const long* __restrict__ centers,
                                  const long* __restrict__ pool, long pool_n,
                                  int neg, unsigned long long seed,
                                  float lr, int G, int dim) {
  int wid = (int)((blockIdx.x * (long)blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  int nwaves = (int)((gridDim.x * (long)blockDim.x) >> 6);
  for (int g = wid; g < G; g += nwaves) {
    float h[DPL], err[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) { h[d] = 0.f; err[d] = 0.f; }
    {
      const float* row = in_emb + in_idx[g] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) h[d] += row[c];
      }
    }
    long pos = centers[g];
    unsigned long long next_random = seed + (unsigned long long)g * 25214903917ull + 11ull;
    // node sequence is computable ahead of the loop
    long node = pos;
    float label = 1.f;
    float* w = out_emb + node * dim;
    float wv[DPL], nv[DPL];
#pragma unroll
    for (int d = 0; d < DPL; ++d) {
      int c = lane + 64 * d;
      wv[d] = (c < dim) ? w[c] : 0.f;
    }
    for (int o = 0; o <= neg; ++o) {
      // issue next row's loads before using this row's values
      long nnode = -1;
      float nlabel = 0.f;
      int no = o + 1;
      while (no <= neg) {
        next_random = next_random * 25214903917ull + 11ull;
        long cand = pool[(long)((next_random >> 8) % (unsigned long long)pool_n)];
        if (cand != pos) { nnode = cand; break; }
        ++no;
      }
      float* nw = nullptr;
      if (nnode >= 0) {
        nw = out_emb + nnode * dim;
#pragma unroll
        for (int d = 0; d < DPL; ++d) {
          int c = lane + 64 * d;
          nv[d] = (c < dim) ? nw[c] : 0.f;
        }
      }
      float f = 0.f;
#pragma unroll
      for (int d = 0; d < DPL; ++d) f += h[d] * wv[d];
#pragma unroll
      for (int s = 32; s; s >>= 1) f += __shfl_xor(f, s, 64);
      f = 1.f / (1.f + expf(-f));
      float e = label - f;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) {
          err[d] += e * wv[d];
          w[c] = wv[d] + e * lr * h[d];
        }
      }
      if (nnode < 0) break;
      node = nnode; label = nlabel; w = nw;
      o = no - 1;
#pragma unroll
      for (int d = 0; d < DPL; ++d) wv[d] = nv[d];
    }
    {
      float* row = in_emb + in_idx[g] * dim;
#pragma unroll
      for (int d = 0; d < DPL; ++d) {
        int c = lane + 64 * d;
        if (c < dim) row[c] += lr * err[d];
      }
    }
  }
}

int main(int argc, char** argv) {
  int G = argc > 1 ? atoi(argv[1]) : 1 << 21;
  int V = argc > 2 ? atoi(argv[2]) : 1000000;  // vocab
  int dim = argc > 3 ? atoi(argv[3]) : 200;
  int neg = argc > 4 ? atoi(argv[4]) : 5;
  int K = 1 + neg;

  float *in_emb, *out_emb, *labels;
  long *in_idx, *out_idx;
  int *in_off, *out_off;
  hipMalloc(&in_emb, (size_t)V * dim * 4);
  hipMalloc(&out_emb, (size_t)V * dim * 4);
  hipMalloc(&in_idx, (size_t)G * 8);
  hipMalloc(&out_idx, (size_t)G * K * 8);
  hipMalloc(&labels, (size_t)G * K * 4);
  hipMalloc(&in_off, (size_t)(G + 1) * 4);
  hipMalloc(&out_off, (size_t)(G + 1) * 4);

  // host-side Zipf-ish ids
  std::vector<long> h_in(G), h_out((size_t)G * K);
  std::vector<int> h_ioff(G + 1), h_ooff(G + 1);
  std::vector<float> h_lab((size_t)G * K);
  srand(7);
  auto zipf = [&](void) -> long {
    double u = (rand() + 1.0) / (RAND_MAX + 2.0);
    long id = (long)(exp(u * log((double)V))) - 1;
    return id < 0 ? 0 : (id >= V ? V - 1 : id);
  };
  for (int g = 0; g < G; ++g) {
    h_in[g] = zipf();
    h_ioff[g] = g;
    h_ooff[g] = g * K;
    for (int k = 0; k < K; ++k) {
      h_out[(size_t)g * K + k] = zipf();
      h_lab[(size_t)g * K + k] = k == 0 ? 1.f : 0.f;
    }
  }
  h_ioff[G] = G; h_ooff[G] = G * K;
  hipMemcpy(in_idx, h_in.data(), G * 8, hipMemcpyHostToDevice);
  hipMemcpy(out_idx, h_out.data(), (size_t)G * K * 8, hipMemcpyHostToDevice);
  hipMemcpy(labels, h_lab.data(), (size_t)G * K * 4, hipMemcpyHostToDevice);
  hipMemcpy(in_off, h_ioff.data(), (G + 1) * 4, hipMemcpyHostToDevice);
  hipMemcpy(out_off, h_ooff.data(), (G + 1) * 4, hipMemcpyHostToDevice);
  hipMemset(in_emb, 0, (size_t)V * dim * 4);
  hipMemset(out_emb, 0, (size_t)V * dim * 4);

  int grid = (int)std::min((long)MAX_GRID, ((long)G * 64 + BLOCK - 1) / BLOCK);

  auto bench = [&](auto kern, const char* name) {
    // warmup
    kern<<<grid, BLOCK>>>(in_emb, out_emb, in_idx, in_off, out_idx, labels,
                          out_off, 0.025f, G, dim);
    hipDeviceSynchronize();
    hipEvent_t a, b;
    hipEventCreate(&a); hipEventCreate(&b);
    hipEventRecord(a);
    for (int r = 0; r < 3; ++r)
      kern<<<grid, BLOCK>>>(in_emb, out_emb, in_idx, in_off, out_idx, labels,
                            out_off, 0.025f, G, dim);
    hipEventRecord(b);
    hipEventSynchronize(b);
    float ms;
    hipEventElapsedTime(&ms, a, b);
    printf("%s: %.3f ms/launch (G=%d)\n", name, ms / 3, G);
  };

  bench(k_w2v_probe<4, 0>, "V0 scalar-atomic");
  bench(k_w2v_probe<4, 1>, "V1 plain-store  ");
  bench(k_w2v_probe_f2<2>, "V2 float2-loads ");
  bench(k_w2v_probe<4, 3>, "V3 read-only    ");
  bench(k_w2v_probe_pf<4, 8>, "V4 prefetch-8   ");
  bench(k_w2v_probe_pf<4, 6>, "V5 prefetch-6   ");

  // NS-mode variants (pool indirection + in-kernel LCG, like k_w2v_ns)
  long pool_n = 700000;
  long* pool;
  hipMalloc(&pool, pool_n * 8);
  {
    std::vector<long> hp(pool_n);
    for (long i = 0; i < pool_n; ++i) hp[i] = zipf();
    hipMemcpy(pool, hp.data(), pool_n * 8, hipMemcpyHostToDevice);
  }
  auto bench_ns = [&](auto kern, const char* name) {
    kern<<<grid, BLOCK>>>(in_emb, out_emb, h_in.data() ? in_idx : in_idx,
                          in_off, in_idx /*centers = in ids*/, pool, pool_n,
                          neg, 12345ull, 0.025f, G, dim);
    hipDeviceSynchronize();
    hipEvent_t a, b;
    hipEventCreate(&a); hipEventCreate(&b);
    hipEventRecord(a);
    for (int r = 0; r < 3; ++r)
      kern<<<grid, BLOCK>>>(in_emb, out_emb, in_idx, in_off, in_idx, pool,
                            pool_n, neg, 12345ull, 0.025f, G, dim);
    hipEventRecord(b);
    hipEventSynchronize(b);
    float ms;
    hipEventElapsedTime(&ms, a, b);
    printf("%s: %.3f ms/launch (G=%d)\n", name, ms / 3, G);
  };
  bench_ns(k_w2v_probe_ns<4, false>, "V6 ns-pool      ");
  bench_ns(k_w2v_probe_ns<4, true>, "V7 ns-pool-nin1 ");
  bench_ns(k_w2v_probe_ns_pf<4>, "V8 ns-pipelined ");
  return 0;
}
