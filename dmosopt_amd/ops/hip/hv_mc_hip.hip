#include "hip/hip_runtime.h"
// Monte Carlo hypervolume sampling kernels (FPRAS / MCM2RV).
//
// Replaces reference hv_adaptive.py:188-461's sample loops: each thread
// draws its own Philox stream, generates samples, performs the dominance /
// first-containing-box test against the (L2-resident) point set, and the
// block-reduced hit count lands with one atomicAdd. No (samples x N)
// containment matrix is ever materialized.

#include "common.h"
#include <math.h>

// MCM2RV: uniform samples in [ideal, ref]; count samples dominated by any
// point (P[j] <= s for all dims).
__global__ void hv_mc_uniform_kernel(const float* __restrict__ P,  // (N, d)
                                     const float* __restrict__ ideal,  // (d,)
                                     const float* __restrict__ ref,    // (d,)
                                     unsigned long long* __restrict__ hits,
                                     long long n_samples, int N, int d,
                                     unsigned long long seed) {
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  float s[16];  // d <= 16 for MC path (high-d routing caps at 8-10 dims)
  unsigned long long my_hits = 0;
  for (long long i = tid; i < n_samples; i += stride) {
    // d uniforms from ceil(d/4) philox calls
    for (int base = 0; base < d; base += 4) {
      const Philox4 r = philox4x32(seed, (unsigned long long)(i * ((d + 3) / 4) + base / 4));
      const unsigned int w[4] = {r.c0, r.c1, r.c2, r.c3};
      for (int k = 0; k < 4 && base + k < d; ++k)
        s[base + k] = ideal[base + k] +
                      u01(w[k]) * (ref[base + k] - ideal[base + k]);
    }
    bool dominated = false;
    for (int j = 0; j < N && !dominated; ++j) {
      bool dom = true;
      for (int k = 0; k < d; ++k)
        if (P[j * d + k] > s[k]) { dom = false; break; }
      dominated = dom;
    }
    if (dominated) ++my_hits;
  }
  // block reduce then one atomic
  __shared__ unsigned long long partial[256];
  partial[threadIdx.x] = my_hits;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) partial[threadIdx.x] += partial[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(hits, partial[0]);
}

// FPRAS (Karp-Luby): box i chosen by binary search of the volume CDF;
// sample uniform in box [P_i, ref]; count iff i is the FIRST box containing
// the sample.
__global__ void hv_fpras_kernel(const float* __restrict__ P,    // (N, d)
                                const float* __restrict__ ref,  // (d,)
                                const float* __restrict__ cdf,  // (N,) inclusive, last == 1
                                unsigned long long* __restrict__ hits,
                                long long n_samples, int N, int d,
                                unsigned long long seed) {
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  float s[16];
  unsigned long long my_hits = 0;
  const int words = (d + 3) / 4 + 1;  // +1 for the box-choice uniform
  for (long long i = tid; i < n_samples; i += stride) {
    const Philox4 rb = philox4x32(seed ^ 0xABCDULL, (unsigned long long)(i * words));
    const float ub = u01(rb.c0);
    // binary search smallest idx with cdf[idx] >= ub
    int lo = 0, hi_i = N - 1;
    while (lo < hi_i) {
      const int mid = (lo + hi_i) >> 1;
      if (cdf[mid] >= ub) hi_i = mid; else lo = mid + 1;
    }
    const int box = lo;
    for (int base = 0; base < d; base += 4) {
      const Philox4 r =
          philox4x32(seed ^ 0xABCDULL,
                     (unsigned long long)(i * words + 1 + base / 4));
      const unsigned int w[4] = {r.c0, r.c1, r.c2, r.c3};
      for (int k = 0; k < 4 && base + k < d; ++k) {
        const float pl = P[box * d + base + k];
        s[base + k] = pl + u01(w[k]) * (ref[base + k] - pl);
      }
    }
    int first = -1;
    for (int j = 0; j < N; ++j) {
      bool contains = true;
      for (int k = 0; k < d; ++k)
        if (P[j * d + k] > s[k]) { contains = false; break; }
      if (contains) { first = j; break; }
    }
    if (first == box) ++my_hits;
  }
  __shared__ unsigned long long partial[256];
  partial[threadIdx.x] = my_hits;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) partial[threadIdx.x] += partial[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(hits, partial[0]);
}

extern "C" void launch_hv_mc_uniform(const float* P, const float* ideal,
                                     const float* ref,
                                     unsigned long long* hits,
                                     long long n_samples, int N, int d,
                                     unsigned long long seed,
                                     hipStream_t stream) {
  hipLaunchKernelGGL(hv_mc_uniform_kernel, dim3(2048), dim3(256), 0, stream,
                     P, ideal, ref, hits, n_samples, N, d, seed);
}

extern "C" void launch_hv_fpras(const float* P, const float* ref,
                                const float* cdf, unsigned long long* hits,
                                long long n_samples, int N, int d,
                                unsigned long long seed, hipStream_t stream) {
  hipLaunchKernelGGL(hv_fpras_kernel, dim3(2048), dim3(256), 0, stream, P,
                     ref, cdf, hits, n_samples, N, d, seed);
}
